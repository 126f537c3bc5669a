"""GPU parity: dictionary tag columns whose compress_block sections are
zstd'd (>=128 B, bytes.go:291-303) — e.g. a full 256-value per-block
dictionary.  The host decompresses them at part registration into the
sidecar arena with raw section framing; predicates and group-by then run
on device exactly as for small plain-framed dictionaries."""
import random

import pytest

import oracle as o
from banyandb_amd import (PartBuilder, Session, VT_INT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
from helpers import oracle_blocks, oracle_scan

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
SVCS = [b"service_%03d" % i for i in range(256)]  # 256 x 11B >> 128B section


def _svc_tags(rng, n, card=256, maxrun=40):
    tags = []
    while len(tags) < n:
        run = min(rng.randint(1, maxrun), n - len(tags))
        tags.extend([SVCS[rng.randrange(card)]] * run)
    return tags


def test_bigdict_stream_is_zstd_compressed():
    rng = random.Random(81)
    b = PartBuilder()
    n = 4000
    b.add_block_i64(1, [T0 + i * MS for i in range(n)], [1] * n,
                    [rng.randint(0, 999) for _ in range(n)])
    b.set_block_tag(_svc_tags(rng, n))
    payload, blocks = oracle_blocks(b)
    d = blocks[0]
    stream = payload[d["tag_off"]: d["tag_off"] + d["tag_len"]]
    assert stream[0] == 10  # ENC_DICTIONARY
    # after the varuint count, the lengths section must be zstd (marker 1)
    p = 1
    while stream[p] & 0x80:
        p += 1
    p += 1
    assert stream[p] == 1


def test_bigdict_predicate():
    rng = random.Random(82)
    b = PartBuilder()
    for sid in range(8):
        n = 5000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)])
        b.set_block_tag(_svc_tags(rng, n))
    g = None
    orc = oracle_scan(b, VT_INT64, pred=b"service_123")[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    s.consume(pred=b"service_123")
    g = s.finalize()[0]
    s.close()
    assert orc.count > 0
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i
    assert g.min_i == orc.min_i and g.max_i == orc.max_i


def test_bigdict_groupby():
    """Group-by over a 64-value domain on a row-varying 256-value
    dictionary with zstd'd sections."""
    rng = random.Random(83)
    b = PartBuilder()
    for sid in range(6):
        n = 4096
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**6, 10**6) for _ in range(n)])
        b.set_block_tag(_svc_tags(rng, n))
    domain = SVCS[:64]
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytag(payload, blocks, VT_INT64, 0, domain)
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 0,
                       domain)
    s.consume()
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
        if oc.count:
            assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_bigdict_composite_with_small_dict():
    """Composite key: slot 0 a zstd'd 256-value dictionary (domain 32),
    slot 1 a small plain-framed dictionary — run-merge across a sidecar
    cursor and a payload cursor."""
    rng = random.Random(84)
    envs = [b"prod", b"dev"]
    b = PartBuilder()
    for sid in range(6):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 9999) for _ in range(n)])
        b.set_block_tag(_svc_tags(rng, n))
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 90), n - len(tags))
            tags.extend([envs[rng.randrange(2)]] * run)
        b.set_block_tag(tags)
    domain0 = SVCS[:32]
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0, 1],
                            [domain0, envs])
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tags(VT_INT64, [AGG_SUM, AGG_COUNT], [0, 1],
                        [domain0, envs])
    s.consume()
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
