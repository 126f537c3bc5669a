"""GPU parity: residual predicate pushdown on plain (non-dictionary) tag
columns.  The >256-distinct dictionary bail
(column.go:266-278, dictionary.go:58) stores a zstd'd bytes block; the
host normalizes it at part registration (where the reference decompresses,
zstd.go:49) and k_resolve_plain evaluates the per-row equality into a
match bitmap consumed by the fold walkers."""
import random

import pytest

from banyandb_amd import (PartBuilder, Session, VT_INT64, VT_FLOAT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
from helpers import oracle_scan

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
ENVS = [b"prod", b"dev", b"staging", b"qa"]


def _users(rng, n, card=300, nil_p=0.0):
    out = []
    for _ in range(n):
        if nil_p and rng.random() < nil_p:
            out.append(None)
        else:
            out.append(b"user_%03d" % rng.randrange(card))
    return out


def run_both(b, vtype, funcs, float_exp=0, **consume_kw):
    orc = oracle_scan(b, vtype, **{k: v for k, v in consume_kw.items()})[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(vtype, funcs, float_exp=float_exp)
    s.consume(**consume_kw)
    g = s.finalize()[0]
    s.close()
    return g, orc


def test_plain_tag_predicate():
    rng = random.Random(71)
    b = PartBuilder()
    for sid in range(8):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)])
        b.set_block_tag(_users(rng, n))
    g, orc = run_both(b, VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                      pred=b"user_042")
    assert orc.count > 0
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i
    assert g.min_i == orc.min_i and g.max_i == orc.max_i


def test_plain_tag_predicate_with_nils_and_clamp():
    rng = random.Random(72)
    b = PartBuilder()
    for sid in range(6):
        n = 4000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 10**6) for _ in range(n)])
        b.set_block_tag(_users(rng, n, nil_p=0.15))
    lo, hi = T0 + 500 * MS, T0 + 3500 * MS
    g, orc = run_both(b, VT_INT64, [AGG_SUM, AGG_COUNT], pred=b"user_007",
                      min_ts=lo, max_ts=hi)
    assert orc.count > 0
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i


def test_plain_and_dict_conjunction():
    """Slot 0 plain (card 300), slot 1 dictionary (4 envs) — the walkers
    mix bitmap mode and RLE mode in one fold."""
    rng = random.Random(73)
    b = PartBuilder()
    for sid in range(6):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**6, 10**6) for _ in range(n)])
        b.set_block_tag(_users(rng, n))                       # slot 0: plain
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 100), n - len(tags))
            tags.extend([ENVS[rng.randrange(4)]] * run)
        b.set_block_tag(tags)                                 # slot 1: dict
    g, orc = run_both(b, VT_INT64, [AGG_SUM, AGG_COUNT],
                      preds=[b"user_100", b"prod"])
    assert orc.count > 0
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i


def test_mixed_plain_and_dict_blocks_same_slot():
    """Some blocks dictionary-encode (low card), others bail to plain —
    per-block mode selection within one consume."""
    rng = random.Random(74)
    b = PartBuilder()
    for sid in range(8):
        n = 2500
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**6, 10**6) for _ in range(n)])
        if sid % 2 == 0:
            b.set_block_tag(_users(rng, n))               # plain (card 300)
        else:
            # low-card: the same value namespace but dictionary-encodable
            b.set_block_tag([b"user_%03d" % rng.randrange(8) for _ in range(n)])
    g, orc = run_both(b, VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                      pred=b"user_003")
    assert orc.count > 0
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i
    assert g.min_i == orc.min_i and g.max_i == orc.max_i


def test_plain_tag_float_column():
    rng = random.Random(75)
    b = PartBuilder()
    for sid in range(5):
        n = 2048
        ts = [T0 + i * MS for i in range(n)]
        cents = [rng.randint(-10**6, 10**6) for _ in range(n)]
        b.add_block_f64(sid + 1, ts, [1] * n, [c / 100.0 for c in cents])
        b.set_block_tag(_users(rng, n))
    exp = b.blocks()[0].exp
    g, orc = run_both(b, VT_FLOAT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                      float_exp=exp, pred=b"user_111")
    import math
    assert orc.count > 0
    assert g.count == orc.count
    assert g.min_f == orc.min_f and g.max_f == orc.max_f
    assert math.isclose(g.sum_f, orc.sum_f, rel_tol=1e-9)


def test_plain_groupby_partial_domain():
    """Group-by on a plain column (landed round 2 —
    k_resolve_plain_groups): a two-value domain folds exactly the rows
    carrying those values; everything else drops.  Parity vs the
    oracle's bytags over the same domain."""
    import oracle as o
    from helpers import oracle_blocks
    rng = random.Random(76)
    b = PartBuilder()
    n = 3000
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64(1, ts, [1] * n, list(range(n)))
    b.set_block_tag(_users(rng, n))
    domain = [b"user_001", b"user_002"]
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0], [domain])
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(VT_INT64, [AGG_SUM, AGG_COUNT], 0, domain)
    s.consume()
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert (g.count, g.sum_i) == (oc.count, oc.sum_i)
