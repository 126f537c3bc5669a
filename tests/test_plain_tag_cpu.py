"""Residual predicate pushdown on plain (non-dictionary) tag columns —
SURVEY §8(f)4: when a tag column has >256 distinct values the dictionary
encode bails (dictionary.go:58) and the reference stores a plain bytes
block (column.go:266-278).  CPU side: the product encoder must emit the
plain form, and the oracle's predicate path must match a direct Python
recompute on the raw rows."""
import random

from banyandb_amd import PartBuilder, VT_INT64, lib
from helpers import oracle_blocks, oracle_scan

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def _mk_part(n_blocks=5, n=2000, card=300, nil_p=0.0, seed=7):
    rng = random.Random(seed)
    b = PartBuilder()
    raw_tags = []
    raw_vals = []
    for sid in range(n_blocks):
        ts = [T0 + i * MS for i in range(n)]
        vals = [rng.randint(-10**9, 10**9) for _ in range(n)]
        tags = []
        for _ in range(n):
            if nil_p and rng.random() < nil_p:
                tags.append(None)
            else:
                tags.append(b"user_%03d" % rng.randrange(card))
        b.add_block_i64(sid + 1, ts, [1] * n, vals)
        b.set_block_tag(tags)
        raw_tags.append(tags)
        raw_vals.append(vals)
    return b, raw_tags, raw_vals


def test_encoder_emits_plain_beyond_256_values():
    b, _, _ = _mk_part(n_blocks=1)
    payload, blocks = oracle_blocks(b)
    d = blocks[0]
    # ENC_PLAIN = 9 leads the stream; a low-cardinality column would be 10
    assert payload[d["tag_off"]] == 9


def test_oracle_plain_predicate_matches_python():
    b, raw_tags, raw_vals = _mk_part()
    pred = b"user_123"
    exp_sum, exp_cnt = 0, 0
    for tags, vals in zip(raw_tags, raw_vals):
        for t, v in zip(tags, vals):
            if t == pred:
                exp_sum += v
                exp_cnt += 1
    assert exp_cnt > 0
    g = oracle_scan(b, VT_INT64, pred=pred)[0]
    assert g.count == exp_cnt
    assert g.sum_i == exp_sum


def test_oracle_plain_predicate_with_nils():
    b, raw_tags, raw_vals = _mk_part(nil_p=0.2, seed=11)
    pred = b"user_007"
    exp_cnt = sum(t == pred for tags in raw_tags for t in tags)
    assert exp_cnt > 0
    g = oracle_scan(b, VT_INT64, pred=pred)[0]
    assert g.count == exp_cnt


def test_normalize_plain_tag_roundtrip():
    """bydb_part_append's host normalization is exercised on the GPU; here
    pin the stream format itself: lens block + payload decompress to the
    raw rows (checked through the oracle's independent decoder)."""
    b, raw_tags, _ = _mk_part(n_blocks=1, n=2000, card=290, nil_p=0.1, seed=3)
    payload, blocks = oracle_blocks(b)
    d = blocks[0]
    stream = payload[d["tag_off"]: d["tag_off"] + d["tag_len"]]
    assert stream[0] == 9
    import oracle as o
    rows = o.bytes_block_decode(stream[1:], 2000)
    assert rows == raw_tags[0]
