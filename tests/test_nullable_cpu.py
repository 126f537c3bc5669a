"""Null-bearing int64 field columns (Plain cell blocks): the fold skips
nulls — count/sum/min/max exclude them (aggregation.go:310 null check;
cells per convert/number.go:33-46).  Oracle vs direct Python recompute."""
import random

from banyandb_amd import PartBuilder, VT_INT64
from helpers import oracle_blocks, oracle_scan

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def _mk(seed=21, n_blocks=4, n=1500, nil_p=0.25):
    rng = random.Random(seed)
    b = PartBuilder()
    raw = []
    for sid in range(n_blocks):
        ts = [T0 + i * MS for i in range(n)]
        vals = [None if rng.random() < nil_p
                else rng.randint(-10**12, 10**12) for _ in range(n)]
        b.add_block_i64_nullable(sid + 1, ts, [1] * n, vals)
        raw.append(vals)
    return b, raw


def test_encoder_emits_plain_field():
    b, _ = _mk(n_blocks=1)
    payload, blocks = oracle_blocks(b)
    d = blocks[0]
    assert payload[d["col_off"]] == 9  # ENC_PLAIN


def test_oracle_nullable_fold_matches_python():
    b, raw = _mk()
    g = oracle_scan(b, VT_INT64)[0]
    flat = [v for vals in raw for v in vals if v is not None]
    assert g.count == len(flat)
    assert g.sum_i == sum(flat) % 2**64 - (2**64 if sum(flat) % 2**64 >= 2**63 else 0)
    assert g.min_i == min(flat)
    assert g.max_i == max(flat)


def test_oracle_nullable_with_clamp():
    b, raw = _mk(seed=22, n_blocks=2, n=2000)
    lo, hi = T0 + 100 * MS, T0 + 1500 * MS
    g = oracle_scan(b, VT_INT64, min_ts=lo, max_ts=hi)[0]
    flat = [v for vals in raw for v in vals[100:1501] if v is not None]
    assert g.count == len(flat)
    assert g.min_i == min(flat) and g.max_i == max(flat)


def test_all_null_block():
    b = PartBuilder()
    n = 300
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64_nullable(1, ts, [1] * n, [None] * n)
    g = oracle_scan(b, VT_INT64)[0]
    assert g.count == 0


def test_oracle_nullable_f64_matches_python():
    import math
    from banyandb_amd import VT_FLOAT64
    rng = random.Random(41)
    b = PartBuilder()
    raw = []
    for sid in range(3):
        n = 1200
        ts = [T0 + i * MS for i in range(n)]
        vals = [None if rng.random() < 0.3
                else rng.uniform(-1e6, 1e6) for _ in range(n)]
        b.add_block_f64_nullable(sid + 1, ts, [1] * n, vals)
        raw.append(vals)
    g = oracle_scan(b, VT_FLOAT64)[0]
    flat = [v for vals in raw for v in vals if v is not None]
    assert g.count == len(flat)
    assert math.isclose(g.sum_f, sum(flat), rel_tol=1e-12)
    assert g.min_f == min(flat)
    assert g.max_f == max(flat)
