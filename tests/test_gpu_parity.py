"""GPU parity: HIP decode+fold kernels vs the CPU oracle on identical
encoded parts.  Bit-exact for count/min/max and int64 sums; float64 sum
within rel tolerance 1e-9 (north-star contract: stated tolerance for
float64 sum/avg, everything else exact)."""
import math
import random

import pytest

import oracle as o
from banyandb_amd import (PartBuilder, Session, VT_INT64, VT_FLOAT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, INT64_MIN,
                         INT64_MAX)
from helpers import oracle_scan

pytestmark = pytest.mark.gpu

FULL = dict(min_ts=INT64_MIN, max_ts=INT64_MAX)
T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def make_builder_styles(seed=11):
    """Part with blocks covering every int64 encode type + sizes 1..8192."""
    rng = random.Random(seed)
    b = PartBuilder()
    sid = 1
    for n in (1, 2, 3, 63, 64, 65, 127, 129, 1000, 8191, 8192):
        ts = [T0 + i * MS for i in range(n)]
        vers = [1] * n
        # Const
        b.add_block_i64(sid, ts, vers, [rng.randint(-10**9, 10**9)] * n); sid += 1
        # DeltaConst (needs n>=2 to select; n==1 -> Const)
        start = rng.randint(-10**6, 10**6)
        b.add_block_i64(sid, ts, vers, [start + i * 37 for i in range(n)]); sid += 1
        # Delta (sign-mixed deltas)
        b.add_block_i64(sid, ts, vers,
                        [rng.randint(-10**12, 10**12) for _ in range(n)]); sid += 1
        # DeltaOfDelta (ascending, varying deltas)
        base = rng.randint(0, 10**9)
        vals = []
        v = base
        for _ in range(n):
            v += rng.randint(0, 9)
            vals.append(v)
        b.add_block_i64(sid, ts, vers, vals); sid += 1
        # multi-byte-heavy Delta (large deltas)
        b.add_block_i64(sid, ts, vers,
                        [rng.randint(-2**61, 2**61) for _ in range(n)]); sid += 1
        # mixed-width Delta: 1-byte runs with periodic multi-byte deltas,
        # exercising the 256-byte fast window -> 64-byte slow step handoff
        v, vals = 0, []
        for i in range(n):
            v += rng.randint(-3, 3) if i % 37 else rng.randint(-10**9, 10**9)
            vals.append(v)
        b.add_block_i64(sid, ts, vers, vals); sid += 1
    return b


def assert_int_exact(gpu, orc):
    assert gpu.count == orc.count
    assert gpu.sum_i == orc.sum_i
    assert gpu.min_i == orc.min_i
    assert gpu.max_i == orc.max_i


def test_i64_all_encodings_sum_count():
    b = make_builder_styles()
    orc = oracle_scan(b, VT_INT64)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
    s.consume(**FULL)
    g = s.finalize()[0]
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i
    s.close()


def test_i64_all_encodings_min_max():
    b = make_builder_styles(seed=12)
    orc = oracle_scan(b, VT_INT64)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    s.consume(**FULL)
    g = s.finalize()[0]
    assert_int_exact(g, orc)
    s.close()


def test_i64_time_clamp():
    rng = random.Random(5)
    b = PartBuilder()
    # DeltaConst timestamps at 1ms stride, several series
    for sid in range(8):
        n = 5000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)])
    # irregular (DoD) timestamps
    for sid in range(8, 12):
        n = 3000
        tss = sorted(rng.randint(T0, T0 + 10**10) for _ in range(n))
        # dedupe to keep strictly increasing (duplicates fine too, keep raw)
        b.add_block_i64(sid + 1, tss, [1] * n,
                        [rng.randint(-10**6, 10**6) for _ in range(n)])
    for (lo, hi) in [
        (T0 + 1000 * MS, T0 + 4000 * MS),          # partial clamp
        (T0, T0 + 10 ** 10),                        # full
        (T0 + 10 ** 13, T0 + 10 ** 14),             # no overlap
        (T0 + 777 * MS + 1, T0 + 888 * MS - 1),     # off-grid bounds
    ]:
        orc = oracle_scan(b, VT_INT64, min_ts=lo, max_ts=hi)[0]
        s = Session(0)
        s.upload_part(b)
        s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
        s.consume(min_ts=lo, max_ts=hi)
        g = s.finalize()[0]
        if orc.count == 0:
            assert g.count == 0
        else:
            assert_int_exact(g, orc)
        s.close()


def test_dense_1byte_clamp_boundaries():
    """All-1-byte delta streams (the dense dot4 path) with clamps landing
    at every alignment of the dword regions."""
    rng = random.Random(21)
    b = PartBuilder()
    for sid in range(6):
        n = 5000
        ts = [T0 + i * MS for i in range(n)]
        vals = []
        v = rng.randint(-10**9, 10**9)
        for _ in range(n):
            v += rng.randint(-3, 3)  # deltas always 1 byte
            vals.append(v)
        b.add_block_i64(sid + 1, ts, [1] * n, vals)
    for lo_row, hi_row in [(0, 4999), (1, 4998), (2, 4997), (3, 4996),
                           (4, 4995), (7, 4993), (63, 4930), (64, 4929),
                           (255, 4700), (256, 4699), (257, 4698),
                           (1000, 1000), (4998, 4999), (0, 0)]:
        lo, hi = T0 + lo_row * MS, T0 + hi_row * MS
        orc = oracle_scan(b, VT_INT64, min_ts=lo, max_ts=hi)[0]
        s = Session(0)
        s.upload_part(b)
        s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
        s.consume(min_ts=lo, max_ts=hi)
        g = s.finalize()[0]
        assert g.count == orc.count, (lo_row, hi_row)
        assert g.sum_i == orc.sum_i, (lo_row, hi_row)
        s.close()


def test_f64_min_max_exact_sum_tolerance():
    b = PartBuilder()
    rng = random.Random(6)
    for sid in range(20):
        n = 4096
        ts = [T0 + i * MS for i in range(n)]
        cents = [rng.randint(-10**7, 10**7) for _ in range(n)]
        b.add_block_f64(sid + 1, ts, [1] * n, [c / 100.0 for c in cents])
    orc = oracle_scan(b, VT_FLOAT64)[0]
    exp = b.blocks()[0].exp
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                float_exp=exp)
    s.consume(**FULL)
    g = s.finalize()[0]
    assert g.count == orc.count
    # min/max bit-exact through the monotone decimal-int restore
    assert g.min_f == orc.min_f
    assert g.max_f == orc.max_f
    # sum within stated tolerance (fold order differs from Go's)
    assert math.isclose(g.sum_f, orc.sum_f, rel_tol=1e-9)
    s.close()


def test_mean_clamp_semantics():
    # values averaging below 1 must clamp to 1 (function.go:36-39)
    b = PartBuilder()
    n = 1000
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64(1, ts, [1] * n, [0] * n)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
    s.consume(**FULL)
    g = s.finalize()[0]
    assert g.sum_i == 0 and g.count == n
    assert g.mean_i == 1  # 0/1000 = 0 < 1 -> clamped to 1
    assert g.mean_i == o.mean_val_i64(g.sum_i, g.count)
    s.close()


def test_grouped_aggregate():
    rng = random.Random(9)
    b = PartBuilder()
    n_groups = 16
    for sid in range(64):
        n = 2048
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)],
                        group_code=sid % n_groups)
    orc = oracle_scan(b, VT_INT64, n_groups=n_groups)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                n_groups=n_groups)
    s.consume(**FULL)
    gs = s.finalize()
    for g, oc in zip(gs, orc):
        assert_int_exact(g, oc)
    s.close()


def test_int64_wrapping_sum():
    # Go int64 addition wraps; the GPU u64 path must match bit-for-bit
    b = PartBuilder()
    n = 4000
    ts = [T0 + i * MS for i in range(n)]
    big = 2 ** 62
    vals = [big, big, big, big] * (n // 4)
    b.add_block_i64(1, ts, [1] * n, vals)
    orc = oracle_scan(b, VT_INT64)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
    s.consume(**FULL)
    g = s.finalize()[0]
    assert g.sum_i == orc.sum_i  # wrapped value
    s.close()


def test_epoch_reset_and_repeat():
    b = make_builder_styles(seed=13)
    orc = oracle_scan(b, VT_INT64)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    for _ in range(3):
        s.reset()
        s.consume(**FULL)
        g = s.finalize()[0]
        assert_int_exact(g, orc)
    s.close()
