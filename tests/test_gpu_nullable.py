"""GPU parity: null-bearing int64 field columns (Plain cell blocks,
normalized host-side into validity bitmap + fixed cells; the kernel
decodes sign-flip cells in-lane and skips nulls — aggregation.go:310)."""
import random

import pytest

from banyandb_amd import (PartBuilder, Session, VT_INT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
from helpers import oracle_scan

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
ENVS = [b"prod", b"dev", b"staging", b"qa"]


def run_both(b, funcs, n_groups=1, **kw):
    orc = oracle_scan(b, VT_INT64, n_groups=n_groups, **kw)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, funcs, n_groups=n_groups)
    s.consume(**{k: v for k, v in kw.items() if k != "n_groups"})
    gs = s.finalize()
    s.close()
    return gs, orc


def _nullable_vals(rng, n, nil_p):
    return [None if rng.random() < nil_p
            else rng.randint(-10**12, 10**12) for _ in range(n)]


def test_nullable_parity():
    rng = random.Random(31)
    b = PartBuilder()
    for sid in range(10):
        n = rng.choice([64, 500, 1024, 4096, 8192])
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64_nullable(sid + 1, ts, [1] * n,
                                 _nullable_vals(rng, n, 0.3))
    gs, orc = run_both(b, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    g, oc = gs[0], orc[0]
    assert oc.count > 0
    assert g.count == oc.count
    assert g.sum_i == oc.sum_i
    assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_nullable_mixed_with_dense_blocks():
    """Nullable and normal int-list blocks in one part."""
    rng = random.Random(32)
    b = PartBuilder()
    for sid in range(8):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        if sid % 2 == 0:
            b.add_block_i64_nullable(sid + 1, ts, [1] * n,
                                     _nullable_vals(rng, n, 0.2))
        else:
            b.add_block_i64(sid + 1, ts, [1] * n,
                            [rng.randint(-10**9, 10**9) for _ in range(n)])
    gs, orc = run_both(b, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    g, oc = gs[0], orc[0]
    assert g.count == oc.count and g.sum_i == oc.sum_i
    assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_nullable_with_clamp_and_groups():
    rng = random.Random(33)
    b = PartBuilder()
    for sid in range(6):
        n = 5000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64_nullable(sid + 1, ts, [1] * n,
                                 _nullable_vals(rng, n, 0.4),
                                 group_code=sid % 3)
    lo, hi = T0 + 777 * MS, T0 + 4321 * MS
    gs, orc = run_both(b, [AGG_SUM, AGG_COUNT], n_groups=3,
                       min_ts=lo, max_ts=hi)
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i


def test_nullable_with_predicate():
    rng = random.Random(34)
    b = PartBuilder()
    for sid in range(6):
        n = 4000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64_nullable(sid + 1, ts, [1] * n,
                                 _nullable_vals(rng, n, 0.25))
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 120), n - len(tags))
            tags.extend([ENVS[rng.randrange(4)]] * run)
        b.set_block_tag(tags)
    gs, orc = run_both(b, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                       pred=b"prod")
    g, oc = gs[0], orc[0]
    assert oc.count > 0
    assert g.count == oc.count and g.sum_i == oc.sum_i
    assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_all_null_block_gpu():
    b = PartBuilder()
    n = 200
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64_nullable(1, ts, [1] * n, [None] * n)
    gs, orc = run_both(b, [AGG_SUM, AGG_COUNT])
    assert gs[0].count == 0 == orc[0].count


def test_nullable_f64_parity():
    """Raw IEEE-754 cell blocks: session configured with FLOAT_RAW_EXP;
    double sums, ordered-bit min/max keys."""
    import math
    from banyandb_amd import VT_FLOAT64, FLOAT_RAW_EXP
    rng = random.Random(42)
    b = PartBuilder()
    for sid in range(6):
        n = 2500
        ts = [T0 + i * MS for i in range(n)]
        vals = [None if rng.random() < 0.25
                else rng.uniform(-1e9, 1e9) for _ in range(n)]
        b.add_block_f64_nullable(sid + 1, ts, [1] * n, vals)
    orc = oracle_scan(b, VT_FLOAT64)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                float_exp=FLOAT_RAW_EXP)
    s.consume()
    g = s.finalize()[0]
    s.close()
    assert orc.count > 0
    assert g.count == orc.count
    assert g.min_f == orc.min_f and g.max_f == orc.max_f
    assert math.isclose(g.sum_f, orc.sum_f, rel_tol=1e-9)


def test_nullable_f64_mixed_domain_fails_loud():
    """A decimal float block inside a raw-float session (or vice versa)
    is a loud device error — the min/max domains are incomparable."""
    from banyandb_amd import VT_FLOAT64, FLOAT_RAW_EXP
    b = PartBuilder()
    n = 100
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_f64_nullable(1, ts, [1] * n, [float(i) for i in range(n)])
    b.add_block_f64(2, ts, [1] * n, [i / 100.0 for i in range(1, 2 * n, 2)])
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT], float_exp=FLOAT_RAW_EXP)
    s.consume()
    with pytest.raises(RuntimeError, match="decode error 5"):
        s.finalize()
    s.close()
    s2 = Session(0)
    s2.upload_part(b)
    s2.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT], float_exp=-2)
    s2.consume()
    with pytest.raises(RuntimeError, match="decode error 5"):
        s2.finalize()
    s2.close()


def test_nullable_with_rowvarying_groupby():
    """Null-bearing int64 fields under per-row (RLE) group-by: the group
    run fold counts only valid rows."""
    import oracle as o
    from helpers import oracle_blocks
    rng = random.Random(36)
    b = PartBuilder()
    for sid in range(6):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64_nullable(sid + 1, ts, [1] * n,
                                 _nullable_vals(rng, n, 0.3))
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 100), n - len(tags))
            tags.extend([ENVS[rng.randrange(4)]] * run)
        b.set_block_tag(tags)
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytag(payload, blocks, VT_INT64, 0, ENVS)
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 0,
                       ENVS)
    s.consume()
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
        if oc.count:
            assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_nullable_f64_with_rowvarying_groupby():
    import math
    import oracle as o
    from helpers import oracle_blocks
    from banyandb_amd import VT_FLOAT64, FLOAT_RAW_EXP
    rng = random.Random(37)
    b = PartBuilder()
    for sid in range(5):
        n = 2000
        ts = [T0 + i * MS for i in range(n)]
        vals = [None if rng.random() < 0.2
                else rng.uniform(-1e6, 1e6) for _ in range(n)]
        b.add_block_f64_nullable(sid + 1, ts, [1] * n, vals)
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 80), n - len(tags))
            tags.extend([ENVS[rng.randrange(4)]] * run)
        b.set_block_tag(tags)
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytag(payload, blocks, VT_FLOAT64, 0, ENVS)
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(VT_FLOAT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 0,
                       ENVS, float_exp=FLOAT_RAW_EXP)
    s.consume()
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        if oc.count:
            assert g.min_f == oc.min_f and g.max_f == oc.max_f
            assert math.isclose(g.sum_f, oc.sum_f, rel_tol=1e-9,
                                abs_tol=1e-6)
