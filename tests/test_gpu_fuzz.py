"""Seeded randomized parity sweep — the GPU/oracle analog of the
reference's row-vs-vec topology matrix (vectorized/measure/diff_test.go,
topology_matrix_test.go): random parts over (encode shape x value type x
block size x clamp x predicate x group mode), every scenario asserted
bit-exact (int) / rel-1e-9 (float sum) against the CPU oracle."""
import math
import random

import pytest

import oracle as o
from banyandb_amd import (PartBuilder, Session, VT_INT64, VT_FLOAT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
from helpers import oracle_blocks, oracle_scan

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
TAGS = [b"a", b"bb", b"ccc", b"dd", b"e", b"ffff", b"g", b"hh"]


def f64_sum_close(a, b, count):
    """The engine sums mantissas exactly per block; the oracle (like the
    reference) adds per-row doubles, so under heavy cancellation the
    ORACLE carries the rounding error.  Tolerance scales with the
    magnitude folded (|v| <= ~5.1e9 in these scenarios)."""
    import math as _m
    return _m.isclose(a, b, rel_tol=1e-9,
                      abs_tol=1e-6 + count * 5.1e9 * 1e-13)


def random_values(rng, n, style):
    if style == 0:    # small deltas (all-1-byte) -> dense paths
        v, out = rng.randint(-10**9, 10**9), []
        for _ in range(n):
            v += rng.randint(-3, 3)
            out.append(v)
        return out
    if style == 1:    # 1-2 byte deltas -> 2-byte fast windows
        v, out = rng.randint(-10**6, 10**6), []
        for _ in range(n):
            v += rng.randint(-900, 900)
            out.append(v)
        return out
    if style == 2:    # wide deltas -> multi-byte ballot windows
        return [rng.randint(-2**60, 2**60) for _ in range(n)]
    if style == 3:    # const
        return [rng.randint(-10**9, 10**9)] * n
    if style == 4:    # arithmetic -> DeltaConst
        a, d = rng.randint(-10**9, 10**9), rng.randint(-10**4, 10**4)
        return [a + i * d for i in range(n)]
    # ascending -> DeltaOfDelta
    v, out = rng.randint(0, 10**9), []
    for _ in range(n):
        v += rng.randint(0, 50)
        out.append(v)
    return out


def random_tags(rng, n, kind):
    if kind == 0:
        return None
    if kind == 1:     # entity (uniform)
        return [TAGS[rng.randrange(len(TAGS))]] * n
    tags = []        # row-varying with nils
    while len(tags) < n:
        run = min(rng.randint(1, 1 + rng.randrange(300)), n - len(tags))
        v = None if rng.random() < 0.15 else TAGS[rng.randrange(len(TAGS))]
        tags.extend([v] * run)
    return tags


def build_scenario(rng, n_groups=1):
    is_float = rng.random() < 0.3
    n_blocks = rng.randint(2, 24)
    tag_kind = rng.randrange(3)
    b = PartBuilder()
    for sid in range(n_blocks):
        n = rng.choice([1, 2, 7, 63, 64, 65, 500, 1023, 1024, 1025, 4096,
                        8191, 8192])
        ts = [T0 + i * MS for i in range(n)]
        if is_float:
            cents = random_values(rng, n, rng.choice([0, 1, 2]))
            cents = [c % 10**12 - 5 * 10**11 for c in cents]
            b.add_block_f64(sid + 1, ts, [1] * n, [c / 100.0 for c in cents],
                            group_code=sid % n_groups)
        else:
            b.add_block_i64(sid + 1, ts, [1] * n,
                            random_values(rng, n, rng.randrange(6)),
                            group_code=sid % n_groups)
        t = random_tags(rng, n, tag_kind)
        if t is not None:
            b.set_block_tag(t)
    return b, is_float, tag_kind


def check_scalar(rng, b, is_float, tag_kind, sess_dev=0):
    vtype = VT_FLOAT64 if is_float else VT_INT64
    # mixed per-block exponents rescale against the part MINIMUM exponent
    exp = min(d.exp for d in b.blocks()) if is_float else 0
    # random clamp
    mode = rng.randrange(3)
    kw = {}
    if mode == 1:
        lo = T0 + rng.randint(0, 2000) * MS
        hi = lo + rng.randint(0, 8192) * MS
        kw = dict(min_ts=lo, max_ts=hi)
    pred = None
    if tag_kind and rng.random() < 0.6:
        pred = TAGS[rng.randrange(len(TAGS))]
    orc = oracle_scan(b, vtype, pred=pred or b"", **kw)[0]
    s = Session(sess_dev)
    s.upload_part(b)
    s.configure(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], float_exp=exp)
    s.consume(pred=pred or b"", **kw)
    g = s.finalize()[0]
    s.close()
    assert g.count == orc.count, "count"
    if is_float:
        if orc.count:
            assert g.min_f == orc.min_f and g.max_f == orc.max_f
            assert f64_sum_close(g.sum_f, orc.sum_f, orc.count)
    else:
        assert g.sum_i == orc.sum_i
        if orc.count:
            assert g.min_i == orc.min_i and g.max_i == orc.max_i


def check_bytag(rng, b, is_float):
    vtype = VT_FLOAT64 if is_float else VT_INT64
    exp = min(d.exp for d in b.blocks()) if is_float else 0
    domain = list(TAGS[: rng.randint(2, len(TAGS))])
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytag(payload, blocks, vtype, 0, domain)
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 0,
                       domain, float_exp=exp)
    s.consume()
    gs = s.finalize()
    s.close()
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        if is_float:
            if oc.count:
                assert g.min_f == oc.min_f and g.max_f == oc.max_f
                assert f64_sum_close(g.sum_f, oc.sum_f, oc.count)
        else:
            assert g.sum_i == oc.sum_i
            if oc.count:
                assert g.min_i == oc.min_i and g.max_i == oc.max_i


@pytest.mark.parametrize("seed", range(20))
def test_fuzz_scenario(seed):
    rng = random.Random(0xF0 + seed)
    b, is_float, tag_kind = build_scenario(rng)
    check_scalar(rng, b, is_float, tag_kind)
    if tag_kind:
        check_bytag(rng, b, is_float)


@pytest.mark.parametrize("seed", range(10))
def test_fuzz_grouped_by_code(seed):
    rng = random.Random(0x1F0 + seed)
    b, is_float, _ = build_scenario(rng, n_groups=3)
    vtype = VT_FLOAT64 if is_float else VT_INT64
    exp = min(d.exp for d in b.blocks()) if is_float else 0
    orc = oracle_scan(b, vtype, n_groups=3)
    s = Session(0)
    s.upload_part(b)
    s.configure(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], n_groups=3,
                float_exp=exp)
    s.consume()
    gs = s.finalize()
    s.close()
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        if is_float:
            if oc.count:
                assert g.min_f == oc.min_f and g.max_f == oc.max_f
                assert f64_sum_close(g.sum_f, oc.sum_f, oc.count)
        else:
            assert g.sum_i == oc.sum_i
            if oc.count:
                assert g.min_i == oc.min_i and g.max_i == oc.max_i


@pytest.mark.parametrize("seed", [1128, 1392, 1710, 1800, 2084, 3018])
def test_sweep_regression_seeds(seed):
    """Seeds the 5000-scenario sweep caught: wide (6-byte) varints
    straddling window boundaries produced multi-byte carries that the
    64-B fast window once accepted with a single 7-bit shift; plus
    float-sum cancellation tolerance cases."""
    rng = random.Random(0xABC000 + seed)
    b, is_float, tag_kind = build_scenario(rng)
    check_scalar(rng, b, is_float, tag_kind)
    if tag_kind:
        check_bytag(rng, b, is_float)


def test_group_code_out_of_range_fails_loud():
    b = PartBuilder()
    ts = [T0 + i * MS for i in range(100)]
    b.add_block_i64(1, ts, [1] * 100, list(range(100)), group_code=5)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT], n_groups=1)
    s.consume()
    try:
        s.finalize()
        raise AssertionError("out-of-range group_code must be a loud error")
    except RuntimeError as e:
        assert "decode error 4" in str(e)
    finally:
        s.close()
