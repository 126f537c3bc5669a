"""GPU parity for the tag-equality predicate (dictionary codes) against the
CPU oracle — config-3 semantics."""
import math
import random

import pytest

import oracle as o
from banyandb_amd import (PartBuilder, Session, VT_INT64, VT_FLOAT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
from helpers import oracle_scan

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
ENVS = [b"prod", b"dev", b"staging", b"qa"]


def entity_tag_part(rng, n_series=12, n=3000, float_vals=False):
    """Entity tag: constant per series (the uniform fast path)."""
    b = PartBuilder()
    for sid in range(n_series):
        ts = [T0 + i * MS for i in range(n)]
        if float_vals:
            cents = [rng.randint(-10**6, 10**6) for _ in range(n)]
            b.add_block_f64(sid + 1, ts, [1] * n, [c / 100.0 for c in cents])
        else:
            b.add_block_i64(sid + 1, ts, [1] * n,
                            [rng.randint(-10**9, 10**9) for _ in range(n)])
        b.set_block_tag([ENVS[sid % 4]] * n)
    return b


def rowvary_tag_part(rng, n_series=8, n=2000):
    """Row-varying tag: multi-run RLE inside each block."""
    b = PartBuilder()
    for sid in range(n_series):
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)])
        tags = []
        i = 0
        while len(tags) < n:
            run = min(rng.randint(1, 200), n - len(tags))
            v = ENVS[rng.randrange(4)] if rng.random() > 0.1 else None
            tags.extend([v] * run)
            i += 1
        b.set_block_tag(tags)
    return b


def run_both(b, vtype, pred, funcs=(AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX),
             float_exp=0):
    orc = oracle_scan(b, vtype, pred=pred)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(vtype, list(funcs), float_exp=float_exp)
    s.consume(pred=pred)
    g = s.finalize()[0]
    s.close()
    return g, orc


def test_entity_tag_predicate_i64():
    rng = random.Random(31)
    b = entity_tag_part(rng)
    for pred in (b"prod", b"dev", b"qa", b"absent"):
        g, orc = run_both(b, VT_INT64, pred)
        assert g.count == orc.count, pred
        assert g.sum_i == orc.sum_i, pred
        if orc.count:
            assert g.min_i == orc.min_i and g.max_i == orc.max_i, pred


def test_rowvarying_tag_predicate_i64():
    rng = random.Random(32)
    b = rowvary_tag_part(rng)
    for pred in (b"prod", b"staging", b"nope"):
        g, orc = run_both(b, VT_INT64, pred)
        assert g.count == orc.count, pred
        assert g.sum_i == orc.sum_i, pred
        if orc.count:
            assert g.min_i == orc.min_i and g.max_i == orc.max_i, pred


def test_predicate_with_const_and_deltaconst_blocks():
    rng = random.Random(33)
    b = PartBuilder()
    n = 1500
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64(1, ts, [1] * n, [42] * n)           # Const
    b.set_block_tag([ENVS[i % 3] for i in range(n)])    # row-varying
    b.add_block_i64(2, ts, [1] * n, [7 * i for i in range(n)])  # DeltaConst
    b.set_block_tag([ENVS[(i // 10) % 4] for i in range(n)])
    b.add_block_i64(3, ts, [1] * n, [rng.randint(0, 100) for _ in range(n)])
    b.set_block_tag([b"prod"] * n)                      # uniform
    for pred in (b"prod", b"dev"):
        g, orc = run_both(b, VT_INT64, pred)
        assert g.count == orc.count, pred
        assert g.sum_i == orc.sum_i, pred
        if orc.count:
            assert g.min_i == orc.min_i and g.max_i == orc.max_i, pred


def test_predicate_block_without_tag_excluded():
    rng = random.Random(34)
    b = PartBuilder()
    n = 1000
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64(1, ts, [1] * n, [rng.randint(0, 99) for _ in range(n)])
    b.set_block_tag([b"prod"] * n)
    b.add_block_i64(2, ts, [1] * n, [10**6] * n)  # NO tag -> excluded by pred
    g, orc = run_both(b, VT_INT64, b"prod")
    assert g.count == orc.count == n
    assert g.sum_i == orc.sum_i
    assert g.max_i == orc.max_i < 100


def test_config3_float_minmax_avg_with_predicate():
    """SURVEY config 3 shape: float64 min/max/avg + one tag equality."""
    rng = random.Random(35)
    b = entity_tag_part(rng, n_series=16, n=4096, float_vals=True)
    exp = b.blocks()[0].exp
    g, orc = run_both(b, VT_FLOAT64, b"prod", float_exp=exp)
    assert g.count == orc.count
    assert g.min_f == orc.min_f
    assert g.max_f == orc.max_f
    assert math.isclose(g.sum_f, orc.sum_f, rel_tol=1e-9)
    # avg (MEAN) via sum/count with the >=1 clamp
    want_mean = o.mean_val_f64(orc.sum_f, float(orc.count))
    assert math.isclose(g.mean_f, want_mean, rel_tol=1e-9)


def test_config5_three_tag_conjunctive():
    """SURVEY config 5 shape: 3-tag conjunctive filter (dict cards 4/16/256)
    + group-by, mixed selectivity."""
    rng = random.Random(41)
    b = PartBuilder()
    n_groups = 32
    regions = [f"r{i}".encode() for i in range(16)]
    svcs = [f"svc{i}".encode() for i in range(256)]
    for sid in range(48):
        n = 2000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)],
                        group_code=sid % n_groups)
        b.set_block_tag([ENVS[sid % 4]] * n)                  # env: entity
        b.set_block_tag([regions[(sid // 4) % 16]] * n)       # region: entity
        # svc: row-varying runs; global card 256, but each block draws from
        # a small subset so the per-block dict values stay in a plain
        # (<128 B) compress_block — the device-parseable form (zstd'd
        # dictionaries are a host-side resolve fallback, later round)
        subset = [svcs[(sid * 7 + k) % 256] for k in range(8)] + [b"svc7"]
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 120), n - len(tags))
            tags.extend([subset[rng.randrange(len(subset))]] * run)
        b.set_block_tag(tags)
    preds = [b"prod", b"r2", b"svc7"]
    orc = oracle_scan(b, VT_INT64, preds=preds, n_groups=n_groups)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                n_groups=n_groups)
    s.consume(preds=preds)
    gs = s.finalize()
    s.close()
    total = sum(oc.count for oc in orc)
    assert total > 0, "predicate selected nothing; bad test setup"
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
        if oc.count:
            assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_two_tag_conjunctive_rowvarying():
    rng = random.Random(42)
    b = PartBuilder()
    n = 3000
    ts = [T0 + i * MS for i in range(n)]
    for sid in range(6):
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 10**6) for _ in range(n)])
        t1, t2 = [], []
        while len(t1) < n:
            run = min(rng.randint(1, 97), n - len(t1))
            t1.extend([ENVS[rng.randrange(4)]] * run)
        while len(t2) < n:
            run = min(rng.randint(1, 53), n - len(t2))
            t2.extend([None if rng.random() < 0.2 else
                       f"z{rng.randrange(8)}".encode()] * run)
        b.set_block_tag(t1)
        b.set_block_tag(t2)
    preds = [b"dev", b"z3"]
    orc = oracle_scan(b, VT_INT64, preds=preds)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    s.consume(preds=preds)
    g = s.finalize()[0]
    s.close()
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i
    if orc.count:
        assert g.min_i == orc.min_i and g.max_i == orc.max_i


def test_predicate_with_time_clamp():
    rng = random.Random(36)
    b = rowvary_tag_part(rng, n_series=4, n=3000)
    lo, hi = T0 + 500 * MS, T0 + 2500 * MS
    orc = oracle_scan(b, VT_INT64, pred=b"prod", min_ts=lo, max_ts=hi)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    s.consume(min_ts=lo, max_ts=hi, pred=b"prod")
    g = s.finalize()[0]
    s.close()
    assert g.count == orc.count
    assert g.sum_i == orc.sum_i
    assert g.min_i == orc.min_i and g.max_i == orc.max_i
