"""GPU parity: per-row group-by on a dictionary tag vs the CPU oracle
(computeKey semantics over tag values, aggregation.go:523; the host
supplies the group domain order)."""
import random

import pytest

import oracle as o
from banyandb_amd import (PartBuilder, Session, VT_INT64, VT_FLOAT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
from helpers import oracle_blocks

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
ENVS = [b"prod", b"dev", b"staging", b"qa"]


def run_both(b, vtype, slot, domain, funcs, min_ts=None, max_ts=None,
             float_exp=0):
    payload, blocks = oracle_blocks(b)
    kw = {}
    if min_ts is not None:
        kw = dict(min_ts=min_ts, max_ts=max_ts)
    orc = o.scan_agg_bytag(payload, blocks, vtype, slot, domain, **kw)
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(vtype, funcs, slot, domain, float_exp=float_exp)
    if min_ts is not None:
        s.consume(min_ts=min_ts, max_ts=max_ts)
    else:
        s.consume()
    gs = s.finalize()
    s.close()
    return gs, orc


def test_entity_tag_grouping():
    """Uniform (entity) tags: resolves to the fast per-block path."""
    rng = random.Random(51)
    b = PartBuilder()
    for sid in range(16):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)])
        b.set_block_tag([ENVS[sid % 4]] * n)
    gs, orc = run_both(b, VT_INT64, 0, ENVS,
                       [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
        if oc.count:
            assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_rowvarying_tag_grouping():
    """Row-varying tags: per-RLE-run folds."""
    rng = random.Random(52)
    b = PartBuilder()
    for sid in range(8):
        n = 4000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)])
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 150), n - len(tags))
            v = None if rng.random() < 0.1 else ENVS[rng.randrange(4)]
            tags.extend([v] * run)
        b.set_block_tag(tags)
    gs, orc = run_both(b, VT_INT64, 0, ENVS,
                       [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX])
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
        if oc.count:
            assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_rowvarying_grouping_with_clamp():
    rng = random.Random(53)
    b = PartBuilder()
    n = 5000
    for sid in range(4):
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 10**6) for _ in range(n)])
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 80), n - len(tags))
            tags.extend([ENVS[rng.randrange(4)]] * run)
        b.set_block_tag(tags)
    lo, hi = T0 + 333 * MS, T0 + 4444 * MS
    gs, orc = run_both(b, VT_INT64, 0, ENVS, [AGG_SUM, AGG_COUNT],
                       min_ts=lo, max_ts=hi)
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i


def test_grouping_on_second_slot_and_subset_domain():
    """Group by tag slot 1; domain omits one value -> those rows drop."""
    rng = random.Random(54)
    b = PartBuilder()
    regions = [b"r0", b"r1", b"r2"]
    for sid in range(6):
        n = 2000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 999) for _ in range(n)])
        b.set_block_tag([ENVS[sid % 4]] * n)             # slot 0
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 60), n - len(tags))
            tags.extend([regions[rng.randrange(3)]] * run)
        b.set_block_tag(tags)                            # slot 1
    domain = [b"r0", b"r2"]  # r1 rows are dropped
    gs, orc = run_both(b, VT_INT64, 1, domain, [AGG_SUM, AGG_COUNT])
    total = sum(oc.count for oc in orc)
    assert 0 < total < 6 * 2000
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i


def test_float_grouping():
    rng = random.Random(55)
    b = PartBuilder()
    for sid in range(8):
        n = 2048
        ts = [T0 + i * MS for i in range(n)]
        cents = [rng.randint(-10**6, 10**6) for _ in range(n)]
        b.add_block_f64(sid + 1, ts, [1] * n, [c / 100.0 for c in cents])
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 100), n - len(tags))
            tags.extend([ENVS[rng.randrange(4)]] * run)
        b.set_block_tag(tags)
    exp = b.blocks()[0].exp
    gs, orc = run_both(b, VT_FLOAT64, 0, ENVS,
                       [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], float_exp=exp)
    import math
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        if oc.count:
            assert g.min_f == oc.min_f and g.max_f == oc.max_f
            assert math.isclose(g.sum_f, oc.sum_f, rel_tol=1e-9)


def test_composite_two_tag_grouping():
    """Composite (env, region) key: gid = env + 4*region."""
    rng = random.Random(61)
    regions = [b"r0", b"r1", b"r2"]
    b = PartBuilder()
    for sid in range(10):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**6, 10**6) for _ in range(n)])
        # slot 0: row-varying env; slot 1: entity region
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 120), n - len(tags))
            tags.extend([ENVS[rng.randrange(4)]] * run)
        b.set_block_tag(tags)
        b.set_block_tag([regions[sid % 3]] * n)
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0, 1],
                            [ENVS, regions])
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tags(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                        [0, 1], [ENVS, regions])
    s.consume()
    gs = s.finalize()
    s.close()
    assert len(gs) == 12
    assert sum(oc.count for oc in orc) == 10 * 3000
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
        if oc.count:
            assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_composite_both_rowvarying():
    rng = random.Random(62)
    regions = [b"x", b"yy"]
    b = PartBuilder()
    for sid in range(6):
        n = 2500
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 9999) for _ in range(n)])
        for table, maxrun in ((ENVS, 90), (regions, 130)):
            tags = []
            while len(tags) < n:
                run = min(rng.randint(1, maxrun), n - len(tags))
                v = None if rng.random() < 0.08 else table[rng.randrange(len(table))]
                tags.extend([v] * run)
            b.set_block_tag(tags)
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0, 1],
                            [ENVS, regions])
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tags(VT_INT64, [AGG_SUM, AGG_COUNT], [0, 1],
                        [ENVS, regions])
    s.consume()
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i


def test_rowvarying_groups_with_rowvarying_predicate():
    """Row-varying group-by slot + row-varying predicate slot: the
    predicate's RLE runs join the group-run merge as extra cursors."""
    rng = random.Random(63)
    regions = [b"r0", b"r1", b"r2", b"r3"]
    b = PartBuilder()
    for sid in range(8):
        n = 4000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)])
        tags = []
        while len(tags) < n:
            run = min(rng.randint(1, 90), n - len(tags))
            v = None if rng.random() < 0.05 else ENVS[rng.randrange(4)]
            tags.extend([v] * run)
        b.set_block_tag(tags)                    # slot 0: grouped
        tags2 = []
        while len(tags2) < n:
            run = min(rng.randint(1, 140), n - len(tags2))
            tags2.extend([regions[rng.randrange(4)]] * run)
        b.set_block_tag(tags2)                   # slot 1: predicated
    import oracle as o
    from helpers import oracle_blocks
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0], [ENVS],
                            preds=[b"", b"r2", b""])
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 0,
                       ENVS)
    s.consume(preds=[b"", b"r2", b""])
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
        if oc.count:
            assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_composite_groups_with_predicate_and_clamp():
    rng = random.Random(64)
    regions = [b"x", b"yy"]
    svcs = [b"s0", b"s1", b"s2"]
    b = PartBuilder()
    for sid in range(6):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 99999) for _ in range(n)])
        for table, maxrun in ((ENVS, 70), (regions, 110)):
            tags = []
            while len(tags) < n:
                run = min(rng.randint(1, maxrun), n - len(tags))
                tags.extend([table[rng.randrange(len(table))]] * run)
            b.set_block_tag(tags)
        tags3 = []
        while len(tags3) < n:
            run = min(rng.randint(1, 60), n - len(tags3))
            tags3.extend([svcs[rng.randrange(3)]] * run)
        b.set_block_tag(tags3)                   # slot 2: predicated
    import oracle as o
    from helpers import oracle_blocks
    payload, blocks = oracle_blocks(b)
    lo, hi = T0 + 200 * MS, T0 + 2600 * MS
    orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0, 1],
                            [ENVS, regions], min_ts=lo, max_ts=hi,
                            preds=[b"", b"", b"s1"])
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tags(VT_INT64, [AGG_SUM, AGG_COUNT], [0, 1],
                        [ENVS, regions])
    s.consume(min_ts=lo, max_ts=hi, preds=[b"", b"", b"s1"])
    gs = s.finalize()
    s.close()
    assert sum(oc.count for oc in orc) > 0
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i
