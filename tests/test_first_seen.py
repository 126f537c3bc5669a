"""Group first-seen materialisation order (bydb_group_first_seen):
the reference creates groups in row-iteration order (computeKey,
vectorized/measure/aggregation.go:523) and NextBatch emits them in that
order; the engine's host-domain results reorder to it by sorting on the
per-group (item, row) first-entry keys."""
import pytest

from banyandb_amd import (PartBuilder, Session, VT_INT64, AGG_SUM,
                          AGG_COUNT)

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def test_first_seen_order_row_varying_and_uniform():
    b = PartBuilder()
    # block 0: runs d(rows 0-49), a(50-99); block 1: c then d;
    # block 2: uniform b.  Expected first-seen: d, a, c, b
    specs = [[b"d", b"a"], [b"c", b"d"], [b"b", b"b"]]
    n = 100
    for i, (v0, v1) in enumerate(specs):
        ts = [T0 + k * MS for k in range(n)]
        b.add_block_i64(i + 1, ts, [1] * n, list(range(n)))
        b.set_block_tag([v0] * 50 + [v1] * 50)
    domain = [b"a", b"b", b"c", b"d", b"never"]
    s = Session(0)
    s.upload_part(b)
    s.configure_by_tag(VT_INT64, [AGG_SUM, AGG_COUNT], 0, domain)
    s.consume()
    order = s.groups_in_first_seen_order()
    gs = s.finalize()
    s.close()
    assert order == [3, 0, 2, 1]   # d, a, c, b; "never" absent
    assert gs[4].count == 0


def test_first_seen_order_block_group_codes():
    b = PartBuilder()
    n = 64
    # blocks in storage order carry group codes 2, 0, 2, 1
    for i, code in enumerate([2, 0, 2, 1]):
        ts = [T0 + k * MS for k in range(n)]
        b.add_block_i64(i + 1, ts, [1] * n, list(range(n)),
                        group_code=code)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT], n_groups=4)
    s.consume()
    order = s.groups_in_first_seen_order()
    s.finalize()
    s.close()
    assert order == [2, 0, 1]   # group 3 never entered


def test_first_seen_stable_across_epochs():
    b = PartBuilder()
    n = 256
    for i, code in enumerate([1, 0]):
        ts = [T0 + k * MS for k in range(n)]
        b.add_block_i64(i + 1, ts, [1] * n, list(range(n)),
                        group_code=code)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM], n_groups=2)
    for _ in range(3):
        s.reset()
        s.consume()
        assert s.groups_in_first_seen_order() == [1, 0]
        s.finalize()
    s.close()
