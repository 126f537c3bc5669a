"""End-to-end GPU query flow: encoded part in HBM -> scan+fold kernel ->
AggModeMap partials -> Reduce combine -> BatchTop selection -> raw
columnar wire frame (the bytes a data node would ship,
frame/encode.go:42-80) -> decode and verify against the oracle."""
import random

import pytest

import ctypes as C

import banyandb_amd as ba
from banyandb_amd import frame as fr
from helpers import oracle_blocks


def _top_by_sum(results, k):
    lib = ba.lib()
    lib.bydb_top_groups.restype = C.c_int
    lib.bydb_top_groups.argtypes = [C.POINTER(ba.Result), C.c_int64, C.c_int,
                                    C.c_int64, C.c_int,
                                    C.POINTER(C.c_int64),
                                    C.POINTER(C.c_int64)]
    n = len(results)
    arr = (ba.Result * n)(*results)
    out = (C.c_int64 * n)()
    out_n = C.c_int64()
    rc = lib.bydb_top_groups(arr, n, 0, k, 0, out, C.byref(out_n))
    assert rc == 0
    return list(out[: out_n.value])

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
SVCS = [b"svc_%d" % i for i in range(8)]


def test_scan_reduce_top_frame_roundtrip():
    import oracle as o
    rng = random.Random(91)
    b = ba.PartBuilder()
    for sid in range(16):
        n = 3000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(0, 10**6) for _ in range(n)],
                        group_code=sid % 8)
    payload, blocks = oracle_blocks(b)
    orc = o.scan_agg(payload, blocks, ba.VT_INT64, n_groups=8)

    # two shards (AggModeMap partials per shard), merged with Reduce
    s = ba.Session(0)
    s.upload_part(b)
    s.configure(ba.VT_INT64, [ba.AGG_SUM, ba.AGG_COUNT], n_groups=8)
    mid = T0 + 1499 * MS
    s.consume(max_ts=mid)
    shard0 = s.finalize_partials()
    s.reset()
    s.consume(min_ts=mid + MS)
    shard1 = s.finalize_partials()
    s.close()
    flat = list(shard0) + list(shard1)
    merged = ba.reduce_partials(flat, 2, 8, ba.VT_INT64)

    for g, oc in zip(merged, orc):
        assert g.count == oc.count
        assert g.sum_i == oc.sum_i

    # BatchTop: top-3 groups by sum desc
    top = _top_by_sum(merged, 3)
    exp = sorted(range(8), key=lambda i: (-orc[i].sum_i, i))[:3]
    assert list(top) == exp

    # emit the winning rows as the reference wire frame and read it back
    fb = fr.FrameBuilder(3)
    fb.add_str(2, "service", "default", [SVCS[i] for i in top])
    fb.add_i64(6, "sum", "", [merged[i].sum_i for i in top])
    fb.add_i64(6, "count", "", [merged[i].count for i in top])
    blob = fb.finish()
    rd = fr.FrameReader(blob)
    assert rd.nrows == 3 and rd.ncols == 3
    assert rd.col_var(0) == [SVCS[i] for i in top]
    assert rd.col_i64(1) == [merged[i].sum_i for i in top]
    assert rd.col_i64(2) == [merged[i].count for i in top]
