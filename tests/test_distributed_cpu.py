"""Multi-shard merge semantics on CPU (gloo, world_size=2): each rank's
time-bucket shard is folded to partials (by the oracle — the merge logic
under test is banyandb_amd.distributed, not the kernels), merged with the
same collective calls the GPU path uses, and compared against the oracle
over the union of both shards."""
import math
import os

import torch
import torch.multiprocessing as mp

import banyandb_amd as ba

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
N_SERIES = 8
N_DP = 20000
N_GROUPS = 4


def _build_shard(rank):
    b = ba.PartBuilder()
    t0_r = T0 + rank * N_DP * MS
    for s in range(N_SERIES):
        b.gen_series_i64(s, N_DP, t0_r, MS, s * 1000, 1, 0xB4DB ^ (rank << 32),
                         group_code=s % N_GROUPS)
    return b


def _oracle_partials(builder):
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "oracle"))
    from helpers import oracle_scan
    res = oracle_scan(builder, ba.VT_INT64, n_groups=N_GROUPS)
    parts = []
    for r in res:
        p = ba.Partial()
        p.sum_i = r.sum_i
        p.count = r.count
        p.min_i = r.min_i
        p.max_i = r.max_i
        p.sum_f = r.sum_f
        parts.append(p)
    return parts


def _worker(rank, world, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29571"
    import torch.distributed as dist
    from banyandb_amd.distributed import (allreduce_partials,
                                          partials_from_structs,
                                          structs_from_partials)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    b = _build_shard(rank)
    parts = _oracle_partials(b)
    t = partials_from_structs(parts)
    allreduce_partials(dist, t, N_GROUPS, need_minmax=True, need_float=False)
    merged = structs_from_partials(t)
    if rank == 0:
        q.put([(p.sum_i, p.count, p.min_i, p.max_i) for p in merged])
    dist.barrier()
    dist.destroy_process_group()


def test_gloo_two_shard_merge():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    merged = q.get(timeout=300)
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0

    # reference: oracle over the union of both shards
    union = ba.PartBuilder()
    for rank in range(2):
        t0_r = T0 + rank * N_DP * MS
        for s in range(N_SERIES):
            union.gen_series_i64(s, N_DP, t0_r, MS, s * 1000, 1,
                                 0xB4DB ^ (rank << 32), group_code=s % N_GROUPS)
    want = _oracle_partials(union)
    for (got, w) in zip(merged, want):
        assert got[0] == w.sum_i
        assert got[1] == w.count
        assert got[2] == w.min_i
        assert got[3] == w.max_i


def test_reduce_partials_finalisation():
    # host-side AggModeReduce combine + Val (incl. MEAN clamp)
    p1 = ba.Partial(sum_i=10, count=4, min_i=-5, max_i=9, sum_f=0.0)
    p2 = ba.Partial(sum_i=-9, count=6, min_i=-50, max_i=2, sum_f=0.0)
    res = ba.reduce_partials([p1, p2], 2, 1, ba.VT_INT64)
    r = res[0]
    assert r.sum_i == 1
    assert r.count == 10
    assert r.min_i == -50
    assert r.max_i == 9
    assert r.mean_i == 1  # 1/10 = 0 -> clamped (function.go:36-39)
