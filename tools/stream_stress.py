"""Stream-ordering stress for the distributed path:
many epochs over tiny parts, 2 ranks (gloo or RCCL) with the REAL
Session — reset / consume / finalize_partials / allreduce raced across
epochs so stream-ordering bugs (collective vs next epoch's reset) show
up as wrong merged partials, checked every epoch against the oracle over
the union of shards.

Launch (1-GPU box, both ranks on device 0):
  BYDB_FORCE_DEVICE=0 BYDB_DIST_BACKEND=gloo \
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 tools/stream_stress.py [epochs]
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))

import torch
import torch.distributed as dist

import banyandb_amd as ba
import oracle as o
from banyandb_amd.distributed import allreduce_partials, partials_from_structs
from helpers import oracle_blocks

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
N_GROUPS = 8


def build_shard(rank, epoch_seed):
    b = ba.PartBuilder()
    for s in range(6):
        b.gen_series_i64(s, 2000 + 37 * (epoch_seed % 5), T0 + rank * 10**10,
                         MS, s * 1000, 1, 0xB4DB ^ (rank << 32) ^ epoch_seed,
                         group_code=s % N_GROUPS)
    return b


def main():
    epochs = int(sys.argv[1]) if len(sys.argv) > 1 else 50
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    device = int(os.environ.get("BYDB_FORCE_DEVICE",
                                os.environ.get("LOCAL_RANK", rank)))
    backend = os.environ.get("BYDB_DIST_BACKEND", "gloo")
    if backend == "nccl":
        torch.cuda.set_device(device)
    dist.init_process_group(backend)

    sess = ba.Session(device)
    part_t = None
    if backend == "nccl":
        part_t = torch.zeros(N_GROUPS * 6, dtype=torch.int64,
                             device=f"cuda:{device}")

    fails = 0
    for e in range(epochs):
        b = build_shard(rank, e)
        sess.upload_part(b)
        sess.configure(ba.VT_INT64,
                       [ba.AGG_SUM, ba.AGG_COUNT, ba.AGG_MIN, ba.AGG_MAX],
                       n_groups=N_GROUPS, mode=ba.MODE_MAP)
        if part_t is not None:
            sess.set_partials_buffer(part_t.data_ptr(), part_t.numel() * 8)
        # two passes per epoch: the SECOND must see a clean reset even
        # though the first epoch's collective may still be in flight
        for _ in range(2):
            sess.reset()
            sess.consume()
            parts = sess.finalize_partials()
            if part_t is not None:
                allreduce_partials(dist, part_t, N_GROUPS, need_minmax=True,
                                   need_float=False)
                torch.cuda.synchronize(device)
                merged = part_t.cpu().view(N_GROUPS, 6)
                got = [(int(merged[g, 0]), int(merged[g, 1]),
                        int(merged[g, 2]), int(merged[g, 3]))
                       for g in range(N_GROUPS)]
            else:
                t = partials_from_structs(parts)
                allreduce_partials(dist, t, N_GROUPS, need_minmax=True,
                                   need_float=False)
                v = t.view(N_GROUPS, 6)
                got = [(int(v[g, 0]), int(v[g, 1]), int(v[g, 2]),
                        int(v[g, 3])) for g in range(N_GROUPS)]
        # oracle over the union of all ranks' shards
        union = ba.PartBuilder()
        for r in range(world):
            for s in range(6):
                union.gen_series_i64(s, 2000 + 37 * (e % 5), T0 + r * 10**10,
                                     MS, s * 1000, 1,
                                     0xB4DB ^ (r << 32) ^ e,
                                     group_code=s % N_GROUPS)
        want = __import__("helpers").oracle_scan(union, ba.VT_INT64,
                                                 n_groups=N_GROUPS)
        for g in range(N_GROUPS):
            w = want[g]
            if got[g] != (w.sum_i, w.count, w.min_i, w.max_i):
                fails += 1
                print(f"rank{rank} epoch {e} group {g}: {got[g]} != "
                      f"({w.sum_i},{w.count},{w.min_i},{w.max_i})",
                      flush=True)
                break
    dist.barrier()
    if rank == 0:
        print(f"STRESS COMPLETE: {epochs} epochs x2 passes, fails={fails}",
              flush=True)
    dist.destroy_process_group()
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
