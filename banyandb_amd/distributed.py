"""Multi-GPU partial merge — the reference's Map->Reduce aggregation
(AggModeMap partial emit + AggModeReduce combine, pkg/query/vectorized/
measure/aggregation.go:57-64, aggregation_reduce.go:120-138) mapped onto
RCCL collectives over xGMI.

Partials are a dense [n_groups x 6] int64 tensor with the bydb_partial
layout: (sum_i, count, min_i, max_i, sum_f-as-bits, pad).  Dense
fixed-cardinality group-bys reduce slot-wise: SUM for sums/counts, MIN/MAX
for extrema, SUM on the float64 view for mantissa sums.  MEAN is finalised
only after the reduce (sum/count, never mean-of-means — the reference's
"Efficient MEAN" doc rule).
"""
import torch


def partials_tensor(n_groups, device):
    """Allocate the accumulation buffer bydb_set_partials_buffer expects."""
    return torch.zeros(n_groups * 6, dtype=torch.int64, device=device)


def allreduce_partials(dist, part_t, n_groups, need_minmax=False,
                       need_float=False):
    """Merge partials across ranks in place (all_reduce — every rank ends
    with the combined partial; AggModeReduce.Combine semantics)."""
    view = part_t.view(n_groups, 6)
    sums = view[:, 0:2].contiguous()
    dist.all_reduce(sums, op=dist.ReduceOp.SUM)
    view[:, 0:2] = sums
    if need_minmax:
        mins = view[:, 2].contiguous()
        dist.all_reduce(mins, op=dist.ReduceOp.MIN)
        view[:, 2] = mins
        maxs = view[:, 3].contiguous()
        dist.all_reduce(maxs, op=dist.ReduceOp.MAX)
        view[:, 3] = maxs
    if need_float:
        sf = view[:, 4].contiguous().view(torch.float64)
        dist.all_reduce(sf, op=dist.ReduceOp.SUM)
        view[:, 4] = sf.view(torch.int64)
    return part_t


def partials_from_structs(parts, device="cpu"):
    """Pack a list of ctypes Partial structs into the dense tensor."""
    import ctypes
    n = len(parts)
    t = torch.empty(n * 6, dtype=torch.int64, device="cpu")
    for i, p in enumerate(parts):
        t[i * 6 + 0] = p.sum_i
        t[i * 6 + 1] = p.count
        t[i * 6 + 2] = p.min_i
        t[i * 6 + 3] = p.max_i
        t[i * 6 + 4] = torch.tensor(p.sum_f, dtype=torch.float64).view(torch.int64)
        t[i * 6 + 5] = 0
    return t.to(device)


def structs_from_partials(part_t):
    """Unpack the dense tensor back into Partial structs."""
    from banyandb_amd import Partial
    cpu = part_t.detach().to("cpu").view(-1, 6)
    out = []
    for row in cpu:
        p = Partial()
        p.sum_i = int(row[0])
        p.count = int(row[1])
        p.min_i = int(row[2])
        p.max_i = int(row[3])
        p.sum_f = float(row[4].view(torch.float64))
        out.append(p)
    return out
