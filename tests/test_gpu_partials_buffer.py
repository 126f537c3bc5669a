"""External (torch CUDA) partials buffer: the RCCL-merge path accumulates
into a caller-owned device tensor (bydb_set_partials_buffer); results
must match the session-owned buffer, and MEAN finalisation applies the
>=1 clamp (meanFunc.Val, function.go:30-45)."""
import random

import pytest

import banyandb_amd as ba
from banyandb_amd import (PartBuilder, Session, VT_INT64,
                         AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def _part(n_groups):
    rng = random.Random(99)
    b = PartBuilder()
    for sid in range(12):
        n = 4000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(sid + 1, ts, [1] * n,
                        [rng.randint(-10**9, 10**9) for _ in range(n)],
                        group_code=sid % n_groups)
    return b


def test_external_torch_partials_matches_internal():
    import torch
    n_groups = 4
    b = _part(n_groups)
    # internal buffer
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                n_groups=n_groups)
    s.consume()
    ref = s.finalize_partials()
    s.close()
    # external torch CUDA tensor
    from banyandb_amd.distributed import (partials_tensor,
                                          structs_from_partials)
    t = partials_tensor(n_groups, "cuda:0")
    s2 = Session(0)
    s2.upload_part(b)
    s2.configure(VT_INT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                 n_groups=n_groups)
    s2.set_partials_buffer(t.data_ptr(), t.numel() * 8)
    s2.reset()
    s2.consume()
    got = s2.finalize_partials()
    torch.cuda.synchronize(0)
    tensor_view = structs_from_partials(t)
    s2.close()
    for r, g, tv in zip(ref, got, tensor_view):
        assert g.count == r.count == tv.count
        assert g.sum_i == r.sum_i == tv.sum_i
        assert g.min_i == r.min_i == tv.min_i
        assert g.max_i == r.max_i == tv.max_i


def test_mean_clamp_on_gpu_results():
    """MEAN = sum/count with the >=1 clamp (function.go:30-45)."""
    b = PartBuilder()
    n = 100
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64(1, ts, [1] * n, [0] * n)        # mean 0 -> clamps to 1
    b.add_block_i64(2, ts, [1] * n, [500] * n, group_code=1)
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT], n_groups=2)
    s.consume()
    gs = s.finalize()
    s.close()
    assert gs[0].count == n and gs[0].sum_i == 0 and gs[0].mean_i == 1
    assert gs[1].mean_i == 500
