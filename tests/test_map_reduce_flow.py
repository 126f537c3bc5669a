"""End-to-end AggModeMap -> wire frame -> AggModeReduce flow on CPU:
partials from two shards (oracle-computed) are emitted as raw frames
(shard_id + tags + value + MEAN count sidecar, the AggModeMap output
shape — aggregation.go:600-614), decoded on the 'liaison' side, replica-
deduped and combined (aggregation_reduce.go:83-138), and the final values
must equal a direct AggModeAll pass over the union."""
import banyandb_amd as ba
from banyandb_amd.frame import FrameBuilder, FrameReader, ROLE_TAG, ROLE_FIELD
from helpers import oracle_scan

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
N_GROUPS = 4


def shard(rank, seed=0xB4DB):
    b = ba.PartBuilder()
    for s in range(8):
        b.gen_series_i64(s, 10000, T0 + rank * 10000 * MS, MS, s * 1000, 1,
                         seed ^ (rank << 32), group_code=s % N_GROUPS)
    return b


def emit_map_frame(shard_id, parts):
    """AggModeMap emit: shard_id first, tag, value, value__agg_count."""
    fb = FrameBuilder(len(parts))
    fb.add_i64(4, "shard_id", "", [shard_id] * len(parts))
    fb.add_str(ROLE_TAG, "service_id", "meta",
               [f"g{g}".encode() for g in range(len(parts))])
    fb.add_i64(ROLE_FIELD, "value", "", [p.sum_i for p in parts])
    fb.add_i64(ROLE_FIELD, "value__agg_count", "", [p.count for p in parts])
    return fb.finish()


def test_map_frames_reduce_to_all():
    # Map phase on each shard (oracle = the reference semantics)
    frames = []
    for rank in range(2):
        res = oracle_scan(shard(rank), ba.VT_INT64, n_groups=N_GROUPS)
        frames.append(emit_map_frame(rank, res))
        # replica: the same shard emitted twice (dedup must drop it)
        frames.append(emit_map_frame(rank, res))

    # Reduce phase: decode frames, dedup on (shard_id, group_key), Combine
    seen = set()
    acc = {}
    for data in frames:
        r = FrameReader(data)
        shard_ids = r.col_i64(0)
        keys = r.col_var(1)
        sums = r.col_i64(2)
        counts = r.col_i64(3)
        for i in range(r.nrows):
            dk = (shard_ids[i], keys[i])
            if dk in seen:        # replica duplicate -> dropped
                continue
            seen.add(dk)
            st = acc.setdefault(keys[i], [0, 0])
            st[0] = (st[0] + sums[i]) % 2 ** 64
            st[1] += counts[i]

    # AggModeAll over the union
    union = ba.PartBuilder()
    for rank in range(2):
        for s in range(8):
            union.gen_series_i64(s, 10000, T0 + rank * 10000 * MS, MS,
                                 s * 1000, 1, 0xB4DB ^ (rank << 32),
                                 group_code=s % N_GROUPS)
    want = oracle_scan(union, ba.VT_INT64, n_groups=N_GROUPS)
    for g in range(N_GROUPS):
        st = acc[f"g{g}".encode()]
        assert st[1] == want[g].count
        assert st[0] % 2 ** 64 == want[g].sum_i % 2 ** 64
