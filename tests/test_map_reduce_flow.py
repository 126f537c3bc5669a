"""End-to-end AggModeMap -> wire frame -> AggModeReduce flow on CPU:
partials from two shards (oracle-computed) are emitted as raw frames
(shard_id + tags + value + MEAN count sidecar, the AggModeMap output
shape — aggregation.go:600-614), then reduced by the PRODUCT entry point
bydb_reduce_frames (replica dedup + Combine + Val,
aggregation_reduce.go:83-138), and the final values must equal a direct
AggModeAll pass over the union."""
import banyandb_amd as ba
from banyandb_amd.frame import (FrameBuilder, reduce_frames, ROLE_TAG,
                                ROLE_FIELD, ROLE_SHARD)
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
    fb.add_i64(ROLE_SHARD, "shard_id", "", [shard_id] * len(parts))
    fb.add_str(ROLE_TAG, "service_id", "meta",
               [f"g{g}".encode() for g in range(len(parts))])
    fb.add_i64(ROLE_FIELD, "value", "", [p.sum_i for p in parts])
    fb.add_i64(ROLE_FIELD, "value__agg_count", "", [p.count for p in parts])
    return fb.finish()


def test_map_frames_reduce_to_all():
    # Map phase on each shard (oracle = the reference semantics); each
    # shard's frame arrives TWICE (a replica) — dedup must drop the copy.
    frames = []
    for rank in range(2):
        res = oracle_scan(shard(rank), ba.VT_INT64, n_groups=N_GROUPS)
        frames.append(emit_map_frame(rank, res))
        frames.append(emit_map_frame(rank, res))

    # Reduce phase: the product C-ABI does dedup + Combine + Val.
    # specs: SUM over col 2, MEAN over col 2 (count sidecar col 3 located
    # by name), plus plain sum over the count column for the row count.
    out = reduce_frames(frames,
                        specs=[(2, ba.AGG_SUM), (2, ba.AGG_MEAN),
                               (3, ba.AGG_SUM)],
                        key_cols=[1], shard_col=0)

    # AggModeAll over the union
    union = ba.PartBuilder()
    for rank in range(2):
        for s in range(8):
            union.gen_series_i64(s, 10000, T0 + rank * 10000 * MS, MS,
                                 s * 1000, 1, 0xB4DB ^ (rank << 32),
                                 group_code=s % N_GROUPS)
    want = oracle_scan(union, ba.VT_INT64, n_groups=N_GROUPS)
    assert len(out) == N_GROUPS
    for g, (key, vals) in enumerate(out):
        # first-seen order follows the frame row order g0..g3; the key is
        # the packed appendKeyComponent form: present marker + uvarint len
        assert key == b"\x01\x02" + f"g{g}".encode()
        s_sum, s_mean, s_cnt = vals
        assert s_cnt[0] == want[g].count
        assert s_sum[0] % 2 ** 64 == want[g].sum_i % 2 ** 64
        mean = want[g].sum_i // want[g].count
        assert s_mean[0] == max(mean, 1)


def test_reduce_frames_dedup_and_shard_semantics():
    """Same group from DIFFERENT shards combines; same (shard, group)
    drops.  shard_col=-1 falls back to key-only dedup (the reference's
    shardIDIdx == -1 fallback)."""
    def frame(shard_id, sums):
        fb = FrameBuilder(len(sums))
        fb.add_i64(ROLE_SHARD, "shard_id", "", [shard_id] * len(sums))
        fb.add_str(ROLE_TAG, "g", "", [b"k%d" % i for i in range(len(sums))])
        fb.add_i64(ROLE_FIELD, "value", "", sums)
        return fb.finish()

    fa, fb_, fa2 = frame(0, [10, 20]), frame(1, [1, 2]), frame(0, [10, 20])
    out = reduce_frames([fa, fb_, fa2], specs=[(2, ba.AGG_SUM)],
                        key_cols=[1], shard_col=0)
    assert [v[0][0] for _, v in out] == [11, 22]

    # key-only dedup: the second shard's rows now count as replicas
    out2 = reduce_frames([fa, fb_], specs=[(2, ba.AGG_SUM)],
                         key_cols=[1], shard_col=-1)
    assert [v[0][0] for _, v in out2] == [10, 20]


def test_reduce_frames_float_min_max_and_nulls():
    """Float specs combine in the float64 domain; MIN/MAX are
    sentinel-aware (function.go:224-228); null value rows skip
    (combinePartial's IsNull check)."""
    def frame(shard_id, mins, nulls=None):
        fb = FrameBuilder(len(mins))
        fb.add_i64(ROLE_SHARD, "shard_id", "", [shard_id] * len(mins))
        fb.add_str(ROLE_TAG, "g", "", [b"k%d" % i for i in range(len(mins))])
        fb.add_f64(ROLE_FIELD, "value", "", mins, nulls=nulls)
        return fb.finish()

    f0 = frame(0, [5.5, -2.0])
    f1 = frame(1, [3.25, 0.0], nulls=[0, 1])  # group k1's row is null
    out = reduce_frames([f0, f1], specs=[(2, ba.AGG_MIN), (2, ba.AGG_MAX)],
                        key_cols=[1], shard_col=0)
    (k0, v0), (k1, v1) = out
    assert v0[0][1] == 3.25 and v0[1][1] == 5.5
    assert v1[0][1] == -2.0 and v1[1][1] == -2.0  # null row skipped


def test_reduce_frames_edge_cases():
    """Edge contracts of the product reduce: zero frames -> zero groups;
    capacity overflow -> loud BYDB_ERR_OOM; numeric key columns."""
    import pytest as _pytest
    assert reduce_frames([], specs=[(2, ba.AGG_SUM)], key_cols=[1],
                         shard_col=0) == []

    def frame(vals, keys):
        fb = FrameBuilder(len(vals))
        fb.add_i64(ROLE_SHARD, "shard_id", "", [0] * len(vals))
        fb.add_i64(ROLE_TAG, "k", "", keys)        # NUMERIC key column
        fb.add_i64(ROLE_FIELD, "value", "", vals)
        return fb.finish()

    # two rows with the SAME (shard, key) in one frame: the second is a
    # replica duplicate and drops (markDedupSeen — a map shard emits one
    # row per group, so a repeat can only be a replica)
    f = frame([5, 7, 9], [100, -3, 100])
    out = reduce_frames([f], specs=[(2, ba.AGG_SUM)], key_cols=[1],
                        shard_col=0)
    assert [v[0][0] for _, v in out] == [5, 7]

    with _pytest.raises(RuntimeError):
        reduce_frames([f], specs=[(2, ba.AGG_SUM)], key_cols=[1],
                      shard_col=0, out_cap=1)


def test_reduce_frames_property_random():
    """Property: random frames (random shard ids, keys, values, nulls,
    replicas) reduce exactly like a Python model of
    aggregation_reduce.go's dedup + Combine + Val."""
    import random as _r
    for seed in range(30):
        rng = _r.Random(7000 + seed)
        n_frames = rng.randint(1, 5)
        keyspace = [b"k%d" % i for i in range(rng.randint(1, 6))]
        frames, rows = [], []
        for _ in range(n_frames):
            nr = rng.randint(1, 8)
            shards = [rng.randint(0, 2) for _ in range(nr)]
            keys = [keyspace[rng.randrange(len(keyspace))]
                    for _ in range(nr)]
            vals = [rng.randint(-10**9, 10**9) for _ in range(nr)]
            nulls = [1 if rng.random() < 0.2 else 0 for _ in range(nr)]
            fb = FrameBuilder(nr)
            fb.add_i64(ROLE_SHARD, "shard_id", "", shards)
            fb.add_str(ROLE_TAG, "g", "", keys)
            fb.add_i64(ROLE_FIELD, "value", "", vals, nulls=nulls)
            frames.append(fb.finish())
            rows.extend(zip(shards, keys, vals, nulls))
        # model: first (shard,key) wins; null value rows still dedup but
        # skip the Combine (IsNull check happens after markDedupSeen)
        seen, acc, order = set(), {}, []
        for sh, k, v, nu in rows:
            if (sh, k) in seen:
                continue
            seen.add((sh, k))
            if k not in acc:
                acc[k] = [0, None, None]
                order.append(k)
            if not nu:
                st = acc[k]
                st[0] = (st[0] + v) % 2 ** 64
                st[1] = v if st[1] is None else min(st[1], v)
                st[2] = v if st[2] is None else max(st[2], v)
        out = reduce_frames(frames,
                            specs=[(2, ba.AGG_SUM), (2, ba.AGG_MIN),
                                   (2, ba.AGG_MAX)],
                            key_cols=[1], shard_col=0)
        assert len(out) == len(order), seed
        for (key, vals_), k in zip(out, order):
            st = acc[k]
            assert key.endswith(k), (seed, key, k)
            assert vals_[0][0] % 2 ** 64 == st[0], seed
            if st[1] is not None:
                assert vals_[1][0] == st[1] and vals_[2][0] == st[2], seed
            else:
                assert vals_[1][0] == 2 ** 63 - 1, seed   # min sentinel
