#!/usr/bin/env python3
"""bench.py — BanyanDB measure scan+aggregate throughput on MI355X.

Workload (BASELINE.json configs[1], the headline single-GPU config):
10k series x 1M int64 datapoints (10^10 dp), sum+count aggregate, no
predicate, whole time range.  Encoded measure blocks (the reference's own
on-disk stream format) are resident in HBM before the timed region; one
step = one pass of the decode+fold hot path over the whole resident part.

N>1 (launched by the driver via torch.distributed.run): weak scaling —
each rank owns one time-bucket shard of the same shape (SURVEY section 8e,
storage/tsdb.go:61 segment model); the group-partial exchange is an RCCL
reduce over the partials tensor.  value = datapoints all ranks processed
per second.

CPU baseline: the CPU oracle (restated reference Go path, 'port') timed on
this box's host cores over a bounded sample of the same workload — a
reported baseline, not the target.
"""
import argparse
import ctypes
import hashlib
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

T0 = 1_700_000_000_000_000_000  # T0 = 1.7e18 ns
STRIDE = 10 ** 6                # 1 ms
SEED = 0xB4DB
ALGO_BYTES_PER_DP = 16          # 8B value + 8B timestamp
HBM_PEAK_GBS = 8000.0           # MI355X_MICROARCH.md: HBM3E 8.0 TB/s spec


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(f"[bench] {msg}", file=sys.stderr, flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--series", type=int,
                    default=int(os.environ.get("BYDB_BENCH_SERIES", 10000)))
    ap.add_argument("--dp", type=int,
                    default=int(os.environ.get("BYDB_BENCH_DP", 1_000_000)))
    ap.add_argument("--skip-cpu-baseline", action="store_true",
                    default=os.environ.get("BYDB_SKIP_CPU_BASELINE") == "1")
    ap.add_argument("--workload", default="i64_sum",
                    choices=["i64_sum", "f64_pred", "group256",
                             "group4096_pred3", "mixed"],
                    help="i64_sum = BASELINE configs[1] (default, the "
                         "headline metric); f64_pred = configs[2] shape; "
                         "group256 = configs[3] shape; group4096_pred3 = "
                         "configs[4] shape (int64 leg); mixed = "
                         "configs[5] shape (half int64 + half float64 "
                         "series, two overlapped sessions)")
    args = ap.parse_args()

    import torch
    import banyandb_amd as ba

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    # test hooks: BYDB_FORCE_DEVICE maps every rank onto one GPU and
    # BYDB_DIST_BACKEND=gloo swaps the merge transport, so the multi-rank
    # code path can be smoke-tested on a single-GPU box
    device = int(os.environ.get("BYDB_FORCE_DEVICE", local_rank))
    backend = os.environ.get("BYDB_DIST_BACKEND", "nccl")
    if args.gpus > 1 and world == 1:
        log("WARNING: --gpus > 1 without torchrun; measuring 1 rank")
    n_gpus = world if world > 1 else 1
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        if backend == "nccl":
            torch.cuda.set_device(device)
        dist.init_process_group(backend)

    n_series, n_dp = args.series, args.dp
    full_config = n_series == 10000 and n_dp == 1_000_000
    # per-workload parameters
    ENVS = [b"prod", b"dev", b"staging", b"qa"]
    REGIONS = [f"r{i}".encode() for i in range(16)]
    SVCS = [f"s{i}".encode() for i in range(256)]
    W = {
        "i64_sum": dict(float=False, n_groups=1, group_mod=0, tags=[],
                        preds=[], funcs=["sum", "count"], algo_bpd=16,
                        dtype="int64", agg="sum+count", enc_bpd=1.05),
        "f64_pred": dict(float=True, n_groups=1, group_mod=0, tags=[ENVS],
                         preds=[b"prod"], funcs=["sum", "count", "min", "max"],
                         algo_bpd=17, dtype="f64", agg="min+max+avg",
                         enc_bpd=2.0),
        "group256": dict(float=False, n_groups=256, group_mod=256, tags=[],
                         preds=[], funcs=["sum", "count"], algo_bpd=16,
                         dtype="int64", agg="sum+count groupby(256)",
                         enc_bpd=1.05),
        "group4096_pred3": dict(float=False, n_groups=4096, group_mod=4096,
                                tags=[ENVS, REGIONS, SVCS],
                                preds=[b"prod", b"r4", b"s4"],
                                funcs=["sum", "count"], algo_bpd=19,
                                dtype="int64",
                                agg="sum+count groupby(4096) 3-tag filter",
                                enc_bpd=1.05),
        # configs[5]: mixed int64/float64 fields — half the series carry an
        # int64 field, half a float64 field; two sessions (one per value
        # type, separate HIP streams, overlapped) fold into separate
        # per-group partials, mirroring one column each of the reference's
        # multi-field MeasureBatch
        "mixed": dict(float="mixed", n_groups=4096, group_mod=4096,
                      tags=[ENVS, REGIONS, SVCS],
                      preds=[b"prod", b"r4", b"s4"],
                      funcs=["sum", "count"], algo_bpd=19,
                      dtype="int64+f64",
                      agg="sum+count groupby(4096) 3-tag filter mixed",
                      enc_bpd=1.6),
    }[args.workload]
    base_name = {"i64_sum": "int64_sum_count", "f64_pred": "f64_minmaxavg_pred",
                 "group256": "int64_group256",
                 "group4096_pred3": "int64_group4096_pred3",
                 "mixed": "mixed_group4096_pred3"}[args.workload]
    workload = (f"10k_series_x_1M_{base_name}" if full_config
                else f"{n_series}_series_x_{n_dp}_{base_name}")
    if args.workload == "i64_sum" and full_config:
        workload = "10k_series_x_1M_int64_sum_count"

    # ---- build + upload the rank's shard (untimed) ----
    threads = max(1, (os.cpu_count() or 8) // max(world, 1))
    t0_r = T0 + rank * n_dp * STRIDE
    seed_r = SEED ^ (rank << 32)
    total_dp_rank = n_series * n_dp
    tag_overhead = 64 * len(W["tags"])  # per-block uniform dict columns
    est_payload = int(total_dp_rank * W["enc_bpd"]) + n_series * (64 + tag_overhead) \
        + n_series * ((n_dp + 8191) // 8192) * tag_overhead
    est_blocks = n_series * ((n_dp + 8191) // 8192)

    mixed = W["float"] == "mixed"
    sess = ba.Session(device)
    sess.reserve(est_payload if not mixed else est_payload // 2 + 1024,
                 est_blocks if not mixed else est_blocks // 2 + 1)
    sess2 = None
    if mixed:
        sess2 = ba.Session(device)
        sess2.reserve(est_payload, est_blocks // 2 + 1)

    gen_t = time.perf_counter()
    chunk = 500
    b = ba.PartBuilder()
    for slot, table in enumerate(W["tags"]):
        b.set_tag_table(slot, table)
    uploaded_dp = 0
    half = n_series // 2
    # mixed: a separate builder per session — a builder's running base
    # offset must track its own session's payload arena
    b2 = None
    if mixed:
        b2 = ba.PartBuilder()
        for slot, table in enumerate(W["tags"]):
            b2.set_tag_table(slot, table)
    s0 = 0
    while s0 < n_series:
        lim = half if (mixed and s0 < half) else n_series
        ns = min(chunk, lim - s0)
        use_float = W["float"] is True or (mixed and s0 >= half)
        bb = b2 if (mixed and s0 >= half) else b
        if use_float:
            bb.gen_bulk_f64(s0, ns, n_dp, t0_r, STRIDE, 10.0, 0.01, seed_r,
                            group_mod=W["group_mod"], threads=threads)
        else:
            bb.gen_bulk_i64(s0, ns, n_dp, t0_r, STRIDE, 1000, 1, seed_r,
                            group_mod=W["group_mod"], threads=threads)
        (sess2 if (mixed and s0 >= half) else sess).append(bb)
        uploaded_dp += ns * n_dp
        bb.drain()
        s0 += ns
    gen_s = time.perf_counter() - gen_t
    log(f"rank {rank}: generated+uploaded {uploaded_dp} dp in {gen_s:.1f}s")

    n_groups = W["n_groups"]
    fmap = {"sum": ba.AGG_SUM, "count": ba.AGG_COUNT, "min": ba.AGG_MIN,
            "max": ba.AGG_MAX}
    sess.configure(ba.VT_FLOAT64 if W["float"] is True else ba.VT_INT64,
                   [fmap[f] for f in W["funcs"]], n_groups=n_groups,
                   float_exp=-2 if W["float"] is True else 0)
    if mixed:
        sess2.configure(ba.VT_FLOAT64, [fmap[f] for f in W["funcs"]],
                        n_groups=n_groups, float_exp=-2)

    # partials live in a torch CUDA tensor so the merge is RCCL over xGMI
    part_t = part_t2 = None
    if world > 1 and backend == "nccl":
        part_t = torch.zeros(n_groups * 6, dtype=torch.int64,
                             device=f"cuda:{device}")
        sess.set_partials_buffer(part_t.data_ptr(), part_t.numel() * 8)
        if mixed:
            part_t2 = torch.zeros(n_groups * 6, dtype=torch.int64,
                                  device=f"cuda:{device}")
            sess2.set_partials_buffer(part_t2.data_ptr(),
                                      part_t2.numel() * 8)

    preds = W["preds"] or None
    need_minmax = "min" in W["funcs"] or "max" in W["funcs"]

    def one_step():
        sess.reset()
        if mixed:
            sess2.reset()
        # consume is asynchronous: with two sessions the int64 and float64
        # scans overlap on their own HIP streams
        sess.consume(min_ts=t0_r, max_ts=t0_r + n_dp * STRIDE, preds=preds)
        if mixed:
            sess2.consume(min_ts=t0_r, max_ts=t0_r + n_dp * STRIDE,
                          preds=preds)
        if world > 1:
            # sync the session stream, then merge the partials
            # (AggModeReduce Combine semantics): RCCL over xGMI on the
            # device tensor, or gloo over host copies in the smoke config
            parts = sess.finalize_partials()
            from banyandb_amd.distributed import (allreduce_partials,
                                                  partials_from_structs)
            if backend == "nccl":
                allreduce_partials(dist, part_t, n_groups,
                                   need_minmax=need_minmax,
                                   need_float=W["float"])
                # the collective runs on torch's stream; the next step's
                # reset() runs on the session stream — order them
                torch.cuda.synchronize(device)
            else:
                t = partials_from_structs(parts)
                allreduce_partials(dist, t, n_groups,
                                   need_minmax=need_minmax,
                                   need_float=W["float"])
            if mixed:
                parts2 = sess2.finalize_partials()
                if backend == "nccl":
                    allreduce_partials(dist, part_t2, n_groups,
                                       need_minmax=need_minmax,
                                       need_float=True)
                    torch.cuda.synchronize(device)
                else:
                    t2 = partials_from_structs(parts2)
                    allreduce_partials(dist, t2, n_groups,
                                       need_minmax=need_minmax,
                                       need_float=True)
                return parts, parts2
            return parts
        if mixed:
            return sess.finalize_partials(), sess2.finalize_partials()
        return sess.finalize_partials()

    # ---- warmup ----
    for _ in range(args.warmup):
        one_step()
    # ---- timed region ----
    launch_ms = []
    if world > 1:
        dist.barrier()
    torch.cuda.synchronize(device) if torch.cuda.is_available() else None
    t_start = time.perf_counter()
    for _ in range(args.steps):
        one_step()
        lm = sess.last_consume_ms()
        if mixed:
            # overlapped streams: the conservative roofline denominator is
            # the SUM of the two launch durations
            lm += sess2.last_consume_ms()
        launch_ms.append(lm)
    torch.cuda.synchronize(device) if torch.cuda.is_available() else None
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t_start
    if world > 1:
        dev = f"cuda:{device}" if backend == "nccl" else "cpu"
        t = torch.tensor([elapsed], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # ---- correctness spot-check (outside the timed region) ----
    if world == 1:
        parts = sess.finalize_partials()
        total_count = sum(p.count for p in parts)
        if mixed:
            total_count += sum(p.count for p in sess2.finalize_partials())
        if not W["preds"]:
            assert total_count == total_dp_rank, \
                f"count {total_count} != {total_dp_rank}"
        else:
            assert 0 < total_count < total_dp_rank, total_count

    ms_per_step = elapsed * 1000.0 / args.steps
    total_dp = total_dp_rank * max(world, 1)
    value = total_dp * args.steps / elapsed

    avg_launch_ms = sum(launch_ms) / len(launch_ms)
    algo_bytes_per_launch = W["algo_bpd"] * total_dp_rank
    achieved_gbs = algo_bytes_per_launch / (avg_launch_ms / 1e3) / 1e9

    # PMC-measured actual HBM bytes per launch, keyed by workload.  The
    # table is only trusted when its kernel-source stamp matches the
    # kernels that are actually running — a changed kernel without a
    # re-profile must NOT report stale traffic (it silently mislabels the
    # roofline fraction).
    traffic = None
    traffic_stale = False
    tpath = os.path.join(REPO, "profiles", "pmc_traffic.json")
    if os.path.exists(tpath):
        try:
            tj = json.load(open(tpath))
            if tj.get("kernels_sha16") != kernels_sha16():
                traffic_stale = True
            else:
                traffic = (tj.get("workloads") or {}).get(workload)
        except Exception:
            pass

    cpu_baseline = None
    if (rank == 0 and world == 1 and not args.skip_cpu_baseline
            and args.workload == "i64_sum"):
        cpu_baseline = run_cpu_baseline(n_dp)

    if rank == 0:
        out = {
            "metric": "datapoints/sec scanned+aggregated",
            "value": value,
            "unit": "datapoints/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": W["dtype"],
            "data": "synthetic",
            "config": {
                "workload": workload,
                "series_per_shard": n_series,
                "datapoints_per_series": n_dp,
                "agg": W["agg"],
                "parallelism": f"time-bucket shards x{max(world,1)}",
            },
            # achieved/frac are ACTUAL HBM bytes moved (rocprofv3 PMC
            # FETCH_SIZE/WRITE_SIZE, gfx950-calibrated, hash-stamped table)
            # over the measured launch duration — frac <= 1 by
            # construction.  The input streams are compressed (~1-2 B/dp),
            # so SURVEY 8d's ALGORITHMIC 16-19 B/dp figure exceeds the HBM
            # peak; it is reported separately as algo_*.  When no fresh
            # PMC measurement exists for these exact kernels, achieved and
            # frac are null (never stale, never algorithmic).
            "roofline": {
                "bound": "hbm",
                "achieved": (traffic / (avg_launch_ms / 1e3) / 1e9)
                            if traffic else None,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": (traffic / (avg_launch_ms / 1e3) / 1e9
                         / HBM_PEAK_GBS) if traffic else None,
                "traffic": traffic,
                "basis": "pmc-actual-bytes" if traffic else
                         ("pmc-table-stale-for-these-kernels" if traffic_stale
                          else "pmc-table-missing"),
                "algo_achieved_gbs": achieved_gbs,
                "algo_frac": achieved_gbs / HBM_PEAK_GBS,
                "kernel_avg_launch_ms": avg_launch_ms,
            },
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(out), flush=True)

    if world > 1:
        dist.destroy_process_group()
    sess.close()
    if mixed:
        sess2.close()


def fast_oracle_part(builder):
    """Vectorised bridge from the product part builder to the oracle's
    (payload, BlockDesc[]) form, for the CPU-baseline leg at full scale.

    Same transformation as tests/helpers.oracle_blocks — the oracle reads
    the on-disk column payload WITH its [type][firstValue-cell] header
    (column.go:183-213), which the product parses off at load — but
    bulk-built with numpy instead of a per-block Python loop (1.2M blocks
    at the headline config).  int64 non-nullable columns only (the bench
    workload).  Returns (payload_buffer, oracle_desc_array, n_blocks)."""
    import ctypes as ct
    import numpy as np
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import banyandb_amd as ba
    import oracle as o
    nb = builder.n_blocks
    plen = builder.payload_len
    pdt = np.dtype([("series_id", "<u8"), ("count", "<u4"),
                    ("ts_enc_wv", "u1"), ("version_enc", "u1"),
                    ("field_enc", "u1"), ("field_vtype", "u1"),
                    ("ts_min", "<i8"), ("ts_max", "<i8"),
                    ("version_first", "<i8"), ("field_first", "<i8"),
                    ("exp", "<i2"), ("_pad", "V6"),
                    ("ts_off", "<u8"), ("ts_len", "<u8"),
                    ("field_off", "<u8"), ("field_len", "<u8"),
                    ("tag_off", "<u8"), ("tag_len", "<u8"),
                    ("tag2_off", "<u8"), ("tag2_len", "<u8"),
                    ("tag3_off", "<u8"), ("tag3_len", "<u8"),
                    ("group_code", "<u4"), ("_pad2", "<u4")])
    assert pdt.itemsize == ct.sizeof(ba.BlockDesc)
    pd = np.frombuffer((ct.c_char * (pdt.itemsize * nb)).from_address(
        ct.addressof(builder.blocks_ptr().contents)), dtype=pdt)
    assert (pd["field_enc"] != 9).all(), "Plain columns: use oracle_blocks"
    assert (pd["field_vtype"] == 2).all(), "int64 only"
    src_np = np.frombuffer((ct.c_char * plen).from_address(
        ct.addressof(builder.raw_payload_ptr().contents)),
        dtype=np.uint8)
    # 9-byte headers: [enc][firstValue as sign-flip BE cell]
    # (convert/number.go:33-46)
    hdr = np.empty((nb, 9), dtype=np.uint8)
    hdr[:, 0] = pd["field_enc"]
    fv = pd["field_first"].astype(np.int64)
    u = np.where(fv >= 0,
                 fv.view(np.uint64) | np.uint64(1 << 63),
                 (np.uint64(1 << 63) - (-fv).view(np.uint64)))
    hdr[:, 1:9] = u[:, None].view(np.uint8).reshape(nb, 8)[:, ::-1]
    # payload = [original streams][hdr0 stream0 hdr1 stream1 ...]
    offs = pd["field_off"].tolist()
    lens = pd["field_len"].tolist()
    pieces = []
    for i in range(nb):
        pieces.append(hdr[i])
        pieces.append(src_np[offs[i]:offs[i] + lens[i]])
    ext = np.concatenate(pieces)
    total = plen + len(ext)
    src = (ct.c_uint8 * total)()
    ct.memmove(src, builder.raw_payload_ptr(), plen)
    np.frombuffer(src, dtype=np.uint8, count=len(ext),
                  offset=plen)[:] = ext
    del ext, pieces
    col_len = pd["field_len"] + 9
    col_off = np.uint64(plen) + np.concatenate(
        ([np.uint64(0)], np.cumsum(col_len, dtype=np.uint64)[:-1]))
    odt = np.dtype([("series_id", "<u8"), ("count", "<u4"),
                    ("ts_enc_with_version", "u1"), ("version_enc", "u1"),
                    ("_pad", "V2"), ("ts_min", "<i8"), ("ts_max", "<i8"),
                    ("version_first", "<i8"), ("ts_off", "<u8"),
                    ("ts_len", "<u8"), ("ver_len", "<u8"),
                    ("col_off", "<u8"), ("col_len", "<u8"),
                    ("tag_off", "<u8"), ("tag_len", "<u8"),
                    ("tag2_off", "<u8"), ("tag2_len", "<u8"),
                    ("tag3_off", "<u8"), ("tag3_len", "<u8"),
                    ("group_code", "<u4"), ("_pad2", "<u4")])
    assert odt.itemsize == ct.sizeof(o.BlockDesc)
    descs = (o.BlockDesc * nb)()
    od = np.frombuffer(descs, dtype=odt)
    od["series_id"] = pd["series_id"]
    od["count"] = pd["count"]
    od["ts_enc_with_version"] = pd["ts_enc_wv"]
    od["version_enc"] = pd["version_enc"]
    od["ts_min"] = pd["ts_min"]
    od["ts_max"] = pd["ts_max"]
    od["version_first"] = pd["version_first"]
    od["ts_off"] = pd["ts_off"]
    od["ts_len"] = pd["ts_len"]
    od["ver_len"] = 0
    od["col_off"] = col_off
    od["col_len"] = col_len
    od["tag_off"] = pd["tag_off"]
    od["tag_len"] = pd["tag_len"]
    od["tag2_off"] = pd["tag2_off"]
    od["tag2_len"] = pd["tag2_len"]
    od["tag3_off"] = pd["tag3_off"]
    od["tag3_len"] = pd["tag3_len"]
    od["group_code"] = 0
    return src, descs, nb


def kernels_sha16():
    """Stamp of the kernel sources the PMC traffic table was measured on."""
    h = hashlib.sha256()
    with open(os.path.join(REPO, "banyandb_amd", "csrc", "kernels.hip"),
              "rb") as f:
        h.update(f.read())
    return h.hexdigest()[:16]


def run_cpu_baseline(n_dp):
    """Time the CPU oracle (restated reference Go path, kind="port" — no Go
    toolchain on the box) on the same workload: single-thread over a
    bounded sample, AND all host cores over the full series set (SURVEY
    section 8d asks for both, core counts stated).  The oracle's scan call
    releases the GIL, so Python threads over disjoint block slices run on
    separate cores."""
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    sys.path.insert(0, os.path.join(REPO, "tests"))
    import ctypes as ct
    import threading
    import banyandb_amd as ba
    import oracle as o
    from helpers import oracle_blocks
    sample_series = int(os.environ.get("BYDB_CPU_BASELINE_SERIES", 1000))
    full_series = int(os.environ.get("BYDB_CPU_BASELINE_FULL_SERIES", 10000))
    nproc = os.cpu_count() or 8

    b = ba.PartBuilder()
    b.gen_bulk_i64(0, sample_series, n_dp, T0, STRIDE, 1000, 1, SEED,
                   group_mod=0, threads=nproc)
    payload, blocks = oracle_blocks(b)
    t = time.perf_counter()
    res = o.scan_agg(payload, blocks, o.VT_INT64)[0]
    dt1 = time.perf_counter() - t
    dp1 = sample_series * n_dp
    assert res.count == dp1

    # all-cores leg over the full series set (one thread per core, each
    # scanning a disjoint block slice of one shared payload)
    all_cores = None
    try:
        bf = ba.PartBuilder()
        bf.gen_bulk_i64(0, full_series, n_dp, T0, STRIDE, 1000, 1, SEED,
                        group_mod=0, threads=nproc)
        src, descs, nb = fast_oracle_part(bf)
        del bf
        src_ptr = ct.cast(src, o.u8p)
        counts = [0] * nproc

        def worker(ti):
            i0 = nb * ti // nproc
            i1 = nb * (ti + 1) // nproc
            if i1 > i0:
                r = o.scan_agg_raw(src_ptr, descs, i0, i1 - i0, o.VT_INT64)
                counts[ti] = r.count
        threads = [threading.Thread(target=worker, args=(ti,))
                   for ti in range(nproc)]
        t = time.perf_counter()
        for th in threads:
            th.start()
        for th in threads:
            th.join()
        dtn = time.perf_counter() - t
        dpn = full_series * n_dp
        assert sum(counts) == dpn, (sum(counts), dpn)
        all_cores = {
            "value": dpn / dtn,
            "cores": nproc,
            "sample": f"{full_series} series x {n_dp} dp, one pass, "
                      f"{nproc} threads ({dtn:.1f}s)",
        }
    except MemoryError:
        pass
    out = {
        "value": dp1 / dt1,
        "unit": "datapoints/s",
        "cores": 1,
        "kind": "port",
        "sample": f"{sample_series} of {full_series} series x {n_dp} dp, "
                  f"one pass, single thread ({dt1:.1f}s)",
    }
    if all_cores:
        out["all_cores"] = all_cores
    return out


if __name__ == "__main__":
    main()
