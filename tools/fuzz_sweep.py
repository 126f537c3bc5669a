"""Wide randomized parity sweep (GPU): N seeds of the fuzz scenarios.
Usage: python tools/fuzz_sweep.py [n_seeds] [seed_base]"""
import random
import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'tests'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'oracle'))

from test_gpu_fuzz import build_scenario, check_scalar, check_bytag

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 100
    base = int(sys.argv[2], 0) if len(sys.argv) > 2 else 0xABC000
    fails = 0
    for seed in range(n):
        rng = random.Random(base + seed)
        try:
            if seed % 2 == 0:
                b, is_float, tag_kind = build_scenario(rng)
                check_scalar(rng, b, is_float, tag_kind)
                if tag_kind:
                    check_bytag(rng, b, is_float)
            else:
                b, kind, envs, regions = build_scenario2(rng)
                check_scenario2(rng, b, kind, envs, regions)
        except AssertionError as e:
            fails += 1
            print(f"seed {seed}: MISMATCH {e}")
        if seed % 50 == 49:
            print(f"{seed + 1}/{n} done, fails={fails}", flush=True)
    print(f"SWEEP COMPLETE: {n} scenarios, {fails} failures")
    sys.exit(1 if fails else 0)


# ---- family 2: the round's wider feature space ----
def build_scenario2(rng):
    """Nullable columns, plain (>256-card) tags, composite groups and
    sparse predicates, in random combination."""
    from banyandb_amd import PartBuilder
    kind = rng.randrange(4)
    b = PartBuilder()
    n_blocks = rng.randint(2, 10)
    envs = [b"prod", b"dev", b"staging", b"qa"]
    regions = [b"r%d" % i for i in range(6)]
    for sid in range(n_blocks):
        n = rng.choice([5, 64, 700, 1024, 2048, 8192])
        ts = [T0 + i * MS for i in range(n)]
        if kind == 0:       # nullable i64
            vals = [None if rng.random() < 0.3
                    else rng.randint(-10**12, 10**12) for _ in range(n)]
            b.add_block_i64_nullable(sid + 1, ts, [1] * n, vals)
        elif kind == 1:     # nullable f64 (raw cells)
            vals = [None if rng.random() < 0.3
                    else rng.uniform(-1e9, 1e9) for _ in range(n)]
            b.add_block_f64_nullable(sid + 1, ts, [1] * n, vals)
        else:
            b.add_block_i64(sid + 1, ts, [1] * n,
                            [rng.randint(-10**9, 10**9) for _ in range(n)])
        # tag slot 0: sometimes plain high-card, sometimes dict
        if kind == 2:
            b.set_block_tag([b"user_%03d" % rng.randrange(300)
                             for _ in range(n)])
        else:
            tags = []
            while len(tags) < n:
                run = min(rng.randint(1, 200), n - len(tags))
                v = None if rng.random() < 0.1 else envs[rng.randrange(4)]
                tags.extend([v] * run)
            b.set_block_tag(tags)
        # tag slot 1: row-varying regions
        tags2 = []
        while len(tags2) < n:
            run = min(rng.randint(1, 150), n - len(tags2))
            tags2.extend([regions[rng.randrange(6)]] * run)
        b.set_block_tag(tags2)
    return b, kind, envs, regions


def check_scenario2(rng, b, kind, envs, regions):
    import math
    import oracle as o
    from helpers import oracle_blocks
    from banyandb_amd import (Session, VT_INT64, VT_FLOAT64, FLOAT_RAW_EXP,
                              AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
    vtype = VT_FLOAT64 if kind == 1 else VT_INT64
    fexp = FLOAT_RAW_EXP if kind == 1 else 0
    payload, blocks = oracle_blocks(b)
    mode = rng.randrange(3)
    preds = None
    if mode == 0 and kind != 2:
        preds = [b"", b"r%d" % rng.randrange(6), b""]
    elif mode == 1 and kind == 2:
        # bitmap-mode (plain-tag) predicate — combines with scalar AND
        # row-varying group folds (fold_range per-row filtering)
        preds = [b"user_%03d" % rng.randrange(300), b"", b""]
    if rng.random() < 0.4:
        # scalar (with optional preds)
        orc = o.scan_agg(payload, blocks, vtype, preds=preds)[0]
        s = Session(0)
        s.upload_part(b)
        s.configure(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                    float_exp=fexp)
        s.consume(preds=preds)
        g = s.finalize()[0]
        s.close()
        pairs = [(g, orc)]
    elif kind == 2 and rng.random() < 0.5:
        # group by the PLAIN high-cardinality tag (slot 0), domain a
        # random subset; optional region predicate on slot 1
        dom = [b"user_%03d" % i
               for i in range(0, 300, rng.choice([1, 2, 3]))]
        gpreds = None
        if rng.random() < 0.5:
            gpreds = [b"", b"r%d" % rng.randrange(6), b""]
        orc = o.scan_agg_bytags(payload, blocks, vtype, [0], [dom],
                                preds=gpreds)
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tag(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 0,
                           dom, float_exp=fexp)
        s.consume(preds=[p_ or b"" for p_ in gpreds] if gpreds else None)
        gs = s.finalize()
        s.close()
        pairs = list(zip(gs, orc))
    elif rng.random() < 0.3:
        # COMPOSITE key over (env slot 0, regions slot 1) — when slot 0
        # is plain (kind==2) this mixes a plain key slot with a
        # dictionary one; optional ts clamp
        doms = [[b"prod", b"dev", b"staging", b"qa"], regions]
        if kind == 2:
            doms[0] = [b"user_%03d" % i for i in range(0, 300, 2)]
        lo = T0 + rng.randint(0, 500) * MS
        hi = T0 + rng.randint(500, 9000) * MS
        orc = o.scan_agg_bytags(payload, blocks, vtype, [0, 1], doms,
                                min_ts=lo, max_ts=hi)
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tags(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                            [0, 1], doms, float_exp=fexp)
        s.consume(min_ts=lo, max_ts=hi)
        gs = s.finalize()
        fsk = s.group_first_seen()
        s.close()
        for g in range(len(gs)):
            if gs[g].count > 0:
                assert fsk[g] != (1 << 64) - 1, "first-seen missing"
        pairs = list(zip(gs, orc))
    else:
        # group by regions (slot 1), optional preds on other slots;
        # sometimes MODE_MAP + host reduce_partials2 instead of ALL
        orc = o.scan_agg_bytags(payload, blocks, vtype, [1], [regions],
                                preds=preds)
        s = Session(0)
        s.upload_part(b)
        if rng.random() < 0.3:
            from banyandb_amd import MODE_MAP, reduce_partials
            s.configure_by_tag(vtype,
                               [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 1,
                               regions, mode=MODE_MAP, float_exp=fexp)
            s.consume(preds=preds)
            parts = s.finalize_partials()
            gs = reduce_partials(parts, 1, len(regions), vtype,
                                 float_exp=fexp)
        else:
            s.configure_by_tag(vtype,
                               [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 1,
                               regions, float_exp=fexp)
            s.consume(preds=preds)
            gs = s.finalize()
        s.close()
        pairs = list(zip(gs, orc))
    for g, oc in pairs:
        assert g.count == oc.count, f"count {g.count} != {oc.count}"
        if kind == 1:
            if oc.count:
                assert g.min_f == oc.min_f and g.max_f == oc.max_f
                assert math.isclose(g.sum_f, oc.sum_f, rel_tol=1e-9,
                                    abs_tol=1e-6)
        else:
            assert g.sum_i == oc.sum_i
            if oc.count:
                assert g.min_i == oc.min_i and g.max_i == oc.max_i


if __name__ == '__main__':
    main()
