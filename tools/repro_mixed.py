"""Reproduce the mixed-bench DERR_BAD_ENC=2 at block 968 with a small slice."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))
import banyandb_amd as ba

T0 = 1_700_000_000_000_000_000
STRIDE = 10**6
SEED = 0xB4DB
ENVS = [b"prod", b"dev", b"staging", b"qa"]
REGIONS = [f"r{i}".encode() for i in range(16)]
SVCS = [f"s{i}".encode() for i in range(256)]
PREDS = [b"prod", b"r4", b"s4"]

def leg(name, float_leg, s_lo, ns, n_dp, funcs, consume_preds=True):
    b = ba.PartBuilder()
    for slot, table in enumerate([ENVS, REGIONS, SVCS]):
        b.set_tag_table(slot, table)
    if float_leg:
        b.gen_bulk_f64(s_lo, ns, n_dp, T0, STRIDE, 10.0, 0.01, SEED,
                       group_mod=4096, threads=16)
    else:
        b.gen_bulk_i64(s_lo, ns, n_dp, T0, STRIDE, 1000, 1, SEED,
                       group_mod=4096, threads=16)
    s = ba.Session(0)
    s.reserve(b.payload_len + 1024, len(b.blocks()))
    s.append(b)
    fmap = {"sum": ba.AGG_SUM, "count": ba.AGG_COUNT, "min": ba.AGG_MIN,
            "max": ba.AGG_MAX}
    s.configure(ba.VT_FLOAT64 if float_leg else ba.VT_INT64,
                [fmap[f] for f in funcs], n_groups=4096,
                float_exp=-2 if float_leg else 0)
    try:
        s.consume(preds=PREDS if consume_preds else None)
        parts = s.finalize_partials()
        cnt = sum(p.count for p in parts)
        print(f"{name}: OK count={cnt}")
    except RuntimeError as e:
        print(f"{name}: FAIL {e}")
    s.close()

# 16 series x 1M: covers block 968 (series 7 block 107)
leg("i64[0:16]x1M sum+count preds", False, 0, 16, 1_000_000, ["sum","count"])
leg("f64[5000:16]x1M sum+count preds", True, 5000, 16, 1_000_000, ["sum","count"])
leg("f64[5000:16]x1M sum+count nopreds", True, 5000, 16, 1_000_000, ["sum","count"], consume_preds=False)
leg("f64[5000:16]x1M +minmax preds", True, 5000, 16, 1_000_000, ["sum","count","min","max"])
leg("i64[0:16]x1M +minmax preds", False, 0, 16, 1_000_000, ["sum","count","min","max"])
# smaller dp
leg("f64[5000:16]x100k sum+count preds", True, 5000, 16, 100_000, ["sum","count"])
leg("i64[0:16]x100k sum+count preds", False, 0, 16, 100_000, ["sum","count"])
