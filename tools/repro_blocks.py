"""Per-block GPU-vs-oracle isolation for a fuzz seed."""
import random, sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'tests'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'oracle'))
from test_gpu_fuzz import build_scenario
from helpers import oracle_blocks
from banyandb_amd import (Session, PartBuilder, VT_FLOAT64,
                          AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, lib)
import banyandb_amd as ba
import ctypes as C
import oracle as o

seed = int(sys.argv[1])
rng = random.Random(0xABC000 + seed)
b, is_float, tag_kind = build_scenario(rng)
payload, blocks = oracle_blocks(b)
cfg_exp = min(d.exp for d in b.blocks())
print("cfg_exp", cfg_exp)
raw = b.payload
descs = b.blocks()
_l = lib()
for i, d in enumerate(descs):
    s = Session(0)
    s._ck(_l.bydb_part_reserve(s._h, len(raw), 1))
    buf = (C.c_uint8 * len(raw)).from_buffer_copy(raw)
    arr = (ba.BlockDesc * 1)(d)
    s._ck(_l.bydb_part_append(s._h, buf, len(raw), arr, 1))
    s.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                float_exp=cfg_exp)
    s.consume()
    g = s.finalize()[0]
    s.close()
    orc = o.scan_agg(payload, [blocks[i]], VT_FLOAT64)[0]
    bad = (g.count != orc.count or g.min_f != orc.min_f or
           g.max_f != orc.max_f or
           abs(g.sum_f - orc.sum_f) > 1e-6 + 1e-9 * abs(orc.sum_f))
    if bad:
        print(f"blk{i}: exp={d.exp} enc={d.field_enc} n={d.count}")
        print(f"  GPU cnt={g.count} sum={g.sum_f} min={g.min_f} max={g.max_f}")
        print(f"  ORC cnt={orc.count} sum={orc.sum_f} min={orc.min_f} max={orc.max_f}")
print("done")
