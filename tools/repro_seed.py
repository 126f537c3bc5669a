"""Reproduce one fuzz-sweep family-1 seed with full detail."""
import random, sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'tests'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'oracle'))

from test_gpu_fuzz import build_scenario, TAGS
from helpers import oracle_scan, oracle_blocks
from banyandb_amd import (Session, VT_INT64, VT_FLOAT64,
                          AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX)
import oracle as o

seed = int(sys.argv[1])
rng = random.Random(0xABC000 + seed)
b, is_float, tag_kind = build_scenario(rng)
print(f"seed={seed} is_float={is_float} tag_kind={tag_kind} "
      f"blocks={len(b.blocks())}")
for i, d in enumerate(b.blocks()):
    print(f"  blk{i}: n={d.count} fenc={d.field_enc} exp={d.exp} "
          f"taglen={d.tag_len}")
vtype = VT_FLOAT64 if is_float else VT_INT64
exp = min(d.exp for d in b.blocks()) if is_float else 0
mode = rng.randrange(3)
kw = {}
T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
if mode == 1:
    lo = T0 + rng.randint(0, 2000) * MS
    hi = lo + rng.randint(0, 8192) * MS
    kw = dict(min_ts=lo, max_ts=hi)
pred = None
if tag_kind and rng.random() < 0.6:
    pred = TAGS[rng.randrange(len(TAGS))]
print(f"mode={mode} kw={kw} pred={pred}")
orc = oracle_scan(b, vtype, pred=pred or b"", **kw)[0]
s = Session(0)
s.upload_part(b)
s.configure(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], float_exp=exp)
s.consume(pred=pred or b"", **kw)
g = s.finalize()[0]
print(f"GPU: count={g.count} sum_i={g.sum_i} min_i={g.min_i} max_i={g.max_i} "
      f"sum_f={g.sum_f} min_f={g.min_f} max_f={g.max_f}")
print(f"ORC: count={orc.count} sum_i={orc.sum_i} min_i={orc.min_i} "
      f"max_i={orc.max_i} sum_f={orc.sum_f} min_f={orc.min_f} "
      f"max_f={orc.max_f}")
# bytag leg too (as the sweep does)
if tag_kind:
    domain = list(TAGS[: rng.randint(2, len(TAGS))])
    payload, blocks = oracle_blocks(b)
    orc2 = o.scan_agg_bytag(payload, blocks, vtype, 0, domain)
    s.configure_by_tag(vtype, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX], 0,
                       domain, float_exp=exp)
    s.consume()
    gs = s.finalize()
    for gi, (gg, oo) in enumerate(zip(gs, orc2)):
        flag = ""
        if gg.count != oo.count or (not is_float and gg.sum_i != oo.sum_i):
            flag = "   <-- MISMATCH"
        if not is_float and oo.count and (gg.min_i != oo.min_i or gg.max_i != oo.max_i):
            flag = "   <-- MISMATCH"
        print(f"  bytag g{gi}: GPU cnt={gg.count} sum={gg.sum_i} "
              f"min={gg.min_i} max={gg.max_i} | ORC cnt={oo.count} "
              f"sum={oo.sum_i} min={oo.min_i} max={oo.max_i}{flag}")
s.close()
