"""Dump per-kernel PMC sums from the newest rocpd .db (GUID-suffixed schema).

Schema note (ROCm 7.2 rocpd): rocpd_pmc_event.event_id references the
kernel DISPATCH row (rocpd_kernel_dispatch.id) and .pmc_id references the
counter (rocpd_info_pmc.id); multiple rows per (dispatch, counter) are
per-shader-engine instances and must be summed."""
import glob
import os
import sqlite3
import sys
import collections

d = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out"
dbs = sorted(glob.glob(os.path.join(d, "**", "*.db"), recursive=True),
             key=os.path.getmtime)
db = dbs[-1]
print("==", db)
c = sqlite3.connect(db)
tabs = [r[0] for r in c.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]


def tab(prefix):
    for t in tabs:
        if t.startswith(prefix + "_0") or t == prefix:
            return t
    for t in tabs:
        if t.startswith(prefix):
            return t
    return None


t_info = tab("rocpd_info_pmc")
t_ev = tab("rocpd_pmc_event")
t_kd = tab("rocpd_kernel_dispatch")
t_sym = tab("rocpd_info_kernel_symbol")
q = f"""
  SELECT s.display_name, i.name, SUM(e.value), COUNT(*), COUNT(DISTINCT k.id)
  FROM {t_ev} e
  JOIN {t_info} i ON i.id = e.pmc_id
  JOIN {t_kd} k ON k.id = e.event_id
  JOIN {t_sym} s ON s.id = k.kernel_id
  GROUP BY s.display_name, i.name"""
rows = c.execute(q).fetchall()
agg = collections.defaultdict(dict)
for name, evn, val, n, nd in rows:
    agg[name.split('(')[0][:70]][evn] = (val, nd)
for k, evs in agg.items():
    print(k)
    for evn, (val, nd) in sorted(evs.items()):
        print(f"   {evn:28s} {val:.5g}  (dispatches={nd}, "
              f"per-dispatch={val / max(nd, 1):.5g})")
