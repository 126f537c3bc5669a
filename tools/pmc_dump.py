"""Dump per-kernel PMC sums from the newest rocpd .db (GUID-suffixed schema)."""
import glob, os, sqlite3, sys, collections

d = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out"
dbs = sorted(glob.glob(os.path.join(d, "**", "*.db"), recursive=True),
             key=os.path.getmtime)
db = dbs[-1]
print("==", db)
c = sqlite3.connect(db)
tabs = [r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table'")]
def tab(prefix):
    for t in tabs:
        if t == prefix or t.startswith(prefix + "_0") or t.startswith(prefix + "_f") or (t.startswith(prefix) and len(t) > len(prefix)):
            return t
    return None
t_info = tab("rocpd_info_pmc")
t_ev = tab("rocpd_pmc_event")
t_kd = tab("rocpd_kernel_dispatch")
t_sym = tab("rocpd_info_kernel_symbol")
print("using:", t_info, t_ev, t_kd, t_sym)
kd_cols = [r[1] for r in c.execute(f"PRAGMA table_info({t_kd})")]
ev_cols = [r[1] for r in c.execute(f"PRAGMA table_info({t_ev})")]
# pmc_id in events references the dispatch row id (or correlation); try both
join_col = "id" if "pmc_id" in ev_cols else None
q = f"""
  SELECT s.display_name, i.name, SUM(e.value), COUNT(*)
  FROM {t_ev} e
  JOIN {t_info} i ON i.id = e.event_id
  JOIN {t_kd} k ON k.id = e.pmc_id
  JOIN {t_sym} s ON s.id = k.kernel_id
  GROUP BY s.display_name, i.name"""
try:
    rows = c.execute(q).fetchall()
except sqlite3.OperationalError as ex:
    print("join failed:", ex)
    print("kd cols:", kd_cols)
    print("ev cols:", ev_cols)
    sys.exit(1)
agg = collections.defaultdict(dict)
for name, evn, val, n in rows:
    agg[name.split('(')[0][:70]][evn] = (val, n)
for k, evs in agg.items():
    print(k)
    for evn, (val, n) in sorted(evs.items()):
        print(f"   {evn:28s} {val:.4g}  (n={n})")
