import json, sys
for line in sys.stdin:
    line = line.strip()
    if not line.startswith('{'):
        print(line)
        continue
    d = json.loads(line)
    r = d.get("roofline") or {}
    print(f'{d["config"]["workload"]}: {d["value"]:.3e} dp/s  wall {d["ms_per_step"]:.2f} ms  '
          f'launch-achieved {r.get("achieved", 0):.0f} GB/s')
