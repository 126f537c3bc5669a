"""Wide randomized parity sweep (GPU): N seeds of the fuzz scenarios.
Usage: python tools/fuzz_sweep.py [n_seeds]"""
import random
import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'tests'))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..', 'oracle'))

from test_gpu_fuzz import build_scenario, check_scalar, check_bytag

n = int(sys.argv[1]) if len(sys.argv) > 1 else 100
fails = 0
for seed in range(n):
    rng = random.Random(0xABC000 + seed)
    try:
        b, is_float, tag_kind = build_scenario(rng)
        check_scalar(rng, b, is_float, tag_kind)
        if tag_kind:
            check_bytag(rng, b, is_float)
    except AssertionError as e:
        fails += 1
        print(f"seed {seed}: MISMATCH {e}")
    if seed % 50 == 49:
        print(f"{seed + 1}/{n} done, fails={fails}", flush=True)
print(f"SWEEP COMPLETE: {n} scenarios, {fails} failures")
sys.exit(1 if fails else 0)
