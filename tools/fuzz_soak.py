#!/usr/bin/env python3
"""Extended fuzz soak (GPU): many more random plan descriptors than
the committed suite runs, same exact-evaluation checker.  One-off
deep parity sweep; any failure prints the iteration seed context."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "oracle"))

import numpy as np

import test_gpu_plan_fuzz as F
from greengage_amd import Engine

eng = Engine(device=0, n_segments=1, segment_id=0)
n_iter = int(sys.argv[1]) if len(sys.argv) > 1 else 150
seed = int(sys.argv[2]) if len(sys.argv) > 2 else 987654321
rng = np.random.default_rng(seed)
fails = 0
for it in range(n_iter):
    try:
        F._run_one(eng, rng, 1000 + it,
                   force_interp=(it % 17 == 0))
    except AssertionError as ex:
        fails += 1
        print(f"FAIL it={it}: {ex}")
        if fails >= 3:
            break
    if (it + 1) % 25 == 0:
        print(f"{it + 1}/{n_iter} plans OK", flush=True)
print(f"SOAK_DONE fails={fails}")
eng.shutdown()
