"""One-off deep fuzz: many random engine configs, HIP vs torch oracle."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import numpy as np
import torch

from test_fuzz_gpu import _rand_case, _run, CUDA, CPU

N = int(sys.argv[1]) if len(sys.argv) > 1 else 150
bad = 0
for case in range(N):
    rng = np.random.default_rng(50_000 + case)
    cfg, spec, k, pm1 = _rand_case(rng)
    g = _run(cfg, spec, k, pm1, CUDA)
    c = _run(cfg, spec, k, pm1, CPU)
    ok = torch.allclose(
        g.local_params().cpu(), c.local_params(), atol=2e-3, rtol=2e-3
    ) and torch.equal(g.state.ages.cpu(), c.state.ages)
    if not ok:
        bad += 1
        print(f"CASE {case} DIVERGED: {cfg}\n  {spec}")
    if case % 25 == 24:
        print(f"{case + 1}/{N} done, {bad} divergent")
print(f"RESULT: {N - bad}/{N} matched")
assert bad == 0
