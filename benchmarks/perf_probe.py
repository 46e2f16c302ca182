"""Per-phase host timing of the flagship bench loop (debug utility)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import (BatchedGossipSimulator, DataArena,
                                EngineConfig, LogRegSpec)

device = torch.device("cuda:0")
n_nodes = 1000
X, y = make_synthetic_classification((4601, 57, 2), seed=7)
rng = np.random.default_rng(7)
idx = rng.permutation(4601)
eval_idx, train_idx = idx[:256], idx[256:]
shards = [(X[s], y[s]) for s in np.array_split(train_idx, n_nodes)]
data = DataArena.from_shards(shards, device, global_eval=(X[eval_idx], y[eval_idx]))
cfg = EngineConfig(n_nodes=n_nodes, delta=100, protocol=AntiEntropyProtocol.PUSH,
                   model_size=116, sampling_eval=0.01, seed=42)
spec = LogRegSpec(d_in=57, n_classes=2, lr=0.1, local_epochs=1, batch_size=32)
sim = BatchedGossipSimulator(cfg, spec, data, device=device)
sim.init_nodes()
print("fast path:", sim._fast_path_ok())

T = {"sched": 0.0, "run": 0.0, "pref": 0.0, "eval": 0.0, "sync": 0.0}
for w in range(5):
    sim.start(n_rounds=1)
torch.cuda.synchronize()

N = 20
t_all0 = time.perf_counter()
for i in range(N):
    r = sim.rounds_done
    t0 = time.perf_counter()
    pre = getattr(sim, "_prefetched", None)
    if pre is not None and pre[0] == r:
        _, sched, flat = pre
    else:
        sched = sim.scheduler.next_round_flat(r)
        flat = sim._maybe_merge(sim.scheduler.last_flat)
    sim._prefetched = None
    sim.pool.ensure(sched.n_slots)
    t1 = time.perf_counter()
    sim._run_round_fast(flat)
    t2 = time.perf_counter()
    s2 = sim.scheduler.next_round_flat(r + 1)
    sim._prefetched = (r + 1, s2, sim._maybe_merge(sim.scheduler.last_flat))
    t3 = time.perf_counter()
    sim._evaluate(sched, (r + 1) * cfg.delta - 1)
    sim.rounds_done += 1
    t4 = time.perf_counter()
    torch.cuda.synchronize()
    t5 = time.perf_counter()
    T["sched"] += t1 - t0
    T["run"] += t2 - t1
    T["pref"] += t3 - t2
    T["eval"] += t4 - t3
    T["sync"] += t5 - t4
t_all1 = time.perf_counter()
for k, v in T.items():
    print(f"{k:6s}: {v/N*1000:8.3f} ms/round")
print(f"total : {(t_all1-t_all0)/N*1000:8.3f} ms/round")
