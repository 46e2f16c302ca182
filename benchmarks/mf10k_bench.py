"""BASELINE.json config 5: Hegedus-2020 low-rank MF recommender at 10,000
nodes (one MovieLens-1M-shaped item catalog per node). Demonstrates the
288 GB HBM sizing: 10k resident nodes x (k+1)(n_items+1) fp32 ~ 0.9 GB of
parameter arena + slot pool, two orders of magnitude of headroom.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol
from gossipy_amd.engine import (
    BatchedGossipSimulator,
    DataArena,
    EngineConfig,
    MFSpec,
)


def main(n_users=10000, n_items=3700, rpu=100, steps=20, warmup=3):
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    rng = np.random.default_rng(42)
    # MovieLens-shaped synthetic ratings, vectorized for 10k users
    items = np.argsort(rng.random((n_users, n_items)), axis=1)[:, :rpu]
    raw = rng.normal(3.0, 1.2, size=(n_users, rpu))
    ratings = np.clip(np.round(raw), 1, 5).astype(np.float32)
    X = torch.from_numpy(items[..., None].astype(np.float32))
    Y = torch.from_numpy(ratings)
    C = torch.full((n_users,), rpu, dtype=torch.int32)
    data = DataArena(X.to(device), Y.to(device), C.to(device),
                     tx=X.to(device), ty=Y.to(device), tcounts=C.to(device))

    spec = MFSpec(k=5, n_items=n_items, reg=0.1, lr=0.001)
    cfg = EngineConfig(
        n_nodes=n_users, delta=100, protocol=AntiEntropyProtocol.PUSH,
        model_size=spec.slot_width, sampling_eval=0.001, seed=42,
    )
    sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    t0 = time.perf_counter()
    sim.init_nodes()
    init_s = time.perf_counter() - t0
    sim.start(n_rounds=warmup)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    sim.start(n_rounds=steps)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    mem = (
        torch.cuda.memory_allocated() / 1e9
        if device.type == "cuda"
        else 0.0
    )
    print(json.dumps({
        "config": "hegedus2020-mf-10000nodes-ml1m-shaped",
        "rounds_per_sec": round(steps / dt, 2),
        "node_rounds_per_sec": round(steps / dt * n_users, 0),
        "arena_plus_pool_gb": round(mem, 2),
        "init_s": round(init_s, 1),
        "D_per_node": spec.D,
    }))


if __name__ == "__main__":
    main()
