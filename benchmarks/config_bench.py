"""Engine throughput on the secondary BASELINE.md configurations.

Row 2: Ormandi-2013 (Pegasos 57-d, 100 nodes, PUSH_PULL, async,
       sampling_eval=0.1) — reference: 1.44 rounds/s on CPU.
Row 3: Hegedus-2021 tokenized (PartitionedTMH logreg 4 parts, 100 nodes,
       20-regular graph, RandomizedTokenAccount(C=20, A=10), UPDATE) —
       reference: 15.7 rounds/s on CPU.

Usage: python benchmarks/config_bench.py [--steps 30] [--warmup 5]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol, CreateModelMode
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import (
    BatchedGossipSimulator,
    BatchedTokenizedGossipSimulator,
    DataArena,
    EngineConfig,
    LogRegSpec,
    MLPSpec,
    PegasosSpec,
)
from gossipy_amd.flow_control import RandomizedTokenAccount
from examples.main_hegedus_2021 import k_regular_csr


def _data(n_nodes, device, pm1=False):
    X, y = make_synthetic_classification((46 * n_nodes, 57, 2), seed=42)
    if pm1:
        y = 2 * y.float() - 1
    idx = np.random.default_rng(42).permutation(len(y))
    cut = int(0.9 * len(y))
    shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], n_nodes)]
    return DataArena.from_shards(
        shards, device, global_eval=(X[idx[cut:]], y[idx[cut:]])
    )


def timed(sim, steps, warmup, device):
    sim.init_nodes()
    sim.start(n_rounds=warmup)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    sim.start(n_rounds=steps)
    if device.type == "cuda":
        torch.cuda.synchronize()
    return steps / (time.perf_counter() - t0)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    args = ap.parse_args()
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")

    # --- row 2: Ormandi 2013
    cfg = EngineConfig(
        n_nodes=100, delta=100, protocol=AntiEntropyProtocol.PUSH_PULL,
        model_size=57, sync=False, sampling_eval=0.1, seed=42,
    )
    sim = BatchedGossipSimulator(
        cfg, PegasosSpec(d_in=57, lam=0.01), _data(100, device, pm1=True),
        device=device,
    )
    r2 = timed(sim, args.steps, args.warmup, device)
    print(json.dumps({
        "config": "ormandi2013-pegasos-100n-pushpull-async",
        "rounds_per_sec": round(r2, 2),
        "reference_cpu": 1.44,
        "speedup": round(r2 / 1.44, 1),
    }))

    # --- row 3: Hegedus 2021 tokenized + partitioned
    indptr, indices = k_regular_csr(100, 20)
    cfg = EngineConfig(
        n_nodes=100, delta=100, protocol=AntiEntropyProtocol.PUSH,
        model_size=116, sampling_eval=0.1, seed=42, n_parts=4,
        peers_indptr=indptr, peers_indices=indices,
    )
    spec = LogRegSpec(
        d_in=57, n_classes=2, lr=0.1, n_parts=4, mode=CreateModelMode.UPDATE
    )
    sim = BatchedTokenizedGossipSimulator(
        cfg, spec, _data(100, device),
        token_account=RandomizedTokenAccount(C=20, A=10),
        # the reference experiment's utility is the constant 1; the int form
        # routes through the native C++ tokenized scheduler (a python
        # callable forces per-message python evaluation instead)
        utility_fun=1,
        device=device,
    )
    r3 = timed(sim, args.steps, args.warmup, device)
    print(json.dumps({
        "config": "hegedus2021-tokenized-partitioned-100n-20regular",
        "rounds_per_sec": round(r3, 2),
        "reference_cpu": 15.7,
        "speedup": round(r3 / 15.7, 1),
    }))


def mlp_bench(steps=20, warmup=3):
    """Giaretta-shaped MLP gossip at 1000 nodes (BASELINE.json config 3
    shape, 1 GPU): 57 -> 100 -> 2 MLP, every tick's GEMMs on MFMA."""
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    spec = MLPSpec(d_in=57, n_classes=2, hidden=(100,), lr=0.05, batch_size=32)
    cfg = EngineConfig(
        n_nodes=1000, delta=100, protocol=AntiEntropyProtocol.PUSH,
        model_size=spec.D, sampling_eval=0.01, seed=42,
    )
    sim = BatchedGossipSimulator(cfg, spec, _data(1000, device), device=device)
    r = timed(sim, steps, warmup, device)
    print(json.dumps({
        "config": "giaretta-shaped-mlp-57-100-2-1000n",
        "rounds_per_sec": round(r, 2),
        "node_rounds_per_sec": round(r * 1000, 0),
    }))


if __name__ == "__main__":
    main()
    mlp_bench()
