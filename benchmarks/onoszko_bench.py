"""BASELINE.json config 4: Onoszko-2021 CIFAR10Net CNN gossip at 100
nodes (CIFAR-shaped synthetic, 40 samples/node, batch 32, lr 0.1,
MERGE_UPDATE, PUSH, delta=100, sampling_eval=0.05) — the EXACT config
benchmarks/ref_baseline_345.py::config4 measures on the reference
(0.694 rounds/s on the dev-box CPU).

The engine path batches every tick's CNN work: vectorized merges +
one vmap(grad) SGD trajectory per wave (grouped conv on MIOpen), instead
of the reference's per-node python loop. ``--loop`` A/Bs the per-node
loop path (GOSSIPY_TORCHMOD_LOOP=1).

Usage: python benchmarks/onoszko_bench.py [--steps 20] [--warmup 3] [--loop]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def cifar10net_factory():
    import torch.nn as nn
    import torch.nn.functional as F

    class CIFAR10Net(nn.Module):
        """The reference PENS CNN (main_onoszko_2021.py:31-60)."""

        def __init__(self):
            super().__init__()
            self.conv1 = nn.Conv2d(3, 32, 3)
            self.pool = nn.MaxPool2d(2, 2)
            self.conv2 = nn.Conv2d(32, 64, 3)
            self.conv3 = nn.Conv2d(64, 64, 3)
            self.fc1 = nn.Linear(64 * 2 * 2, 64)
            self.fc2 = nn.Linear(64, 10)

        def forward(self, x):
            x = self.pool(F.relu(self.conv1(x)))
            x = self.pool(F.relu(self.conv2(x)))
            x = self.pool(F.relu(self.conv3(x)))
            x = x.view(-1, 64 * 2 * 2)
            x = F.relu(self.fc1(x))
            return self.fc2(x)

    return CIFAR10Net()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--nodes", type=int, default=100)
    ap.add_argument("--loop", action="store_true",
                    help="A/B: per-node loop instead of the batched path")
    args = ap.parse_args()
    if args.loop:
        os.environ["GOSSIPY_TORCHMOD_LOOP"] = "1"

    from gossipy_amd.core import AntiEntropyProtocol
    from gossipy_amd.engine import (
        BatchedGossipSimulator,
        DataArena,
        EngineConfig,
        TorchModuleSpec,
    )

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    n, per = args.nodes, 40
    rng = np.random.default_rng(42)
    labels = rng.integers(0, 10, size=n * per)
    x = rng.normal(0, 0.3, size=(len(labels), 3, 32, 32)).astype(np.float32)
    for c in range(10):
        x[labels == c, c % 3] += 0.8 + 0.25 * c
    X = torch.from_numpy(x.reshape(len(labels), -1))
    y = torch.from_numpy(labels).float()
    shards = [
        (X[i * per : (i + 1) * per], y[i * per : (i + 1) * per])
        for i in range(n)
    ]
    # fixed-size global eval set (the 100-node config's 4000 samples) so
    # the eval sweep doesn't scale quadratically with the population
    ev = rng.permutation(len(labels))[:4000]
    data = DataArena.from_shards(shards, device, global_eval=(X[ev], y[ev]))
    spec = TorchModuleSpec(
        cifar10net_factory, input_shape=(3, 32, 32), lr=0.1, batch_size=32
    )
    cfg = EngineConfig(
        n_nodes=n, delta=100, protocol=AntiEntropyProtocol.PUSH,
        model_size=spec.D, sampling_eval=0.05, seed=42,
    )
    sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    sim.init_nodes()
    sim.start(n_rounds=args.warmup)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    sim.start(n_rounds=args.steps)
    if device.type == "cuda":
        torch.cuda.synchronize()
    rps = args.steps / (time.perf_counter() - t0)
    print(json.dumps({
        "config": f"onoszko-cifar10net-{n}n-mergeupdate"
                  + ("-LOOP" if args.loop else ""),
        "rounds_per_sec": round(rps, 3),
        "node_rounds_per_sec": round(rps * n, 1),
        "reference_cpu": 0.694,
        "speedup_vs_ref": round(rps / 0.694, 1),
        "device": str(device),
    }))


if __name__ == "__main__":
    main()
