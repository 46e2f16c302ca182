"""Koloskova et al. 2020 — decentralized weighted averaging (all2all).

Engine-first equivalent of the reference's main_all2all.py (All2All
simulator + WeightedTMH + UniformMixing): every timed-out node merges its
accumulated neighbor models with mixing weights and broadcasts to all
peers.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import (
    BatchedAll2AllGossipSimulator,
    DataArena,
    EngineConfig,
    LogRegSpec,
)
from gossipy_amd.simul import SimulationReport


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=50)
    ap.add_argument("--rounds", type=int, default=50)
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    n, d = args.nodes, 57
    X, y = make_synthetic_classification((46 * n, d, 2), seed=42, margin=2.0)
    idx = np.random.default_rng(42).permutation(len(y))
    cut = int(0.9 * len(y))
    shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], n)]
    data = DataArena.from_shards(
        shards, device, global_eval=(X[idx[cut:]], y[idx[cut:]])
    )
    cfg = EngineConfig(
        n_nodes=n,
        delta=100,
        protocol=AntiEntropyProtocol.PUSH,
        model_size=2 * d + 2,
        sampling_eval=0.1,
        seed=42,
    )
    sim = BatchedAll2AllGossipSimulator(
        cfg, LogRegSpec(d_in=d, n_classes=2, lr=0.1), data, device=device
    )
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)
    print(f"final global eval: {report.get_evaluation(False)[-1][1]}")


if __name__ == "__main__":
    main()
