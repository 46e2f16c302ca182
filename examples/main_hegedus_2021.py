"""Hegedus et al. 2021 — partitioned gossip learning with token-account
flow control.

Engine-first equivalent of the reference's main_hegedus_2021.py:43-60:
PartitionedTMH logistic regression (4 partitions), 100 nodes, 20-regular
topology, TokenizedGossipSimulator + RandomizedTokenAccount(C=20, A=10),
constant utility. Synthetic spambase-shaped data.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol, StaticP2PNetwork
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import (
    BatchedTokenizedGossipSimulator,
    DataArena,
    EngineConfig,
    LogRegSpec,
)
from gossipy_amd.flow_control import RandomizedTokenAccount
from gossipy_amd.simul import SimulationReport


def k_regular_csr(n: int, k: int, seed: int = 0):
    """k-regular ring-lattice topology (each node links its k nearest)."""
    half = k // 2
    indptr = np.arange(0, n * k + 1, k, dtype=np.int64)
    indices = np.empty(n * k, dtype=np.int64)
    for i in range(n):
        nbrs = [(i + off) % n for off in range(-half, half + 1) if off != 0]
        indices[i * k : (i + 1) * k] = nbrs[:k]
    return indptr, indices


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=100)
    ap.add_argument("--rounds", type=int, default=100)
    ap.add_argument("--parts", type=int, default=4)
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

    indptr, indices = k_regular_csr(n, 20)
    cfg = EngineConfig(
        n_nodes=n,
        delta=100,
        protocol=AntiEntropyProtocol.PUSH,
        model_size=2 * d + 2,
        sampling_eval=0.1,
        seed=42,
        n_parts=args.parts,
        peers_indptr=indptr,
        peers_indices=indices,
    )
    spec = LogRegSpec(d_in=d, n_classes=2, lr=0.1, n_parts=args.parts)
    sim = BatchedTokenizedGossipSimulator(
        cfg,
        spec,
        data,
        token_account=RandomizedTokenAccount(C=20, A=10),
        utility_fun=lambda recv, sender, t: 1,  # main_hegedus_2021.py:57
        device=device,
    )
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)
    print(f"final global eval: {report.get_evaluation(False)[-1][1]}")
    print(f"messages sent: {report._sent_messages}, failed: {report._failed_messages}")


if __name__ == "__main__":
    main()
