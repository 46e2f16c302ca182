"""Hegedus et al. 2020 — decentralized matrix-factorization recommender.

Engine-first equivalent of the reference's main_hegedus_2020.py:32-54
(MFModelHandler dim=5 on MovieLens, RecSysDataDispatcher, 20-regular
topology, MERGE_UPDATE). Runs on synthetic MovieLens-shaped ratings.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import torch

from gossipy_amd.core import AntiEntropyProtocol
from gossipy_amd.data import make_synthetic_recsys
from gossipy_amd.engine import (
    BatchedGossipSimulator,
    DataArena,
    EngineConfig,
    MFSpec,
)
from gossipy_amd.simul import SimulationReport
from examples.main_hegedus_2021 import k_regular_csr


def build_arena(n_users, n_items, device):
    ratings, _, _ = make_synthetic_recsys(n_users, n_items, 80, seed=42)
    shards, tests = [], []
    for u in range(n_users):
        rs = ratings[u]
        cut = max(1, int(0.8 * len(rs)))
        shards.append(
            (torch.tensor([[i] for i, _ in rs[:cut]], dtype=torch.float32),
             torch.tensor([r for _, r in rs[:cut]]))
        )
        tests.append(
            (torch.tensor([[i] for i, _ in rs[cut:]], dtype=torch.float32),
             torch.tensor([r for _, r in rs[cut:]]))
        )
    return DataArena.from_shards(shards, device, test_shards=tests)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--users", type=int, default=100)
    ap.add_argument("--items", type=int, default=500)
    ap.add_argument("--rounds", type=int, default=50)
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    data = build_arena(args.users, args.items, device)
    indptr, indices = k_regular_csr(args.users, 20)
    spec = MFSpec(k=5, n_items=args.items, reg=0.1, lr=0.001)
    cfg = EngineConfig(
        n_nodes=args.users,
        delta=100,
        protocol=AntiEntropyProtocol.PUSH,
        model_size=spec.slot_width,
        sampling_eval=0.1,
        seed=42,
        peers_indptr=indptr,
        peers_indices=indices,
    )
    sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)
    ev = report.get_evaluation(True)
    print(f"first local RMSE: {ev[0][1]}  final: {ev[-1][1]}")


if __name__ == "__main__":
    main()
