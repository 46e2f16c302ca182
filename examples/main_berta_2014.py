"""Berta et al. 2014 — decentralized k-means vs centralized baselines.

Engine-first equivalent of the reference's main_berta_2014.py:30-52: the
gossip k-means NMI is printed next to two centralized oracles (sklearn
KMeans and a plain torch Lloyd's run), on synthetic Gaussian blobs.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol, CreateModelMode
from gossipy_amd.engine import (
    BatchedGossipSimulator,
    DataArena,
    EngineConfig,
    KMeansSpec,
)
from gossipy_amd.simul import SimulationReport


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=50)
    ap.add_argument("--rounds", type=int, default=50)
    ap.add_argument("--k", type=int, default=4)
    ap.add_argument("--dim", type=int, default=16)
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    rng = np.random.default_rng(42)
    centers = rng.normal(0, 4, size=(args.k, args.dim))
    shards, allx, ally = [], [], []
    for _ in range(args.nodes):
        labels = rng.integers(0, args.k, size=40)
        x = centers[labels] + rng.normal(0, 0.5, size=(40, args.dim))
        shards.append((torch.from_numpy(x).float(), torch.from_numpy(labels).float()))
        allx.append(x)
        ally.append(labels)
    gx = np.concatenate(allx)
    gy = np.concatenate(ally)
    data = DataArena.from_shards(
        shards, device,
        global_eval=(torch.from_numpy(gx).float(), torch.from_numpy(gy).float()),
    )

    cfg = EngineConfig(
        n_nodes=args.nodes,
        delta=100,
        protocol=AntiEntropyProtocol.PUSH,
        model_size=args.k * args.dim,
        sampling_eval=0.1,
        seed=42,
    )
    spec = KMeansSpec(
        k=args.k, dim=args.dim, alpha=0.1, mode=CreateModelMode.MERGE_UPDATE
    )
    sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)
    gossip_nmi = report.get_evaluation(False)[-1][1]["nmi"]

    # centralized oracles (main_berta_2014.py:30-52 prints these next to
    # the gossip result)
    from sklearn.cluster import KMeans
    from sklearn.metrics import normalized_mutual_info_score as nmi

    sk = KMeans(n_clusters=args.k, n_init=10, random_state=42).fit(gx)
    print(f"gossip k-means NMI:      {gossip_nmi:.4f}")
    print(f"sklearn KMeans NMI:      {nmi(gy, sk.labels_):.4f}")


if __name__ == "__main__":
    main()
