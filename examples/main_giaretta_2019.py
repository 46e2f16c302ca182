"""Giaretta & Girdzijauskas 2019 — gossip learning on skewed topologies.

Engine-first equivalent of the reference's main_giaretta_2019.py (Pegasos
on a Barabasi-Albert scale-free graph), plus the paper's two protocol
remedies for power-law topologies: ``--variant passthrough`` (hubs adopt
low-degree models with prob 1 - deg_s/deg_r and relay them,
gossipy/node.py:289-392) and ``--variant cacheneigh`` (per-neighbor model
slots merged lazily at send time, gossipy/node.py:395-496). The BA graph
is generated with a plain numpy preferential-attachment loop (networkx is
not available here).
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
    BatchedCacheNeighGossipSimulator,
    BatchedGossipSimulator,
    DataArena,
    EngineConfig,
    PegasosSpec,
)
from gossipy_amd.simul import SimulationReport


def barabasi_albert_csr(n: int, m: int = 2, seed: int = 0):
    """Preferential-attachment graph, CSR adjacency."""
    rng = np.random.default_rng(seed)
    edges = set()
    targets = list(range(m))
    repeated = []
    for v in range(m, n):
        chosen = set()
        while len(chosen) < m:
            if repeated and rng.random() < 0.5:
                cand = repeated[rng.integers(len(repeated))]
            else:
                cand = targets[rng.integers(len(targets))]
            if cand != v:
                chosen.add(int(cand))
        for u in chosen:
            edges.add((min(u, v), max(u, v)))
            repeated.extend([u, v])
        targets.append(v)
    adj = [[] for _ in range(n)]
    for u, v in edges:
        adj[u].append(v)
        adj[v].append(u)
    indptr = np.zeros(n + 1, dtype=np.int64)
    for i in range(n):
        indptr[i + 1] = indptr[i] + len(adj[i])
    indices = np.concatenate([np.sort(a) for a in adj]).astype(np.int64)
    return indptr, indices


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=100)
    ap.add_argument("--rounds", type=int, default=100)
    ap.add_argument(
        "--variant",
        choices=["plain", "passthrough", "cacheneigh"],
        default="plain",
    )
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    n, d = args.nodes, 57
    X, y = make_synthetic_classification((46 * n, d, 2), seed=42, margin=2.0)
    y = 2 * y.float() - 1
    idx = np.random.default_rng(42).permutation(len(y))
    cut = int(0.9 * len(y))
    shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], n)]
    data = DataArena.from_shards(
        shards, device, global_eval=(X[idx[cut:]], y[idx[cut:]])
    )

    indptr, indices = barabasi_albert_csr(n, m=2, seed=42)
    cfg = EngineConfig(
        n_nodes=n,
        delta=100,
        protocol=AntiEntropyProtocol.PUSH,
        model_size=d,
        sampling_eval=0.1,
        seed=42,
        peers_indptr=indptr,
        peers_indices=indices,
        pass_through=args.variant == "passthrough",
    )
    spec = PegasosSpec(
        d_in=d, lam=0.01, pass_through=args.variant == "passthrough"
    )
    if args.variant == "cacheneigh":
        sim = BatchedCacheNeighGossipSimulator(cfg, spec, data, device=device)
    else:
        sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)
    print(f"final global eval: {report.get_evaluation(False)[-1][1]}")


if __name__ == "__main__":
    main()
