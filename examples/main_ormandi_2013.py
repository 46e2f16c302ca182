"""Ormandi et al. 2013 — gossip learning with Pegasos / AdaLine.

Engine-first equivalent of the reference's main_ormandi_2013.py:22-55
(Pegasos + AdaLine on spambase, PUSH, async, online_prob=.2, drop=.1).
Runs on synthetic spambase-shaped data (no network in this environment) on
the batched MI355X engine; pass ``--model adaline`` for the delta-rule
variant.
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
    AdaLineSpec,
    BatchedGossipSimulator,
    DataArena,
    EngineConfig,
    PegasosSpec,
)
from gossipy_amd.simul import SimulationReport
from gossipy_amd.utils import plot_evaluation


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=100)
    ap.add_argument("--rounds", type=int, default=100)
    ap.add_argument("--model", choices=["pegasos", "adaline"], default="pegasos")
    ap.add_argument("--plot", action="store_true")
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    n, d = args.nodes, 57
    X, y = make_synthetic_classification((46 * n, d, 2), seed=42, margin=2.0)
    y = 2 * y.float() - 1  # {-1, +1} labels (AdaLine/Pegasos convention)
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
        model_size=d,
        drop_prob=0.1,
        online_prob=0.2,
        sync=False,  # async timeouts ~ N(delta, delta/10) (gossipy/node.py:79)
        sampling_eval=0.1,
        seed=42,
    )
    spec = (
        PegasosSpec(d_in=d, lam=0.01)
        if args.model == "pegasos"
        else AdaLineSpec(d_in=d, lr=0.01)
    )
    sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)

    ev = report.get_evaluation(False)
    print(f"final global eval: {ev[-1][1]}")
    if args.plot:
        plot_evaluation([ev], title=f"Ormandi 2013 ({args.model})")


if __name__ == "__main__":
    main()
