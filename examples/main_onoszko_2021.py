"""Onoszko et al. 2021 — PENS: neighbor selection for non-IID data.

Object-layer equivalent of the reference's main_onoszko_2021.py: PENSNode
gossip where step 1 scores received models on local data to discover
similar peers and step 2 gossips only with the selected ones. The
reference uses a CNN on rotated CIFAR-10 (torchvision downloads are
unavailable offline); this runs the same protocol on a rotated synthetic
task with a TorchMLP — the non-IID structure (two client groups with
feature-rotated distributions) is preserved.

``--engine`` runs the batched MI355X PENS instead (logreg family, device
-side candidate scoring + top-m merges, one host sync at the step
boundary).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import numpy as np
import torch

from gossipy_amd import set_seed
from gossipy_amd.core import AntiEntropyProtocol, CreateModelMode, StaticP2PNetwork
from gossipy_amd.data import DataDispatcher, make_synthetic_classification
from gossipy_amd.data.handler import ClassificationDataHandler
from gossipy_amd.model.handler import TorchModelHandler
from gossipy_amd.model.nn import TorchMLP
from gossipy_amd.node import PENSNode
from gossipy_amd.simul import GossipSimulator, SimulationReport


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=20)
    ap.add_argument("--rounds", type=int, default=30)
    ap.add_argument("--engine", action="store_true",
                    help="batched MI355X PENS instead of the object layer")
    ap.add_argument("--engine-cnn", action="store_true",
                    help="batched engine with the reference CIFAR10Net CNN"
                         " (TorchModuleSpec; convs on MIOpen)")
    ap.add_argument("--pens", action="store_true",
                    help="with --engine-cnn: run the paper's PENS protocol"
                         " (vmap-scored candidates, top-m merges) instead"
                         " of plain merge-update gossip")
    args = ap.parse_args()

    if args.engine_cnn:
        return main_engine_cnn(args)
    if args.engine:
        return main_engine(args)

    set_seed(98765)
    d = 20
    X, y = make_synthetic_classification((50 * args.nodes, d, 2), seed=42, margin=2.0)
    # two rotated client groups (the reference rotates CIFAR images,
    # main_onoszko_2021.py:63-95)
    rot = torch.from_numpy(
        np.linalg.qr(np.random.default_rng(7).normal(size=(d, d)))[0]
    ).float()
    half = len(X) // 2
    X = torch.cat([X[:half], X[half:] @ rot])
    handler = ClassificationDataHandler(X, y, test_size=0.1, seed=42)
    dispatcher = DataDispatcher(handler, n=args.nodes, eval_on_user=True)

    topology = StaticP2PNetwork(args.nodes)
    nodes = PENSNode.generate(
        data_dispatcher=dispatcher,
        p2p_net=topology,
        model_proto=TorchModelHandler(
            net=TorchMLP(d, 2, (32,)),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        ),
        round_len=100,
        sync=True,
        n_sampled=10,
        m_top=2,
        step1_rounds=10,
    )
    simulator = GossipSimulator(
        nodes=nodes,
        data_dispatcher=dispatcher,
        delta=100,
        protocol=AntiEntropyProtocol.PUSH,
        sampling_eval=0.5,
    )
    report = SimulationReport()
    simulator.add_receiver(report)
    simulator.init_nodes(seed=42)
    simulator.start(n_rounds=args.rounds)
    print(f"final local eval: {report.get_evaluation(True)[-1][1]}")




def main_engine(args):
    from gossipy_amd.core import AntiEntropyProtocol
    from gossipy_amd.data import make_synthetic_classification
    from gossipy_amd.engine import (
        BatchedPENSGossipSimulator,
        DataArena,
        EngineConfig,
        LogRegSpec,
    )
    from gossipy_amd.simul import SimulationReport

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    n, d = args.nodes, 20
    X, y = make_synthetic_classification((50 * n, d, 2), seed=42, margin=2.0)
    idx = np.random.default_rng(42).permutation(len(y))
    cut = int(0.9 * len(y))
    shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], n)]
    data = DataArena.from_shards(
        shards, device, global_eval=(X[idx[cut:]], y[idx[cut:]])
    )
    cfg = EngineConfig(
        n_nodes=n, delta=100, protocol=AntiEntropyProtocol.PUSH,
        model_size=2 * d + 2, sampling_eval=0.25, seed=42,
    )
    sim = BatchedPENSGossipSimulator(
        cfg, LogRegSpec(d_in=d, n_classes=2, lr=0.1), data,
        n_sampled=6, m_top=2, step1_rounds=max(2, args.rounds // 3),
        device=device,
    )
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)
    print(f"final global eval: {report.get_evaluation(False)[-1][1]}")
    n_best = sum(len(b) for b in sim.scheduler.best_nodes or [])
    print(f"selected neighbors across nodes: {n_best}")




def main_engine_cnn(args):
    """CIFAR-shaped synthetic gossip with the reference CNN on the batched
    engine (BASELINE.json config 4 shape)."""
    from gossipy_amd.core import AntiEntropyProtocol
    from gossipy_amd.engine import (
        BatchedGossipSimulator,
        DataArena,
        EngineConfig,
        TorchModuleSpec,
    )
    from gossipy_amd.simul import SimulationReport
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

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    n = args.nodes
    rng = np.random.default_rng(42)
    per = 40
    labels = rng.integers(0, 10, size=n * per)
    x = rng.normal(0, 0.3, size=(len(labels), 3, 32, 32)).astype(np.float32)
    for c in range(10):
        x[labels == c, c % 3] += 0.8 + 0.25 * c
    X = torch.from_numpy(x.reshape(len(labels), -1))
    y = torch.from_numpy(labels).float()
    shards = [(X[i * per : (i + 1) * per], y[i * per : (i + 1) * per]) for i in range(n)]
    data = DataArena.from_shards(shards, device, global_eval=(X, y))
    spec = TorchModuleSpec(CIFAR10Net, input_shape=(3, 32, 32), lr=0.1, batch_size=32)
    cfg = EngineConfig(
        n_nodes=n, delta=10, protocol=AntiEntropyProtocol.PUSH,
        model_size=spec.D, sampling_eval=0.25, seed=42,
    )
    if getattr(args, "pens", False):
        # the paper's actual protocol: PENS neighbor selection with the CNN
        from gossipy_amd.engine import BatchedPENSGossipSimulator

        sim = BatchedPENSGossipSimulator(
            cfg, spec, data, n_sampled=6, m_top=2,
            step1_rounds=max(2, args.rounds // 3), device=device,
        )
    else:
        sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()
    sim.start(n_rounds=args.rounds)
    print(f"final global eval: {report.get_evaluation(False)[-1][1]}")
    if getattr(args, "pens", False):
        n_best = sum(len(b) for b in sim.scheduler.best_nodes or [])
        print(f"selected neighbors across nodes: {n_best}")


if __name__ == "__main__":
    main()
