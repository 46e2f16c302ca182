"""Measure the REFERENCE (makgyver/gossipy at /root/reference) on BASELINE
configs 3/4/5, exactly as BASELINE.md did for configs 1-2 — CPU, synthetic
data of the same shapes the engine benches use, rounds/s over a few rounds.

Config 3: Giaretta-shaped MLP 57->100->2, 1000 nodes, PUSH, full mesh,
          delta=100, sampling_eval=0.01 (cf. benchmarks/config_bench.py
          mlp_bench — engine row: 492 rounds/s).
Config 4: Onoszko CIFAR10Net CNN, 100 nodes, PUSH, delta=100, CIFAR-shaped
          synthetic (40/node), merge_update (cf. examples/main_onoszko_2021
          --engine-cnn shape).
Config 5: Hegedus-2020 MF (k=5, 3700 items, 100 ratings/user), 10,000 nodes
          (use --mf-nodes to reduce), PUSH, delta=100, sampling_eval=0.001
          (cf. benchmarks/mf10k_bench.py — engine row: 173 rounds/s).

Shims (documented in BASELINE.md): torchvision stub for the module-level
import in gossipy/data/__init__.py:13; roc_auc_score wrapped to np.float64
(reference calls .astype(float) on it, gossipy/model/handler.py:328).

Usage: python benchmarks/ref_baseline_345.py [--config 3|4|5] [--rounds N]
"""

import argparse
import json
import sys
import time
import types

import numpy as np
import torch

# --- shims so the 2021-era reference imports on the 2026 stack ----------
_tv = types.ModuleType("torchvision")
_tv.datasets = types.ModuleType("torchvision.datasets")
_tv.transforms = types.ModuleType("torchvision.transforms")
sys.modules.setdefault("torchvision", _tv)
sys.modules.setdefault("torchvision.datasets", _tv.datasets)
sys.modules.setdefault("torchvision.transforms", _tv.transforms)

import sklearn.metrics as _skm

_orig_auc = _skm.roc_auc_score
_skm.roc_auc_score = lambda *a, **k: np.float64(_orig_auc(*a, **k))

sys.path.insert(0, "/root/reference")

from gossipy import set_seed  # noqa: E402
from gossipy.core import AntiEntropyProtocol, CreateModelMode, StaticP2PNetwork  # noqa: E402
from gossipy.data import DataDispatcher, RecSysDataDispatcher  # noqa: E402
from gossipy.data.handler import ClassificationDataHandler, RecSysDataHandler  # noqa: E402
from gossipy.model.handler import MFModelHandler, TorchModelHandler  # noqa: E402
from gossipy.model.nn import TorchMLP  # noqa: E402
from gossipy.node import GossipNode  # noqa: E402
from gossipy.simul import GossipSimulator, SimulationReport  # noqa: E402


def synthetic_classification(shape, seed=42, margin=1.0):
    """Same generator as gossipy_amd.data.make_synthetic_classification
    (kept inline so this script never imports the new package next to the
    reference one)."""
    n, d, c = shape
    rng = np.random.default_rng(seed)
    centers = rng.normal(0.0, margin, size=(c, d))
    y = rng.integers(0, c, size=n)
    X = centers[y] + rng.normal(0.0, 1.0, size=(n, d))
    return torch.from_numpy(X.astype(np.float32)), torch.from_numpy(y.astype(np.int64))


def timed_rounds(sim, rounds):
    t0 = time.perf_counter()
    sim.start(n_rounds=rounds)
    return rounds / (time.perf_counter() - t0)


def config3(rounds):
    set_seed(42)
    n = 1000
    X, y = synthetic_classification((46 * n, 57, 2), seed=42)
    handler = ClassificationDataHandler(X, y, test_size=0.1, seed=42)
    dispatcher = DataDispatcher(handler, n=n, eval_on_user=False)
    nodes = GossipNode.generate(
        data_dispatcher=dispatcher,
        p2p_net=StaticP2PNetwork(n),
        model_proto=TorchModelHandler(
            net=TorchMLP(57, 2, (100,)),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.05},
            criterion=torch.nn.CrossEntropyLoss(),
            batch_size=32,
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        ),
        round_len=100,
        sync=True,
    )
    sim = GossipSimulator(
        nodes=nodes, data_dispatcher=dispatcher, delta=100,
        protocol=AntiEntropyProtocol.PUSH, sampling_eval=0.01,
    )
    sim.add_receiver(SimulationReport())
    t0 = time.perf_counter()
    sim.init_nodes(seed=42)
    init_s = time.perf_counter() - t0
    rps = timed_rounds(sim, rounds)
    print(json.dumps({
        "config": "ref-giaretta-shaped-mlp-57-100-2-1000n",
        "rounds_per_sec": round(rps, 3), "init_s": round(init_s, 1),
        "rounds": rounds,
    }), flush=True)


class CIFAR10Net(torch.nn.Module):
    """The reference PENS CNN (/root/reference/main_onoszko_2021.py:31-60)
    re-stated: conv 3->32->64->64 (3x3, maxpool 2) + fc 256->64->10."""

    def __init__(self):
        super().__init__()
        import torch.nn as nn
        self.conv1 = nn.Conv2d(3, 32, 3)
        self.pool = nn.MaxPool2d(2, 2)
        self.conv2 = nn.Conv2d(32, 64, 3)
        self.conv3 = nn.Conv2d(64, 64, 3)
        self.fc1 = nn.Linear(64 * 2 * 2, 64)
        self.fc2 = nn.Linear(64, 10)

    def forward(self, x):
        import torch.nn.functional as F
        x = x.view(-1, 3, 32, 32)
        x = self.pool(F.relu(self.conv1(x)))
        x = self.pool(F.relu(self.conv2(x)))
        x = self.pool(F.relu(self.conv3(x)))
        x = x.view(-1, 64 * 2 * 2)
        x = F.relu(self.fc1(x))
        return self.fc2(x)

    # TorchModel surface the reference handler expects
    def init_weights(self):
        pass

    def get_size(self):
        return sum(p.numel() for p in self.parameters())


def config4(rounds):
    set_seed(42)
    n, per = 100, 40
    rng = np.random.default_rng(42)
    labels = rng.integers(0, 10, size=n * per)
    x = rng.normal(0, 0.3, size=(len(labels), 3, 32, 32)).astype(np.float32)
    for c in range(10):
        x[labels == c, c % 3] += 0.8 + 0.25 * c
    X = torch.from_numpy(x.reshape(len(labels), -1))
    y = torch.from_numpy(labels.astype(np.int64))
    handler = ClassificationDataHandler(X, y, test_size=0.1, seed=42)
    dispatcher = DataDispatcher(handler, n=n, eval_on_user=False)
    nodes = GossipNode.generate(
        data_dispatcher=dispatcher,
        p2p_net=StaticP2PNetwork(n),
        model_proto=TorchModelHandler(
            net=CIFAR10Net(),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            batch_size=32,
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        ),
        round_len=100,
        sync=True,
    )
    sim = GossipSimulator(
        nodes=nodes, data_dispatcher=dispatcher, delta=100,
        protocol=AntiEntropyProtocol.PUSH, sampling_eval=0.05,
    )
    sim.add_receiver(SimulationReport())
    t0 = time.perf_counter()
    sim.init_nodes(seed=42)
    init_s = time.perf_counter() - t0
    rps = timed_rounds(sim, rounds)
    print(json.dumps({
        "config": "ref-onoszko-cifar10net-100n-mergeupdate",
        "rounds_per_sec": round(rps, 4), "init_s": round(init_s, 1),
        "rounds": rounds,
    }), flush=True)


def config5(rounds, n_users):
    set_seed(42)
    n_items, rpu = 3700, 100
    rng = np.random.default_rng(42)
    items = np.argsort(rng.random((n_users, n_items)), axis=1)[:, :rpu]
    raw = rng.normal(3.0, 1.2, size=(n_users, rpu))
    ratings_arr = np.clip(np.round(raw), 1, 5).astype(np.float64)
    ratings = {
        u: [(int(items[u, j]), float(ratings_arr[u, j])) for j in range(rpu)]
        for u in range(n_users)
    }
    handler = RecSysDataHandler(ratings, n_users, n_items, test_size=0.2, seed=42)
    dispatcher = RecSysDataDispatcher(handler)
    dispatcher.assign(seed=42)
    nodes = GossipNode.generate(
        data_dispatcher=dispatcher,
        p2p_net=StaticP2PNetwork(n_users),
        model_proto=MFModelHandler(
            dim=5, n_items=n_items, lam_reg=0.1, learning_rate=0.001,
            create_model_mode=CreateModelMode.UPDATE,
        ),
        round_len=100,
        sync=True,
    )
    sim = GossipSimulator(
        nodes=nodes, data_dispatcher=dispatcher, delta=100,
        protocol=AntiEntropyProtocol.PUSH, sampling_eval=0.001,
    )
    sim.add_receiver(SimulationReport())
    t0 = time.perf_counter()
    sim.init_nodes(seed=42)
    init_s = time.perf_counter() - t0
    rps = timed_rounds(sim, rounds)
    print(json.dumps({
        "config": f"ref-hegedus2020-mf-{n_users}nodes-ml1m-shaped",
        "rounds_per_sec": round(rps, 4), "init_s": round(init_s, 1),
        "rounds": rounds,
    }), flush=True)


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=int, default=0, help="3, 4, 5 (0 = all)")
    ap.add_argument("--rounds", type=int, default=3)
    ap.add_argument("--mf-nodes", type=int, default=10000)
    args = ap.parse_args()
    if args.config in (0, 3):
        config3(args.rounds)
    if args.config in (0, 4):
        config4(args.rounds)
    if args.config in (0, 5):
        config5(args.rounds, args.mf_nodes)
