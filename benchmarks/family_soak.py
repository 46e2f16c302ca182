"""Stability soak: every engine family runs a few hundred rounds on the
GPU; asserts finite parameters and flat memory."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol, CreateModelMode
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import (
    AdaLineSpec,
    BatchedAll2AllGossipSimulator,
    BatchedCacheNeighGossipSimulator,
    BatchedGossipSimulator,
    BatchedPENSGossipSimulator,
    BatchedTokenizedGossipSimulator,
    DataArena,
    EngineConfig,
    KMeansSpec,
    LogRegSpec,
    MFSpec,
    MLPSpec,
    PegasosSpec,
)
from gossipy_amd.flow_control import RandomizedTokenAccount

dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")


def class_data(n, d=57, k=2, pm1=False):
    X, y = make_synthetic_classification((46 * n, d, k), seed=1, margin=2.0)
    if pm1:
        y = 2 * y.float() - 1
    shards = [(X[s], y[s]) for s in np.array_split(np.arange(len(y)), n)]
    return DataArena.from_shards(shards, dev, global_eval=(X, y))


def run(name, sim, rounds):
    sim.init_nodes()
    sim.start(n_rounds=rounds)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    assert torch.isfinite(sim.local_params()).all(), name
    print(f"{name}: OK ({rounds} rounds)")


N = 128
cfgk = dict(n_nodes=N, delta=10, model_size=200, sampling_eval=0.05, seed=3)

run("logreg push_pull", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH_PULL, **cfgk),
    LogRegSpec(d_in=57, n_classes=2), class_data(N), device=dev), 300)
run("pegasos pull", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PULL, **cfgk),
    PegasosSpec(d_in=57), class_data(N, pm1=True), device=dev), 300)
run("adaline drop/online/delay", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, drop_prob=0.2,
                 online_prob=0.8, **cfgk),
    AdaLineSpec(d_in=57), class_data(N, pm1=True), device=dev), 300)
run("mlp mfma", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, **cfgk),
    MLPSpec(d_in=57, n_classes=2, hidden=(100,)), class_data(N), device=dev), 150)
run("partitioned async", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, sync=False, n_parts=4, **cfgk),
    LogRegSpec(d_in=57, n_classes=2, n_parts=4), class_data(N), device=dev), 200)
run("sampled", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, sampled=True, **cfgk),
    LogRegSpec(d_in=57, n_classes=2, sample_size=0.3), class_data(N), device=dev), 200)
run("tokenized", BatchedTokenizedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, **cfgk),
    LogRegSpec(d_in=57, n_classes=2), class_data(N),
    token_account=RandomizedTokenAccount(C=20, A=10), device=dev), 200)
run("all2all", BatchedAll2AllGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, n_nodes=32, delta=10,
                 model_size=200, sampling_eval=0.05, seed=3),
    LogRegSpec(d_in=57, n_classes=2), class_data(32), device=dev), 100)
run("cacheneigh", BatchedCacheNeighGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH_PULL, **cfgk),
    LogRegSpec(d_in=57, n_classes=2), class_data(N), device=dev), 200)
run("pens", BatchedPENSGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, **cfgk),
    LogRegSpec(d_in=57, n_classes=2), class_data(N),
    n_sampled=6, m_top=2, step1_rounds=30, device=dev), 100)

mf_items = 200
rng = np.random.default_rng(4)
items = np.argsort(rng.random((N, mf_items)), axis=1)[:, :40]
ratings = np.clip(np.round(rng.normal(3, 1.2, size=(N, 40))), 1, 5).astype(np.float32)
mf_data = DataArena(
    torch.from_numpy(items[..., None].astype(np.float32)).to(dev),
    torch.from_numpy(ratings).to(dev),
    torch.full((N,), 40, dtype=torch.int32).to(dev),
)
run("mf", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, n_nodes=N, delta=10,
                 model_size=1200, sampling_eval=0.0, seed=5),
    MFSpec(k=5, n_items=mf_items), mf_data, device=dev), 200)

km_rng = np.random.default_rng(9)
centers = km_rng.normal(0, 4, size=(4, 16))
shards = []
for _ in range(N):
    lb = km_rng.integers(0, 4, size=30)
    xx = centers[lb] + km_rng.normal(0, 0.4, size=(30, 16))
    shards.append((torch.from_numpy(xx).float(), torch.from_numpy(lb).float()))
km_data = DataArena.from_shards(shards, dev)
run("kmeans", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, n_nodes=N, delta=10,
                 model_size=64, sampling_eval=0.0, seed=6),
    KMeansSpec(k=4, dim=16, mode=CreateModelMode.MERGE_UPDATE),
    km_data, device=dev), 200)

run("kmeans hungarian", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, n_nodes=N, delta=10,
                 model_size=64, sampling_eval=0.0, seed=6),
    KMeansSpec(k=4, dim=16, mode=CreateModelMode.MERGE_UPDATE,
               matching="hungarian"),
    km_data, device=dev), 150)


# torchmod CNN (vmap-batched + hipGraph trajectories) — r2 paths
def cnn_factory():
    import torch.nn as nn
    import torch.nn.functional as F

    class Tiny(nn.Module):
        def __init__(self):
            super().__init__()
            self.conv1 = nn.Conv2d(3, 8, 3)
            self.pool = nn.MaxPool2d(2, 2)
            self.fc = nn.Linear(8 * 15 * 15, 10)

        def forward(self, x):
            x = self.pool(F.relu(self.conv1(x)))
            return self.fc(x.view(-1, 8 * 15 * 15))

    return Tiny()


from gossipy_amd.engine import TorchModuleSpec  # noqa: E402

cnn_rng = np.random.default_rng(11)
lbl = cnn_rng.integers(0, 10, size=32 * 12)
cx = cnn_rng.normal(0, 0.3, size=(len(lbl), 3, 32, 32)).astype(np.float32)
CX = torch.from_numpy(cx.reshape(len(lbl), -1))
CY = torch.from_numpy(lbl).float()
cnn_shards = [(CX[i * 12 : (i + 1) * 12], CY[i * 12 : (i + 1) * 12]) for i in range(32)]
cnn_data = DataArena.from_shards(cnn_shards, dev, global_eval=(CX, CY))
cnn_spec = TorchModuleSpec(cnn_factory, input_shape=(3, 32, 32), lr=0.05,
                           batch_size=6)
run("torchmod cnn", BatchedGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, n_nodes=32, delta=10,
                 model_size=cnn_spec.D, sampling_eval=0.1, seed=13),
    cnn_spec, cnn_data, device=dev), 120)
run("pens cnn", BatchedPENSGossipSimulator(
    EngineConfig(protocol=AntiEntropyProtocol.PUSH, n_nodes=32, delta=10,
                 model_size=cnn_spec.D, sampling_eval=0.0, seed=14),
    TorchModuleSpec(cnn_factory, input_shape=(3, 32, 32), lr=0.05,
                    batch_size=6),
    cnn_data, n_sampled=4, m_top=2, step1_rounds=40, device=dev), 100)

if dev.type == "cuda":
    print(f"memory allocated: {torch.cuda.memory_allocated()/1e6:.1f} MB")
print("ALL FAMILIES STABLE")
