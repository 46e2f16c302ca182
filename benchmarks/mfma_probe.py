"""Tiny MFMA exercise for PMC capture: wide-MLP update on the engine."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import DataArena, MLPSpec, NodeStateArena, RandomTape
from gossipy_amd.engine.backend import HIPBackend, TorchBackend

dev = torch.device("cuda:0")
spec = MLPSpec(d_in=64, n_classes=16, hidden=(128,), lr=0.02, batch_size=0)
X, y = make_synthetic_classification((1600, 64, 16), seed=4)
shards = [(X[s], y[s]) for s in np.array_split(np.arange(1600), 64)]
data = DataArena.from_shards(shards, dev, global_eval=(X, y))
state = NodeStateArena(64, spec.D, dev)
TorchBackend().init_params(state, spec, RandomTape(6), 64)
be = HIPBackend()
nodes = torch.arange(64)
for _ in range(20):
    be.update(state, data, spec, nodes)
torch.cuda.synchronize()
print("mfma probe done")
