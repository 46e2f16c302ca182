"""Probe: did hipGraph capture of the batched CNN SGD succeed, and what
does a pure replay cost vs the eager body?"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from benchmarks.onoszko_bench import cifar10net_factory
from gossipy_amd.engine import DataArena, TorchModuleSpec
from gossipy_amd.engine.arena import NodeStateArena
from gossipy_amd.engine.backend import TorchBackend


def main():
    dev = torch.device("cuda:0")
    spec = TorchModuleSpec(
        cifar10net_factory, input_shape=(3, 32, 32), lr=0.1, batch_size=32
    )
    n, per = 64, 40
    rng = np.random.default_rng(0)
    X = torch.from_numpy(
        rng.normal(0, 0.3, size=(n * per, 3 * 32 * 32)).astype(np.float32)
    )
    y = torch.from_numpy(rng.integers(0, 10, size=n * per).astype(np.float32))
    shards = [(X[i * per : (i + 1) * per], y[i * per : (i + 1) * per]) for i in range(n)]
    data = DataArena.from_shards(shards, dev, global_eval=(X, y))
    st = NodeStateArena(n, spec.D, dev)
    torch.manual_seed(0)
    st.params.normal_(0, 0.05)
    be = TorchBackend()
    nodes = torch.arange(32)

    # trigger capture
    be._update_torchmod(st.params, st.ages, data, spec, nodes)
    cache = getattr(be, "_tm_graphs", {})
    print("graph cache entries:", {k[1:]: (v is not None) for k, v in cache.items()})

    # steady-state full call
    for _ in range(3):
        be._update_torchmod(st.params, st.ages, data, spec, nodes)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        be._update_torchmod(st.params, st.ages, data, spec, nodes)
    torch.cuda.synchronize()
    print("full update call ms:", (time.perf_counter() - t0) / 20 * 1000)

    # pure replay
    for k, v in cache.items():
        if v is None:
            continue
        graph = v[0]
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(20):
            graph.replay()
        torch.cuda.synchronize()
        print(f"pure replay {k[1:]} ms:", (time.perf_counter() - t0) / 20 * 1000)


if __name__ == "__main__":
    main()
