import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from benchmarks.onoszko_bench import cifar10net_factory
from gossipy_amd.core import AntiEntropyProtocol
from gossipy_amd.engine import BatchedGossipSimulator, DataArena, EngineConfig, TorchModuleSpec

def build():
    n, per = 100, 40
    rng = np.random.default_rng(42)
    labels = rng.integers(0, 10, size=n*per)
    x = rng.normal(0,0.3,size=(len(labels),3,32,32)).astype(np.float32)
    X = torch.from_numpy(x.reshape(len(labels),-1)); y = torch.from_numpy(labels).float()
    shards=[(X[i*per:(i+1)*per], y[i*per:(i+1)*per]) for i in range(n)]
    dev = torch.device("cuda:0")
    data = DataArena.from_shards(shards, dev, global_eval=(X,y))
    spec = TorchModuleSpec(cifar10net_factory, input_shape=(3,32,32), lr=0.1, batch_size=32)
    cfg = EngineConfig(n_nodes=n, delta=100, protocol=AntiEntropyProtocol.PUSH, model_size=spec.D, sampling_eval=0.05, seed=42)
    sim = BatchedGossipSimulator(cfg, spec, data, device=dev)
    sim.init_nodes()
    return sim

sim = build()
sim._evaluate = lambda *a, **k: None   # no eval
sim.start(n_rounds=3)
torch.cuda.synchronize(); t0=time.perf_counter()
sim.start(n_rounds=10)
torch.cuda.synchronize()
print("no-eval ms/round:", (time.perf_counter()-t0)/10*1000)

sim2 = build()
sim2.start(n_rounds=3)
torch.cuda.synchronize(); t0=time.perf_counter()
sim2.start(n_rounds=10)
torch.cuda.synchronize()
print("with-eval ms/round:", (time.perf_counter()-t0)/10*1000)

# steady-state single batched update timing
from gossipy_amd.engine.backend import TorchBackend
be = sim2.backend
nodes = torch.arange(32)
for _ in range(3):
    be._update_torchmod(sim2.state.params, sim2.state.ages, sim2.data, sim2.spec, nodes)
torch.cuda.synchronize(); t0=time.perf_counter()
for _ in range(10):
    be._update_torchmod(sim2.state.params, sim2.state.ages, sim2.data, sim2.spec, nodes)
torch.cuda.synchronize()
print("update B=32 ms:", (time.perf_counter()-t0)/10*1000)
