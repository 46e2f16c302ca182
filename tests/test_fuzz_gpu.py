"""Randomized config fuzz: the HIP engine on CUDA must match the torch
oracle on CPU across random protocol / fault / model-shape settings (same
seeds; the schedule is device-independent by construction)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from gossipy_amd.core import AntiEntropyProtocol, CreateModelMode
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import (
    AdaLineSpec,
    BatchedGossipSimulator,
    DataArena,
    EngineConfig,
    LogRegSpec,
    MLPSpec,
    PegasosSpec,
)

CUDA = torch.device("cuda:0")
CPU = torch.device("cpu")


def _rand_case(rng):
    d = int(rng.integers(5, 80))
    fam = rng.choice(["logreg", "mlp", "pegasos", "adaline", "logreg_part",
                      "logreg_samp"])
    k = int(rng.integers(2, 5))
    proto = rng.choice([
        AntiEntropyProtocol.PUSH,
        AntiEntropyProtocol.PULL,
        AntiEntropyProtocol.PUSH_PULL,
    ])
    mode = rng.choice([
        CreateModelMode.MERGE_UPDATE,
        CreateModelMode.UPDATE,
        CreateModelMode.UPDATE_MERGE,
        CreateModelMode.PASS,
    ])
    n_nodes = int(rng.integers(8, 48))
    kw = dict(
        n_nodes=n_nodes,
        delta=int(rng.integers(3, 12)),
        protocol=proto,
        model_size=d,
        drop_prob=float(rng.choice([0.0, 0.2])),
        online_prob=float(rng.choice([1.0, 0.8])),
        sync=bool(rng.random() < 0.7),
        sampling_eval=0.0,
        seed=int(rng.integers(0, 10**6)),
    )
    pm1 = False
    if fam in ("logreg_part", "logreg_samp") and mode == CreateModelMode.PASS:
        mode = CreateModelMode.MERGE_UPDATE  # PASS rejected by those handlers
    if fam == "logreg":
        spec = LogRegSpec(
            d_in=d, n_classes=k, lr=0.1, mode=mode,
            batch_size=int(rng.choice([0, 8, 32])),
            local_epochs=int(rng.integers(1, 3)),
            weight_decay=float(rng.choice([0.0, 0.01])),
        )
    elif fam == "mlp":
        spec = MLPSpec(
            d_in=d, n_classes=k, hidden=(int(rng.integers(8, 40)),),
            lr=0.05, mode=mode, batch_size=int(rng.choice([0, 16])),
        )
    elif fam == "pegasos":
        spec = PegasosSpec(d_in=d, lam=0.01, mode=mode)
        k, pm1 = 2, True
    elif fam == "adaline":
        spec = AdaLineSpec(d_in=d, lr=0.01, mode=mode)
        k, pm1 = 2, True
    elif fam == "logreg_part":
        if mode == CreateModelMode.UPDATE_MERGE and rng.random() < 0.5:
            mode = CreateModelMode.MERGE_UPDATE
        P = int(rng.integers(2, 7))
        spec = LogRegSpec(d_in=d, n_classes=k, lr=0.1, mode=mode, n_parts=P)
        kw["n_parts"] = P
    else:  # sampled
        spec = LogRegSpec(
            d_in=d, n_classes=k, lr=0.1, mode=mode,
            sample_size=float(rng.uniform(0.1, 0.6)),
        )
        kw["sampled"] = True
    return EngineConfig(**kw), spec, k, pm1


def _run(cfg, spec, k, pm1, device):
    n_samp = cfg.n_nodes * 8
    X, y = make_synthetic_classification((n_samp, spec.d_in, k), seed=cfg.seed)
    if pm1:
        y = 2 * y.float() - 1
    shards = [
        (X[s], y[s]) for s in np.array_split(np.arange(n_samp), cfg.n_nodes)
    ]
    data = DataArena.from_shards(shards, device, global_eval=(X, y))
    sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    sim.init_nodes()
    sim.start(n_rounds=3)
    if device.type == "cuda":
        torch.cuda.synchronize()
    return sim


@pytest.mark.parametrize("case", range(40))
def test_fuzz_hip_matches_oracle(case):
    rng = np.random.default_rng(1000 + case)
    cfg, spec, k, pm1 = _rand_case(rng)
    g = _run(cfg, spec, k, pm1, CUDA)
    c = _run(cfg, spec, k, pm1, CPU)
    assert torch.allclose(
        g.local_params().cpu(), c.local_params(), atol=2e-3, rtol=2e-3
    ), f"params diverge: {cfg} {spec}"
    assert torch.equal(g.state.ages.cpu(), c.state.ages), f"ages: {cfg} {spec}"


def _rand_runner_case(rng):
    """Random protocol-runner config (tokenized / cacheneigh / all2all /
    pens) over logreg."""
    kind = rng.choice(["tokenized", "cacheneigh", "all2all", "pens"])
    n_nodes = int(rng.integers(8, 32))
    d = int(rng.integers(8, 60))
    proto = AntiEntropyProtocol.PUSH
    if kind == "cacheneigh" and rng.random() < 0.5:
        proto = AntiEntropyProtocol.PUSH_PULL
    kw = dict(
        n_nodes=n_nodes,
        delta=int(rng.integers(3, 9)),
        protocol=proto,
        model_size=d,
        drop_prob=float(rng.choice([0.0, 0.2])),
        online_prob=float(rng.choice([1.0, 0.85])),
        sync=bool(rng.random() < 0.7),
        sampling_eval=0.0,
        seed=int(rng.integers(0, 10**6)),
    )
    spec = LogRegSpec(d_in=d, n_classes=2, lr=0.1)
    return kind, EngineConfig(**kw), spec


def _run_runner(kind, cfg, spec, device):
    from gossipy_amd.engine import (
        BatchedAll2AllGossipSimulator,
        BatchedCacheNeighGossipSimulator,
        BatchedPENSGossipSimulator,
        BatchedTokenizedGossipSimulator,
    )
    from gossipy_amd.flow_control import RandomizedTokenAccount

    n_samp = cfg.n_nodes * 8
    X, y = make_synthetic_classification((n_samp, spec.d_in, 2), seed=cfg.seed)
    shards = [
        (X[s], y[s]) for s in np.array_split(np.arange(n_samp), cfg.n_nodes)
    ]
    data = DataArena.from_shards(shards, device, global_eval=(X, y))
    if kind == "tokenized":
        sim = BatchedTokenizedGossipSimulator(
            cfg, spec, data, token_account=RandomizedTokenAccount(C=10, A=4),
            device=device,
        )
    elif kind == "cacheneigh":
        sim = BatchedCacheNeighGossipSimulator(cfg, spec, data, device=device)
    elif kind == "all2all":
        sim = BatchedAll2AllGossipSimulator(cfg, spec, data, device=device)
    else:
        sim = BatchedPENSGossipSimulator(
            cfg, spec, data, n_sampled=3, m_top=1, step1_rounds=2,
            device=device,
        )
    sim.init_nodes()
    sim.start(n_rounds=4)
    if device.type == "cuda":
        torch.cuda.synchronize()
    return sim


@pytest.mark.parametrize("case", range(16))
def test_fuzz_protocol_runners(case):
    rng = np.random.default_rng(70_000 + case)
    kind, cfg, spec = _rand_runner_case(rng)
    g = _run_runner(kind, cfg, spec, CUDA)
    c = _run_runner(kind, cfg, spec, CPU)
    assert torch.allclose(
        g.local_params().cpu(), c.local_params(), atol=2e-3, rtol=2e-3
    ), f"{kind}: {cfg}"
    assert torch.equal(g.state.ages.cpu(), c.state.ages), f"{kind}: {cfg}"
