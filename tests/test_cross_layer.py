"""Cross-layer validation: the object layer (reference-parity path) and the
batched engine must produce comparable learning on the same task — the
object layer is the semantic oracle for the engine at the system level
(SURVEY.md §4 'the in-process simulator doubles as our own fake backend
oracle')."""

import numpy as np
import pytest
import torch

from gossipy_amd import set_seed
from gossipy_amd.core import AntiEntropyProtocol, CreateModelMode, StaticP2PNetwork
from gossipy_amd.data import DataDispatcher, make_synthetic_classification
from gossipy_amd.data.handler import ClassificationDataHandler
from gossipy_amd.engine import (
    BatchedGossipSimulator,
    DataArena,
    EngineConfig,
    LogRegSpec,
)
from gossipy_amd.model.handler import TorchModelHandler
from gossipy_amd.model.nn import LogisticRegression
from gossipy_amd.node import GossipNode
from gossipy_amd.simul import GossipSimulator, SimulationReport


N, D = 30, 57


def _object_layer_run(rounds=10):
    set_seed(98765)
    X, y = make_synthetic_classification((46 * N, D, 2), seed=42, margin=2.0)
    handler = ClassificationDataHandler(X, y, test_size=0.1, seed=42)
    dispatcher = DataDispatcher(handler, n=N, eval_on_user=False)
    nodes = GossipNode.generate(
        data_dispatcher=dispatcher,
        p2p_net=StaticP2PNetwork(N),
        model_proto=TorchModelHandler(
            net=LogisticRegression(D, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        ),
        round_len=10,
        sync=True,
    )
    sim = GossipSimulator(
        nodes=nodes,
        data_dispatcher=dispatcher,
        delta=10,
        protocol=AntiEntropyProtocol.PUSH,
        sampling_eval=0.3,
    )
    rep = SimulationReport()
    sim.add_receiver(rep)
    sim.init_nodes(seed=42)
    sim.start(n_rounds=rounds)
    return rep.get_evaluation(False)[-1][1]["accuracy"]


def _engine_run(rounds=10):
    X, y = make_synthetic_classification((46 * N, D, 2), seed=42, margin=2.0)
    idx = np.random.default_rng(42).permutation(len(y))
    cut = int(0.9 * len(y))
    shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], N)]
    data = DataArena.from_shards(
        shards, torch.device("cpu"), global_eval=(X[idx[cut:]], y[idx[cut:]])
    )
    cfg = EngineConfig(
        n_nodes=N, delta=10, protocol=AntiEntropyProtocol.PUSH,
        model_size=116, sampling_eval=0.3, seed=42,
    )
    sim = BatchedGossipSimulator(
        cfg, LogRegSpec(d_in=D, n_classes=2, lr=0.1), data
    )
    rep = SimulationReport()
    sim.add_receiver(rep)
    sim.init_nodes()
    sim.start(n_rounds=rounds)
    return rep.get_evaluation(False)[-1][1]["accuracy"]


@pytest.mark.timeout(600)
def test_object_layer_and_engine_agree():
    """Same task, same hyperparameters, independent RNG streams: both
    layers must converge to high accuracy (they share semantics, not
    bit-streams — the object layer mirrors the reference's global-RNG
    consumption, the engine uses the counter tape)."""
    obj_acc = _object_layer_run()
    eng_acc = _engine_run()
    assert obj_acc > 0.9, obj_acc
    assert eng_acc > 0.9, eng_acc
    assert abs(obj_acc - eng_acc) < 0.08, (obj_acc, eng_acc)
