"""Integration tests for the three simulators: vanilla, tokenized, all2all,
plus report accounting and checkpoint save/load."""

import os

import numpy as np
import pytest
import torch

from gossipy_amd import CACHE, set_seed
from gossipy_amd.core import (
    AntiEntropyProtocol,
    ConstantDelay,
    CreateModelMode,
    StaticP2PNetwork,
    UniformDelay,
    UniformMixing,
)
from gossipy_amd.data import DataDispatcher, make_synthetic_classification
from gossipy_amd.data.handler import ClassificationDataHandler
from gossipy_amd.flow_control import RandomizedTokenAccount
from gossipy_amd.model.handler import PegasosHandler, TorchModelHandler, WeightedTMH
from gossipy_amd.model.nn import AdaLine, LogisticRegression
from gossipy_amd.node import All2AllGossipNode, GossipNode
from gossipy_amd.simul import (
    All2AllGossipSimulator,
    GossipSimulator,
    SimulationReport,
    TokenizedGossipSimulator,
)


def _setup(n=20, d=6, margin=2.0, pegasos=True, eval_on_user=False):
    set_seed(42)
    X, y = make_synthetic_classification((200, d, 2), seed=0, margin=margin)
    if pegasos:
        y = 2 * y.float() - 1
    handler = ClassificationDataHandler(X, y, test_size=0.1, seed=0)
    dispatcher = DataDispatcher(handler, n=n, eval_on_user=eval_on_user)
    topology = StaticP2PNetwork(n)
    if pegasos:
        proto = PegasosHandler(
            net=AdaLine(d),
            learning_rate=0.01,
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        )
    else:
        proto = TorchModelHandler(
            net=LogisticRegression(d, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        )
    return dispatcher, topology, proto


class TestGossipSimulator:
    def test_learning_curve_rises(self):
        dispatcher, topology, proto = _setup()
        nodes = GossipNode.generate(
            data_dispatcher=dispatcher,
            p2p_net=topology,
            model_proto=proto,
            round_len=10,
            sync=True,
        )
        sim = GossipSimulator(
            nodes=nodes,
            data_dispatcher=dispatcher,
            delta=10,
            protocol=AntiEntropyProtocol.PUSH,
        )
        report = SimulationReport()
        sim.add_receiver(report)
        sim.init_nodes(seed=42)
        sim.start(n_rounds=10)
        evals = report.get_evaluation(False)
        assert len(evals) == 10
        assert evals[-1][1]["accuracy"] > 0.8

    def test_push_pull_and_delay(self):
        dispatcher, topology, proto = _setup(n=10)
        nodes = GossipNode.generate(
            data_dispatcher=dispatcher,
            p2p_net=topology,
            model_proto=proto,
            round_len=10,
            sync=False,
        )
        sim = GossipSimulator(
            nodes=nodes,
            data_dispatcher=dispatcher,
            delta=10,
            protocol=AntiEntropyProtocol.PUSH_PULL,
            delay=UniformDelay(0, 3),
            drop_prob=0.1,
            online_prob=0.9,
        )
        report = SimulationReport()
        sim.add_receiver(report)
        sim.init_nodes(seed=1)
        sim.start(n_rounds=5)
        assert report._sent_messages > 0

    def test_report_accounting_no_drops(self):
        dispatcher, topology, proto = _setup(n=10)
        nodes = GossipNode.generate(
            data_dispatcher=dispatcher,
            p2p_net=topology,
            model_proto=proto,
            round_len=10,
            sync=True,
        )
        sim = GossipSimulator(
            nodes=nodes,
            data_dispatcher=dispatcher,
            delta=10,
            protocol=AntiEntropyProtocol.PUSH,
            delay=ConstantDelay(0),
        )
        report = SimulationReport()
        sim.add_receiver(report)
        sim.init_nodes(seed=1)
        sim.start(n_rounds=3)
        # sync nodes with no drops: one send per node per round, all delivered
        assert report._sent_messages == 10 * 3
        assert report._failed_messages == 0
        assert len(CACHE) == 0, "all snapshots consumed"

    def test_receivers_are_per_instance(self):
        dispatcher, topology, proto = _setup(n=4)
        nodes = GossipNode.generate(
            data_dispatcher=dispatcher,
            p2p_net=topology,
            model_proto=proto,
            round_len=10,
            sync=True,
        )
        s1 = GossipSimulator(nodes, dispatcher, 10, AntiEntropyProtocol.PUSH)
        s2 = GossipSimulator(nodes, dispatcher, 10, AntiEntropyProtocol.PUSH)
        r = SimulationReport()
        s1.add_receiver(r)
        assert r not in s2._receivers, "receiver list must not be shared (ref bug)"

    def test_save_load_roundtrip(self, tmp_path):
        dispatcher, topology, proto = _setup(n=6)
        nodes = GossipNode.generate(
            data_dispatcher=dispatcher,
            p2p_net=topology,
            model_proto=proto,
            round_len=10,
            sync=True,
        )
        sim = GossipSimulator(
            nodes=nodes,
            data_dispatcher=dispatcher,
            delta=10,
            protocol=AntiEntropyProtocol.PUSH,
        )
        sim.init_nodes(seed=7)
        sim.start(n_rounds=2)
        path = os.path.join(tmp_path, "ckpt.dill")
        sim.save(path)
        CACHE.clear()
        loaded = GossipSimulator.load(path)
        assert loaded.n_nodes == sim.n_nodes
        assert loaded.delta == sim.delta
        # resumed simulation continues without error
        loaded.start(n_rounds=1)


class TestTokenizedSimulator:
    def test_runs_and_learns(self):
        dispatcher, topology, proto = _setup(n=10, pegasos=False)
        nodes = GossipNode.generate(
            data_dispatcher=dispatcher,
            p2p_net=topology,
            model_proto=proto,
            round_len=10,
            sync=True,
        )
        sim = TokenizedGossipSimulator(
            nodes=nodes,
            data_dispatcher=dispatcher,
            token_account=RandomizedTokenAccount(C=20, A=10),
            utility_fun=lambda mh1, mh2, msg: 1,
            delta=10,
            protocol=AntiEntropyProtocol.PUSH,
            sampling_eval=0.5,
        )
        report = SimulationReport()
        sim.add_receiver(report)
        sim.init_nodes(seed=2)
        sim.start(n_rounds=10)
        assert len(sim.accounts) == 10
        evals = report.get_evaluation(False)
        assert evals, "evaluations recorded"


class TestAll2AllSimulator:
    def test_runs_with_mixing(self):
        set_seed(5)
        X, y = make_synthetic_classification((120, 6, 2), seed=5, margin=2.0)
        handler = ClassificationDataHandler(X, y, test_size=0.1, seed=5)
        dispatcher = DataDispatcher(handler, n=6, eval_on_user=False)
        topology = StaticP2PNetwork(6)
        proto = WeightedTMH(
            net=LogisticRegression(6, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        )
        nodes = All2AllGossipNode.generate(
            data_dispatcher=dispatcher,
            p2p_net=topology,
            model_proto=proto,
            round_len=10,
            sync=True,
        )
        sim = All2AllGossipSimulator(
            nodes=nodes,
            data_dispatcher=dispatcher,
            delta=10,
            protocol=AntiEntropyProtocol.PUSH,
        )
        report = SimulationReport()
        sim.add_receiver(report)
        sim.init_nodes(seed=5)
        sim.start(UniformMixing(topology), n_rounds=5)
        evals = report.get_evaluation(False)
        assert len(evals) == 5
        assert evals[-1][1]["accuracy"] > 0.6


def test_throughput_tracer_records_windows():
    import time

    from gossipy_amd.simul import ThroughputTracer

    tr = ThroughputTracer()
    for t in (99, 199, 299):
        tr.update_timestep(t)
        time.sleep(0.01)
    assert len(tr.round_times) == 2
    assert tr.rounds_per_sec > 0
    tr.update_end()
