"""Unit tests for the node-behavior layer (protocol send/receive, CACHE
interplay, PENS peer selection, all-to-all slots)."""

import numpy as np
import pytest
import torch

from gossipy_amd import CACHE, set_seed
from gossipy_amd.core import (
    AntiEntropyProtocol,
    CreateModelMode,
    MessageType,
    StaticP2PNetwork,
    UniformMixing,
)
from gossipy_amd.data import DataDispatcher, make_synthetic_classification
from gossipy_amd.data.handler import ClassificationDataHandler
from gossipy_amd.model.handler import (
    PegasosHandler,
    SamplingTMH,
    PartitionedTMH,
    TorchModelHandler,
    WeightedTMH,
)
from gossipy_amd.model.nn import AdaLine, LogisticRegression
from gossipy_amd.model.sampling import TorchModelPartition
from gossipy_amd.node import (
    All2AllGossipNode,
    CacheNeighNode,
    GossipNode,
    PartitioningBasedNode,
    PassThroughNode,
    PENSNode,
    SamplingBasedNode,
)


def _dispatcher(n=4, d=6, binary_pm1=False, eval_on_user=True):
    X, y = make_synthetic_classification((80, d, 2), seed=0)
    if binary_pm1:
        y = 2 * y.float() - 1
    handler = ClassificationDataHandler(X, y, test_size=0.2, seed=0)
    return DataDispatcher(handler, n=n, eval_on_user=eval_on_user)


def _pegasos_proto(d=6, mode=CreateModelMode.MERGE_UPDATE):
    return PegasosHandler(net=AdaLine(d), learning_rate=0.01, create_model_mode=mode)


def _tmh_proto(d=6, mode=CreateModelMode.MERGE_UPDATE):
    return TorchModelHandler(
        net=LogisticRegression(d, 2),
        optimizer=torch.optim.SGD,
        optimizer_params={"lr": 0.1},
        criterion=torch.nn.CrossEntropyLoss(),
        create_model_mode=mode,
    )


def _gen(cls, proto, n=4, sync=True, **kwargs):
    set_seed(1)
    disp = _dispatcher(n=n, binary_pm1=isinstance(proto, PegasosHandler))
    net = StaticP2PNetwork(n)
    return cls.generate(
        data_dispatcher=disp,
        p2p_net=net,
        model_proto=proto,
        round_len=10,
        sync=sync,
        **kwargs,
    )


class TestGossipNode:
    def test_generate_builds_independent_handlers(self):
        nodes = _gen(GossipNode, _pegasos_proto())
        assert len(nodes) == 4
        assert nodes[0].model_handler is not nodes[1].model_handler

    def test_timed_out_sync(self):
        nodes = _gen(GossipNode, _pegasos_proto(), sync=True)
        node = nodes[0]
        assert node.timed_out(node.delta)
        assert node.timed_out(node.delta + 10)
        assert not node.timed_out(node.delta + 1)

    def test_get_peer_excludes_self(self):
        nodes = _gen(GossipNode, _pegasos_proto())
        for _ in range(20):
            assert nodes[2].get_peer() != 2

    def test_push_send_caches_model(self):
        nodes = _gen(GossipNode, _pegasos_proto())
        nodes[0].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        assert msg.type == MessageType.PUSH
        key = msg.value[0]
        assert CACHE[key] is not None
        assert len(CACHE) == 1

    def test_pull_send_has_no_payload(self):
        nodes = _gen(GossipNode, _pegasos_proto())
        nodes[0].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PULL)
        assert msg.type == MessageType.PULL
        assert msg.value is None
        assert len(CACHE) == 0

    def test_receive_push_pops_cache_and_merges(self):
        nodes = _gen(GossipNode, _pegasos_proto())
        nodes[0].init_model()
        nodes[1].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        reply = nodes[1].receive(1, msg)
        assert reply is None
        assert len(CACHE) == 0, "push key must be popped on delivery"

    def test_receive_push_pull_returns_reply(self):
        nodes = _gen(GossipNode, _pegasos_proto())
        nodes[0].init_model()
        nodes[1].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH_PULL)
        reply = nodes[1].receive(1, msg)
        assert reply is not None and reply.type == MessageType.REPLY
        assert reply.receiver == 0
        assert len(CACHE) == 1, "reply key cached"

    def test_evaluate_local_and_external(self):
        nodes = _gen(GossipNode, _pegasos_proto())
        nodes[0].init_model()
        res = nodes[0].evaluate()
        assert "accuracy" in res
        X, y = make_synthetic_classification((20, 6, 2), seed=3)
        res2 = nodes[0].evaluate((X, 2 * y.float() - 1))
        assert "accuracy" in res2


class TestPassThroughNode:
    def test_send_carries_degree(self):
        nodes = _gen(PassThroughNode, _pegasos_proto())
        nodes[0].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        key, deg = msg.value
        assert deg == nodes[0].n_neighs
        CACHE.pop(key)

    def test_receive_always_processes(self):
        set_seed(0)
        nodes = _gen(PassThroughNode, _pegasos_proto())
        nodes[0].init_model()
        nodes[1].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        nodes[1].receive(1, msg)
        assert len(CACHE) == 0


class TestCacheNeighNode:
    def test_receive_parks_model_in_slot(self):
        nodes = _gen(CacheNeighNode, _pegasos_proto())
        nodes[0].init_model()
        nodes[1].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        # sender had an empty cache: model snapshot stays in CACHE
        nodes[1].receive(1, msg)
        assert 0 in nodes[1].local_cache
        assert len(CACHE) == 1, "parked, not consumed"

    def test_send_consumes_random_slot(self):
        set_seed(0)
        nodes = _gen(CacheNeighNode, _pegasos_proto())
        nodes[0].init_model()
        nodes[1].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        nodes[1].receive(1, msg)
        nodes[1].send(2, 0, AntiEntropyProtocol.PUSH)
        assert nodes[1].local_cache == {}

    def test_duplicate_sender_replaces_slot(self):
        nodes = _gen(CacheNeighNode, _pegasos_proto())
        nodes[0].init_model()
        nodes[1].init_model()
        m1 = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        nodes[1].receive(1, m1)
        nodes[0].model_handler.n_updates += 1  # new version
        m2 = nodes[0].send(2, 1, AntiEntropyProtocol.PUSH)
        nodes[1].receive(3, m2)
        assert len(nodes[1].local_cache) == 1
        assert len(CACHE) == 1


class TestSamplingBasedNode:
    def test_roundtrip(self):
        proto = SamplingTMH(
            sample_size=0.5,
            net=LogisticRegression(6, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
        )
        nodes = _gen(SamplingBasedNode, proto)
        nodes[0].init_model()
        nodes[1].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        key, ss = msg.value
        assert ss == 0.5
        nodes[1].receive(1, msg)
        assert len(CACHE) == 0


class TestPartitioningBasedNode:
    def test_roundtrip(self):
        set_seed(0)
        net = LogisticRegression(6, 2)
        proto = PartitionedTMH(
            net=net,
            tm_partition=TorchModelPartition(net, 3),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
        )
        nodes = _gen(PartitioningBasedNode, proto)
        nodes[0].init_model()
        nodes[1].init_model()
        msg = nodes[0].send(0, 1, AntiEntropyProtocol.PUSH)
        key, pid = msg.value
        assert 0 <= pid < 3
        nodes[1].receive(1, msg)
        assert len(CACHE) == 0


class TestPENSNode:
    def test_step1_counts_and_batch_merge(self):
        set_seed(2)
        proto = _tmh_proto()
        nodes = _gen(PENSNode, proto, n=4, n_sampled=2, m_top=1, step1_rounds=1)
        for n in nodes.values():
            n.init_model()
        recv = nodes[3]
        # two pushes fill the sample batch of size n_sampled=2
        m0 = nodes[0].send(0, 3, AntiEntropyProtocol.PUSH)
        recv.receive(0, m0)
        assert len(recv.cache) == 1
        m1 = nodes[1].send(0, 3, AntiEntropyProtocol.PUSH)
        recv.receive(0, m1)
        assert recv.cache == {}, "batch merged and reset"
        assert sum(recv.neigh_counter.values()) == 1  # top-1 counted

    def test_step_transition_selects_best(self):
        set_seed(2)
        proto = _tmh_proto()
        nodes = _gen(PENSNode, proto, n=4, n_sampled=2, m_top=1, step1_rounds=1)
        node = nodes[0]
        node.neigh_counter = {1: 5, 2: 0, 3: 0}
        node.selected = {1: 5, 2: 5, 3: 5}
        assert node.step == 1
        node.timed_out(node.round_len * 1 + node.delta)  # past step1_rounds
        assert node.step == 2
        assert node.best_nodes == [1]
        for _ in range(10):
            assert node.get_peer() == 1


class TestAll2AllNode:
    def test_accumulate_then_merge_at_timeout(self):
        set_seed(3)
        proto = WeightedTMH(
            net=LogisticRegression(6, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            create_model_mode=CreateModelMode.MERGE_UPDATE,
        )
        nodes = _gen(All2AllGossipNode, proto)
        for n in nodes.values():
            n.init_model()
        recv = nodes[0]
        for s in (1, 2):
            msg = nodes[s].send(0, 0, AntiEntropyProtocol.PUSH)
            recv.receive(0, msg)
        assert len(recv.local_cache) == 2
        net = StaticP2PNetwork(4)
        w = UniformMixing(net)[0][:3]  # self + 2 cached
        assert recv.timed_out(recv.delta, w)
        assert recv.local_cache == {}
        assert len(CACHE) == 0
