"""Unit tests for the core runtime and protocol layer (cache, messages,
delays, topologies, mixing matrices)."""

import numpy as np
import pytest
import torch
from scipy.sparse import csr_matrix

from gossipy_amd import CACHE, CacheItem, CacheKey, GlobalSettings, Sizeable, set_seed
from gossipy_amd.core import (
    AntiEntropyProtocol,
    ConstantDelay,
    CreateModelMode,
    LinearDelay,
    Message,
    MessageType,
    MetropolisHastingsMixing,
    StaticP2PNetwork,
    UniformDelay,
    UniformMixing,
)


class _Fixed(Sizeable):
    def __init__(self, n):
        self.n = n

    def get_size(self):
        return self.n


class TestCache:
    def test_push_pop_single_ref(self):
        key = CacheKey(0, 1)
        CACHE.push(key, "model-a")
        assert len(CACHE) == 1
        assert CACHE.pop(key) == "model-a"
        assert len(CACHE) == 0, "refcount 0 must evict"

    def test_push_same_key_adds_ref_not_overwrite(self):
        key = CacheKey(0, 1)
        CACHE.push(key, "first")
        CACHE.push(CacheKey(0, 1), "second")
        assert len(CACHE) == 1
        # both pops return the FIRST value (reference semantics:
        # gossipy/__init__.py:297-311)
        assert CACHE.pop(CacheKey(0, 1)) == "first"
        assert len(CACHE) == 1
        assert CACHE.pop(CacheKey(0, 1)) == "first"
        assert len(CACHE) == 0

    def test_pop_unknown_key_returns_none(self):
        assert CACHE.pop(CacheKey(9, 9)) is None

    def test_getitem_does_not_deref(self):
        key = CacheKey(1, 2)
        CACHE.push(key, "v")
        assert CACHE[key] == "v"
        assert len(CACHE) == 1

    def test_load_get_cache_roundtrip(self):
        key = CacheKey(3, 4)
        CACHE.push(key, "v")
        raw = CACHE.get_cache()
        CACHE.clear()
        assert len(CACHE) == 0
        CACHE.load(raw)
        assert CACHE.pop(key) == "v"

    def test_cache_item_refcounting(self):
        item = CacheItem("x")
        assert item.is_referenced()
        item.add_ref()
        assert item.del_ref() == "x"
        assert item.is_referenced()
        assert item.del_ref() == "x"
        assert not item.is_referenced()

    def test_cache_key_size_dereferences_cache(self):
        key = CacheKey(0, 0)
        CACHE.push(key, _Fixed(42))
        assert key.get_size() == 42


class TestMessage:
    def test_size_of_sizeable_payload(self):
        msg = Message(0, 1, 2, MessageType.PUSH, (_Fixed(10),))
        assert msg.get_size() == 10

    def test_size_of_empty_payload_is_one(self):
        msg = Message(0, 1, 2, MessageType.PULL, None)
        assert msg.get_size() == 1

    def test_size_mixed_payload(self):
        msg = Message(0, 1, 2, MessageType.PUSH, (_Fixed(5), 3))
        assert msg.get_size() == 6  # 5 + one scalar


class TestDelays:
    def test_constant(self):
        d = ConstantDelay(4)
        assert d.get(Message(0, 0, 1, MessageType.PUSH, None)) == 4

    def test_uniform_bounds(self):
        set_seed(0)
        d = UniformDelay(2, 6)
        vals = {d.get(Message(0, 0, 1, MessageType.PUSH, None)) for _ in range(200)}
        assert vals <= set(range(2, 7))
        assert len(vals) > 1

    def test_linear_scales_with_size(self):
        d = LinearDelay(0.5, 3)
        msg = Message(0, 0, 1, MessageType.PUSH, (_Fixed(10),))
        assert d.get(msg) == int(0.5 * 10) + 3


class TestTopology:
    def test_fully_connected_default(self):
        net = StaticP2PNetwork(5)
        assert net.size() == 5
        assert net.get_peers(2) == [0, 1, 3, 4]

    def test_from_dense_adjacency(self):
        adj = np.array([[0, 1, 0], [1, 0, 1], [0, 1, 0]])
        net = StaticP2PNetwork(3, adj)
        assert net.get_peers(0) == [1]
        assert net.get_peers(1) == [0, 2]

    def test_from_csr(self):
        adj = csr_matrix(np.array([[0, 1], [1, 0]]))
        net = StaticP2PNetwork(2, adj)
        assert net.get_peers(0) == [1]

    def test_size_node_zero_quirk(self):
        # reference parity: `if node:` means node 0 returns the global size
        # (gossipy/core.py:346-349)
        adj = np.array([[0, 1, 0], [1, 0, 1], [0, 1, 0]])
        net = StaticP2PNetwork(3, adj)
        assert net.size(0) == 3
        assert net.size(1) == 2


class TestMixing:
    def test_uniform_weights_sum(self):
        net = StaticP2PNetwork(4)
        mix = UniformMixing(net)
        w = mix[1]
        assert len(w) == net.size(1) + 1
        assert np.allclose(w, 1.0 / (net.size(1) + 1))

    def test_mh_weights_shape(self):
        net = StaticP2PNetwork(4)
        mix = MetropolisHastingsMixing(net)
        w = mix[1]
        assert len(w) == 1 + len(net.get_peers(1))


class TestGlobals:
    def test_set_seed_reproducible(self):
        set_seed(7)
        a = (np.random.rand(3), torch.rand(3))
        set_seed(7)
        b = (np.random.rand(3), torch.rand(3))
        assert np.allclose(a[0], b[0])
        assert torch.allclose(a[1], b[1])

    def test_global_settings_singleton(self):
        s1, s2 = GlobalSettings(), GlobalSettings()
        assert s1 is s2
        s1.set_device("cpu")
        assert str(s2.get_device()) == "cpu"

    def test_enums_complete(self):
        assert {m.name for m in CreateModelMode} == {
            "UPDATE",
            "MERGE_UPDATE",
            "UPDATE_MERGE",
            "PASS",
        }
        assert {p.name for p in AntiEntropyProtocol} == {"PUSH", "PULL", "PUSH_PULL"}
        assert {m.name for m in MessageType} == {"PUSH", "PULL", "REPLY", "PUSH_PULL"}
