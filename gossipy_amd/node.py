"""Node behaviors: per-peer protocol logic of the object layer.

Parity layer for the reference's ``gossipy/node.py`` (per-class citations
below). These classes define the *semantics* of each gossip protocol; the
batched engine (:mod:`gossipy_amd.engine`) executes the same event sequences
as node-batched kernels and uses these implementations as its CPU oracle.

Deliberate divergence: ``CacheNeighNode.send`` draws the random slot with
``random.choice(list(...))`` — the reference calls ``random.choice(set(...))``
which raises ``TypeError`` (gossipy/node.py:449,463), i.e. that path is broken
as shipped; we fix the crash and keep the intended uniform-slot semantics.
"""

from __future__ import annotations

import random
from typing import Any, Dict, Iterable, Optional, Tuple, Union

import numpy as np
import torch

from . import CACHE, LOG
from .core import AntiEntropyProtocol, CreateModelMode, Message, MessageType, P2PNetwork
from .data import DataDispatcher
from .model.handler import ModelHandler, PartitionedTMH, SamplingTMH, WeightedTMH
from .model.sampling import TorchModelSampling

__all__ = [
    "GossipNode",
    "PassThroughNode",
    "CacheNeighNode",
    "SamplingBasedNode",
    "PartitioningBasedNode",
    "PENSNode",
    "All2AllGossipNode",
]

NodeData = Union[
    Tuple[torch.Tensor, Optional[torch.Tensor]],
    Tuple[np.ndarray, Optional[np.ndarray]],
    Tuple[Any, Any],
]


class GossipNode:
    """A generic gossip node (gossipy/node.py:34-286).

    Each node owns a data shard, a model handler, and a timeout offset
    ``delta``: synchronous nodes fire once per round at offset
    ``delta ~ U(0, round_len)``; asynchronous nodes fire every
    ``delta ~ N(round_len, round_len/10)`` timesteps.
    """

    def __init__(
        self,
        idx: int,
        data: NodeData,
        round_len: int,
        model_handler: ModelHandler,
        p2p_net: P2PNetwork,
        sync: bool = True,
    ):
        self.idx = idx
        self.data = data
        self.round_len = round_len
        self.model_handler = model_handler
        self.p2p_net = p2p_net
        self.sync = sync
        self.delta = (
            np.random.randint(0, round_len)
            if sync
            else int(np.random.normal(round_len, round_len / 10))
        )

    def init_model(self, local_train: bool = True, *args, **kwargs) -> None:
        """Initialize the local model, optionally followed by one local
        training pass (gossipy/node.py:82-94)."""
        self.model_handler.init()
        if local_train:
            self.model_handler._update(self.data[0])

    def get_peer(self) -> Optional[int]:
        """Pick a uniformly random peer (gossipy/node.py:96-109)."""
        peers = self.p2p_net.get_peers(self.idx)
        if not peers:
            LOG.warning("Node %d has no peers." % self.idx)
            return None
        return random.choice(peers)

    def timed_out(self, t: int) -> bool:
        """Whether the node fires at timestep ``t`` (gossipy/node.py:111-125)."""
        if self.sync:
            return (t % self.round_len) == self.delta
        return (t % self.delta) == 0

    def send(self, t: int, peer: int, protocol: AntiEntropyProtocol) -> Message:
        """Build the outgoing message for ``peer`` (gossipy/node.py:127-169).

        PUSH / PUSH_PULL snapshot the local model into the CACHE and ship the
        key; PULL ships an empty model request.
        """
        if protocol == AntiEntropyProtocol.PUSH:
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, peer, MessageType.PUSH, (key,))
        if protocol == AntiEntropyProtocol.PULL:
            return Message(t, self.idx, peer, MessageType.PULL, None)
        if protocol == AntiEntropyProtocol.PUSH_PULL:
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, peer, MessageType.PUSH_PULL, (key,))
        raise ValueError("Unknown protocol %s." % protocol)

    def receive(self, t: int, msg: Message) -> Optional[Message]:
        """Process an incoming message; return the REPLY for PULL/PUSH_PULL
        (gossipy/node.py:171-204)."""
        msg_type = msg.type
        key = msg.value[0] if msg.value else None
        if msg_type in (MessageType.PUSH, MessageType.REPLY, MessageType.PUSH_PULL):
            recv_model = CACHE.pop(key)
            self.model_handler(recv_model, self.data[0])
        if msg_type in (MessageType.PULL, MessageType.PUSH_PULL):
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, msg.sender, MessageType.REPLY, (key,))
        return None

    def evaluate(self, ext_data: Optional[Any] = None) -> Dict[str, float]:
        """Evaluate on the local test set, or on ``ext_data`` when given
        (gossipy/node.py:206-224)."""
        if ext_data is None:
            return self.model_handler.evaluate(self.data[1])
        return self.model_handler.evaluate(ext_data)

    def has_test(self) -> bool:
        """Whether the node holds a local test set (gossipy/node.py:227-238)."""
        if isinstance(self.data, tuple):
            return self.data[1] is not None
        return True

    def __repr__(self) -> str:
        return str(self)

    def __str__(self) -> str:
        return f"{self.__class__.__name__} #{self.idx} (Δ={self.delta})"

    @classmethod
    def generate(
        cls,
        data_dispatcher: DataDispatcher,
        p2p_net: P2PNetwork,
        model_proto: ModelHandler,
        round_len: int,
        sync: bool,
        **kwargs,
    ) -> Dict[int, "GossipNode"]:
        """Build the node dictionary, one handler copy per node
        (gossipy/node.py:247-286)."""
        return {
            idx: cls(
                idx=idx,
                data=data_dispatcher[idx],
                round_len=round_len,
                model_handler=model_proto.copy(),
                p2p_net=p2p_net,
                sync=sync,
                **kwargs,
            )
            for idx in range(p2p_net.size())
        }


class PassThroughNode(GossipNode):
    """Giaretta-2019 degree-aware pass-through node (gossipy/node.py:289-392).

    On receive, runs the full merge+update with probability
    ``min(1, deg_sender / deg_receiver)``; otherwise adopts the received model
    as-is (PASS mode). The sender's degree travels in the payload.
    """

    def __init__(self, idx, data, round_len, model_handler, p2p_net, sync=True):
        super().__init__(idx, data, round_len, model_handler, p2p_net, sync)
        self.n_neighs = p2p_net.size(idx)

    def send(self, t: int, peer: int, protocol: AntiEntropyProtocol) -> Message:
        if protocol == AntiEntropyProtocol.PUSH:
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, peer, MessageType.PUSH, (key, self.n_neighs))
        if protocol == AntiEntropyProtocol.PULL:
            return Message(t, self.idx, peer, MessageType.PULL, None)
        if protocol == AntiEntropyProtocol.PUSH_PULL:
            key = self.model_handler.caching(self.idx)
            return Message(
                t, self.idx, peer, MessageType.PUSH_PULL, (key, self.n_neighs)
            )
        raise ValueError("Unknown protocol %s." % protocol)

    def receive(self, t: int, msg: Message) -> Optional[Message]:
        msg_type = msg.type
        if msg_type in (MessageType.PUSH, MessageType.REPLY, MessageType.PUSH_PULL):
            key, deg = msg.value
            recv_model = CACHE.pop(key)
            if np.random.rand() < min(1, deg / self.n_neighs):
                self.model_handler(recv_model, self.data[0])
            else:  # pass-through: adopt without merge/update
                prev_mode = self.model_handler.mode
                self.model_handler.mode = CreateModelMode.PASS
                self.model_handler(recv_model, self.data[0])
                self.model_handler.mode = prev_mode
        if msg_type in (MessageType.PULL, MessageType.PUSH_PULL):
            key = self.model_handler.caching(self.idx)
            return Message(
                t, self.idx, msg.sender, MessageType.REPLY, (key, self.n_neighs)
            )
        return None


class CacheNeighNode(GossipNode):
    """Giaretta-2019 cache-per-neighbor node (gossipy/node.py:395-496).

    Received models are parked in a per-sender slot; only when the node's own
    timeout fires does it pop a random slot and merge+update with it before
    gossiping. (Crash fix vs reference noted in the module docstring.)
    """

    def __init__(self, idx, data, round_len, model_handler, p2p_net, sync=True):
        super().__init__(idx, data, round_len, model_handler, p2p_net, sync)
        self.local_cache: Dict[int, Any] = {}

    def _consume_random_slot(self) -> None:
        if self.local_cache:
            k = random.choice(list(self.local_cache.keys()))
            cached_model = CACHE.pop(self.local_cache[k])
            del self.local_cache[k]
            self.model_handler(cached_model, self.data[0])

    def send(self, t: int, peer: int, protocol: AntiEntropyProtocol) -> Message:
        if protocol == AntiEntropyProtocol.PUSH:
            self._consume_random_slot()
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, peer, MessageType.PUSH, (key,))
        if protocol == AntiEntropyProtocol.PULL:
            return Message(t, self.idx, peer, MessageType.PULL, None)
        if protocol == AntiEntropyProtocol.PUSH_PULL:
            self._consume_random_slot()
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, peer, MessageType.PUSH_PULL, (key,))
        raise ValueError("Unknown protocol %s." % protocol)

    def receive(self, t: int, msg: Message) -> Optional[Message]:
        sender, msg_type = msg.sender, msg.type
        key = msg.value[0] if msg.value else None
        if msg_type in (MessageType.PUSH, MessageType.REPLY, MessageType.PUSH_PULL):
            if sender in self.local_cache:
                CACHE.pop(self.local_cache[sender])
            self.local_cache[sender] = key
        if msg_type in (MessageType.PULL, MessageType.PUSH_PULL):
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, msg.sender, MessageType.REPLY, (key,))
        return None


class SamplingBasedNode(GossipNode):
    """Hegedus-2021 subsampled-merge node (gossipy/node.py:499-562).

    The payload carries the handler's ``sample_size``; the receiver draws the
    coordinate sample and merges only those coordinates.
    """

    model_handler: SamplingTMH

    def send(self, t: int, peer: int, protocol: AntiEntropyProtocol) -> Message:
        if protocol == AntiEntropyProtocol.PUSH:
            key = self.model_handler.caching(self.idx)
            return Message(
                t, self.idx, peer, MessageType.PUSH, (key, self.model_handler.sample_size)
            )
        if protocol == AntiEntropyProtocol.PULL:
            return Message(t, self.idx, peer, MessageType.PULL, None)
        if protocol == AntiEntropyProtocol.PUSH_PULL:
            key = self.model_handler.caching(self.idx)
            return Message(
                t,
                self.idx,
                peer,
                MessageType.PUSH_PULL,
                (key, self.model_handler.sample_size),
            )
        raise ValueError("Unknown protocol %s." % protocol)

    def receive(self, t: int, msg: Message) -> Optional[Message]:
        msg_type = msg.type
        if msg_type in (MessageType.PUSH, MessageType.REPLY, MessageType.PUSH_PULL):
            key, sample_size = msg.value
            recv_model = CACHE.pop(key)
            sample = TorchModelSampling.sample(sample_size, recv_model.model)
            self.model_handler(recv_model, self.data[0], sample)
        if msg_type in (MessageType.PULL, MessageType.PUSH_PULL):
            key = self.model_handler.caching(self.idx)
            return Message(
                t,
                self.idx,
                msg.sender,
                MessageType.REPLY,
                (key, self.model_handler.sample_size),
            )
        return None


class PartitioningBasedNode(GossipNode):
    """Hegedus-2021 partitioned-model node (gossipy/node.py:566-659).

    Every send picks a random partition id; the receiver merges only that
    partition (age-weighted).
    """

    model_handler: PartitionedTMH

    def _random_pid(self) -> int:
        return int(np.random.randint(0, self.model_handler.tm_partition.n_parts))

    def send(self, t: int, peer: int, protocol: AntiEntropyProtocol) -> Message:
        if protocol == AntiEntropyProtocol.PUSH:
            pid = self._random_pid()
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, peer, MessageType.PUSH, (key, pid))
        if protocol == AntiEntropyProtocol.PULL:
            return Message(t, self.idx, peer, MessageType.PULL, None)
        if protocol == AntiEntropyProtocol.PUSH_PULL:
            pid = self._random_pid()
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, peer, MessageType.PUSH_PULL, (key, pid))
        raise ValueError("Unknown protocol %s." % protocol)

    def receive(self, t: int, msg: Message) -> Optional[Message]:
        msg_type = msg.type
        if msg_type in (MessageType.PUSH, MessageType.REPLY, MessageType.PUSH_PULL):
            key, pid = msg.value
            recv_model = CACHE.pop(key)
            self.model_handler(recv_model, self.data[0], pid)
        if msg_type in (MessageType.PULL, MessageType.PUSH_PULL):
            pid = self._random_pid()
            key = self.model_handler.caching(self.idx)
            return Message(t, self.idx, msg.sender, MessageType.REPLY, (key, pid))
        return None


class PENSNode(GossipNode):
    """Onoszko-2021 PENS node (gossipy/node.py:663-785).

    Step 1 (first ``step1_rounds`` rounds): evaluate received models on the
    local *training* data, keep the top-``m_top`` of every batch of
    ``n_sampled`` received models, and count how often each peer makes the
    cut. Step 2: gossip only with the peers selected above expectation.
    PUSH-only.
    """

    def __init__(
        self,
        idx,
        data,
        round_len,
        model_handler,
        p2p_net,
        n_sampled: int = 10,
        m_top: int = 2,
        step1_rounds: int = 200,
        sync: bool = True,
    ):
        super().__init__(idx, data, round_len, model_handler, p2p_net, sync)
        assert self.model_handler.mode == CreateModelMode.MERGE_UPDATE, (
            "PENSNode can only be used with MERGE_UPDATE mode."
        )
        self.cache: Dict[int, Tuple[Any, float]] = {}
        self.n_sampled = n_sampled
        self.m_top = m_top
        known_nodes = p2p_net.get_peers(self.idx)
        if not known_nodes:
            known_nodes = [j for j in range(self.p2p_net.size()) if j != self.idx]
        self.neigh_counter = {i: 0 for i in known_nodes}
        self.selected = {i: 0 for i in known_nodes}
        self.step1_rounds = step1_rounds
        self.step = 1
        self.best_nodes: Optional[list] = None

    def _select_neighbors(self) -> None:
        self.best_nodes = [
            i
            for i, cnt in self.neigh_counter.items()
            if cnt > self.selected[i] * (self.m_top / self.n_sampled)
        ]

    def timed_out(self, t: int) -> bool:
        if self.step == 1 and (t // self.round_len) >= self.step1_rounds:
            self.step = 2
            self._select_neighbors()
        return super().timed_out(t)

    def get_peer(self) -> Optional[int]:
        if self.step == 1 or not self.best_nodes:
            peer = super().get_peer()
            if peer is None:
                return None
            if self.step == 1:
                self.selected[peer] += 1
            return peer
        return random.choice(self.best_nodes)

    def send(self, t: int, peer: int, protocol: AntiEntropyProtocol) -> Message:
        if protocol != AntiEntropyProtocol.PUSH:
            LOG.warning("PENSNode only supports PUSH protocol.")
        key = self.model_handler.caching(self.idx)
        return Message(t, self.idx, peer, MessageType.PUSH, (key,))

    def receive(self, t: int, msg: Message) -> None:
        sender, msg_type, key = msg.sender, msg.type, msg.value[0]
        if msg_type != MessageType.PUSH:
            LOG.warning("PENSNode only supports PUSH protocol.")
        if self.step == 1:
            evaluation = CACHE[key].evaluate(self.data[0])
            # one slot per peer: keep the latest model, score = -accuracy
            self.cache[sender] = (key, -evaluation["accuracy"])
            if len(self.cache) >= self.n_sampled:
                top_m = sorted(self.cache, key=lambda k: self.cache[k][1])[: self.m_top]
                recv_models = [CACHE.pop(self.cache[k][0]) for k in top_m]
                self.model_handler(recv_models, self.data[0])
                self.cache = {}
                for i in top_m:
                    self.neigh_counter[i] += 1
        else:
            recv_model = CACHE.pop(key)
            self.model_handler(recv_model, self.data[0])
        return None


class All2AllGossipNode(GossipNode):
    """Koloskova-2020 all-to-all averaging node (gossipy/node.py:789-869).

    Accumulates neighbor models in per-sender slots; at its own timeout it
    merges all of them with the supplied mixing weights, then pushes its model
    to *every* peer. PUSH-only.
    """

    model_handler: WeightedTMH

    def __init__(self, idx, data, round_len, model_handler, p2p_net, sync=True):
        super().__init__(idx, data, round_len, model_handler, p2p_net, sync)
        self.local_cache: Dict[int, Any] = {}

    def timed_out(self, t: int, weights: Iterable[float]) -> bool:
        tout = super().timed_out(t)
        if tout and self.local_cache:
            self.model_handler(
                [CACHE.pop(k) for k in self.local_cache.values()],
                self.data[0],
                weights,
            )
            self.local_cache = {}
        return tout

    def get_peers(self) -> list:
        """All reachable peers (the all-to-all fan-out set)."""
        return self.p2p_net.get_peers(self.idx)

    def send(self, t: int, peer: int, protocol: AntiEntropyProtocol) -> Message:
        if protocol == AntiEntropyProtocol.PUSH:
            return super().send(t, peer, protocol)
        raise ValueError("All2AllGossipNode only supports PUSH protocol.")

    def receive(self, t: int, msg: Message) -> None:
        sender, msg_type = msg.sender, msg.type
        key = msg.value[0] if msg.value else None
        if msg_type == MessageType.PUSH:
            if sender in self.local_cache:
                CACHE.pop(self.local_cache[sender])
            self.local_cache[sender] = key
        return None
