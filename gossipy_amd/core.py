"""Protocol and network model: enums, messages, delays, topologies, mixing.

Parity layer for the reference's ``gossipy/core.py`` (cited per class below).
All randomness goes through numpy's global RNG like the reference so seeded
runs are reproducible; the batched engine uses its own counter-based tape
instead (:mod:`gossipy_amd.engine.rng`).
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from enum import Enum
from typing import Any, Dict, List, Optional, Tuple, Union

import numpy as np
from scipy.sparse import csr_matrix

from . import Sizeable

__all__ = [
    "CreateModelMode",
    "AntiEntropyProtocol",
    "MessageType",
    "Message",
    "Delay",
    "ConstantDelay",
    "UniformDelay",
    "LinearDelay",
    "P2PNetwork",
    "StaticP2PNetwork",
    "MixingMatrix",
    "UniformMixing",
    "MetropolisHastingsMixing",
]


class CreateModelMode(Enum):
    """How a received model is combined with the local one (gossipy/core.py:31-44)."""

    UPDATE = 1  #: train the *received* model on local data and adopt it
    MERGE_UPDATE = 2  #: merge received into local, then local training
    UPDATE_MERGE = 3  #: train both on local data, then merge
    PASS = 4  #: adopt the received model as-is


class AntiEntropyProtocol(Enum):
    """Gossip exchange protocol (gossipy/core.py:47-58)."""

    PUSH = 1
    PULL = 2
    PUSH_PULL = 3


class MessageType(Enum):
    """Wire message types (gossipy/core.py:61-75)."""

    PUSH = 1
    PULL = 2
    REPLY = 3
    PUSH_PULL = 4


class Message(Sizeable):
    """A gossip message ``(timestamp, sender, receiver, type, value)``.

    The payload is typically ``(CacheKey,)`` or ``(CacheKey, extra)``;
    ``None`` represents an ACK / model request. Size accounting recurses over
    Sizeable payload entries (gossipy/core.py:78-152).
    """

    __slots__ = ("timestamp", "sender", "receiver", "type", "value")

    def __init__(
        self,
        timestamp: int,
        sender: int,
        receiver: int,
        type: MessageType,
        value: Optional[Tuple[Any, ...]],
    ):
        self.timestamp = timestamp
        self.sender = sender
        self.receiver = receiver
        self.type = type
        self.value = value

    def get_size(self) -> int:
        """Estimated message size in atomic scalars."""
        from . import _size_of

        if self.value is None:
            return 1
        if isinstance(self.value, (tuple, list)):
            total = sum(_size_of(v, strict=True) for v in self.value if v is not None)
            return max(total, 1)
        return _size_of(self.value, strict=True)

    def __repr__(self) -> str:
        payload = "ACK" if self.value is None else str(self.value)
        return "T%d [%d -> %d] {%s}: %s" % (
            self.timestamp,
            self.sender,
            self.receiver,
            self.type.name,
            payload,
        )


class Delay(ABC):
    """Maps a message to a latency in simulation timesteps (gossipy/core.py:155-176)."""

    @abstractmethod
    def get(self, msg: Message) -> int:
        """Latency (in timesteps) for ``msg``."""
        raise NotImplementedError


class ConstantDelay(Delay):
    """Fixed latency (gossipy/core.py:179-216)."""

    def __init__(self, delay: int = 0):
        assert delay >= 0, "Delay must be non-negative!"
        self._delay = int(delay)

    def get(self, msg: Message) -> int:
        return self._delay

    def __repr__(self):
        return f"ConstantDelay({self._delay})"


class UniformDelay(Delay):
    """Latency drawn uniformly from ``[min_delay, max_delay]`` (gossipy/core.py:219-259)."""

    def __init__(self, min_delay: int, max_delay: int):
        assert 0 <= min_delay <= max_delay, (
            "The minimum delay must be non-negative and <= the maximum delay!"
        )
        self._min_delay = int(min_delay)
        self._max_delay = int(max_delay)

    def get(self, msg: Message) -> int:
        return int(np.random.randint(self._min_delay, self._max_delay + 1))

    def __repr__(self):
        return f"UniformDelay({self._min_delay}, {self._max_delay})"


class LinearDelay(Delay):
    """Bandwidth model: ``floor(timexunit * size(msg)) + overhead`` (gossipy/core.py:262-307)."""

    def __init__(self, timexunit: float, overhead: int):
        assert timexunit >= 0 and overhead >= 0
        self._timexunit = timexunit
        self._overhead = int(overhead)

    def get(self, msg: Message) -> int:
        return int(self._timexunit * msg.get_size()) + self._overhead

    def __repr__(self):
        return f"LinearDelay(time_x_unit={self._timexunit}, overhead={self._overhead})"


class P2PNetwork(ABC):
    """Network topology as an adjacency map ``node -> peer list``.

    Built from a dense adjacency matrix, a scipy CSR matrix, or fully
    connected when ``topology is None`` (gossipy/core.py:311-361).
    """

    def __init__(
        self,
        num_nodes: int,
        topology: Optional[Union[np.ndarray, csr_matrix]] = None,
    ):
        assert num_nodes > 0, "The number of nodes must be positive!"
        if topology is not None:
            assert topology.shape[0] == num_nodes, (
                "The number of nodes must match the number of rows of the topology!"
            )
        self._num_nodes = num_nodes
        self._topology: Dict[int, List[int]] = {}
        if topology is None:
            self._topology = {
                i: [j for j in range(num_nodes) if j != i] for i in range(num_nodes)
            }
        elif isinstance(topology, np.ndarray):
            for node in range(num_nodes):
                self._topology[node] = list(np.where(topology[node, :] > 0)[-1])
        else:
            for node in range(num_nodes):
                self._topology[node] = list(topology.getrow(node).nonzero()[-1])

    def size(self, node: Optional[int] = None) -> int:
        """Number of nodes, or the degree of ``node``.

        Quirk parity (gossipy/core.py:346-349): the reference tests ``if node:``
        so node 0 — and an empty peer list — fall back to global counts; the
        Metropolis–Hastings mixing weights depend on this behavior.
        """
        if node:
            peers = self._topology[node]
            return len(peers) if peers else self._num_nodes - 1
        return self._num_nodes

    @abstractmethod
    def get_peers(self, node_id: int) -> List[int]:
        """Peer list of ``node_id``."""
        raise NotImplementedError


class StaticP2PNetwork(P2PNetwork):
    """Fixed-topology network (gossipy/core.py:364-389)."""

    def get_peers(self, node_id: int) -> List[int]:
        assert 0 <= node_id < self._num_nodes
        return self._topology[node_id]

    def to_adjacency_lists(self) -> Dict[int, List[int]]:
        """Expose the adjacency map (used by the batched engine to build a
        device-resident CSR adjacency)."""
        return self._topology


class MixingMatrix(ABC):
    """Per-node mixing weights for all-to-all averaging (gossipy/core.py:392-416)."""

    def __init__(self, p2p_net: P2PNetwork) -> None:
        self.p2p_net = p2p_net

    @abstractmethod
    def get(self, node_id: int) -> np.ndarray:
        """Mixing weights for ``node_id`` (self first, then peers)."""
        raise NotImplementedError

    def __getitem__(self, node_id: int) -> np.ndarray:
        return self.get(node_id)

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}({self.p2p_net})"


class UniformMixing(MixingMatrix):
    """``1/(deg+1)`` uniform weights (gossipy/core.py:419-434)."""

    def get(self, node_id: int) -> np.ndarray:
        size = self.p2p_net.size(node_id) + 1
        return np.ones(size) / size


class MetropolisHastingsMixing(MixingMatrix):
    """Metropolis–Hastings weights ``1/(min(deg_k, deg_i)+1)`` (gossipy/core.py:437-453)."""

    def get(self, node_id: int) -> np.ndarray:
        size = self.p2p_net.size(node_id)
        peers = self.p2p_net.get_peers(node_id)
        return np.array(
            [1.0 / size]
            + [1.0 / (min(self.p2p_net.size(k), size) + 1) for k in peers]
        )
