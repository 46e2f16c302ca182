"""Simulators: the object-layer runtime (event loop, observer report, checkpoints).

Parity layer for the reference's ``gossipy/simul.py``. The classes here run
the simulation one node at a time (anywhere, CPU included) and serve as the
semantic oracle for the batched GPU engine (:mod:`gossipy_amd.engine`), which
executes the same per-timestep event schedule as node-batched HIP kernels.

Deliberate divergences from the reference (documented, all bug fixes):

* ``SimulationEventSender._receivers`` is a *per-instance* list — the
  reference uses a class attribute shared across all simulators
  (gossipy/simul.py:94), so two simulators cross-notify each other's reports.
* In ``TokenizedGossipSimulator``, the reactive burst is sent by the message
  *receiver* — the reference reuses the stale loop variable ``node`` (the
  last timed-out node, gossipy/simul.py:638-641), so reactive messages
  originate from an unrelated node; ``sender_mh`` is also unbound for
  valueless messages (gossipy/simul.py:621-633). We bind both correctly.

Replicated reference quirks (they shape the learning curves / accounting):

* a timed-out node with no peers ``break``s the whole per-timestep node loop
  (gossipy/simul.py:397-399);
* a message is counted *sent* before the drop coin-flip and *failed* again if
  dropped (gossipy/simul.py:401-407);
* replies to replies are discarded (gossipy/simul.py:423-430);
* the drop test for fresh messages is ``random() >= drop_prob`` but for
  replies ``random() > drop_prob`` (gossipy/simul.py:403,414).
"""

from __future__ import annotations

import json
from abc import ABC, abstractmethod
from collections import defaultdict
from copy import deepcopy
from typing import Callable, Dict, List, Optional, Tuple

import dill
import numpy as np

from . import CACHE, LOG, CacheKey
from .core import AntiEntropyProtocol, ConstantDelay, Delay, Message, MixingMatrix
from .data import DataDispatcher
from .flow_control import TokenAccount
from .model.handler import ModelHandler
from .node import All2AllGossipNode, GossipNode
from .utils import StringEncoder

__all__ = [
    "SimulationEventReceiver",
    "SimulationEventSender",
    "SimulationReport",
    "GossipSimulator",
    "TokenizedGossipSimulator",
    "All2AllGossipSimulator",
    "ThroughputTracer",
]


def _progress(iterable, description: str):
    """Progress-bar wrapper (rich when available, plain iterator otherwise)."""
    try:
        from rich.progress import track

        return track(iterable, description=description)
    except Exception:  # pragma: no cover
        return iterable


class SimulationEventReceiver(ABC):
    """Observer interface for simulation events (gossipy/simul.py:37-90)."""

    @abstractmethod
    def update_message(self, failed: bool, msg: Optional[Message] = None) -> None:
        """A message was sent (``failed=False``) or dropped (``failed=True``)."""
        raise NotImplementedError

    def update_evaluation(
        self, round: int, on_user: bool, evaluation: List[Dict[str, float]]
    ) -> None:
        """An evaluation sweep completed (local when ``on_user`` else global)."""

    @abstractmethod
    def update_end(self) -> None:
        """The simulation ended."""
        raise NotImplementedError

    @abstractmethod
    def update_timestep(self, t: int) -> None:
        """Timestep ``t`` completed."""
        raise NotImplementedError


class SimulationEventSender(ABC):
    """Observer subject: manages receivers and dispatches notifications
    (gossipy/simul.py:92-177; receiver list made per-instance here)."""

    def __init__(self):
        self._receivers: List[SimulationEventReceiver] = []

    def add_receiver(self, receiver: SimulationEventReceiver) -> None:
        """Attach ``receiver`` (idempotent)."""
        if receiver not in self._receivers:
            self._receivers.append(receiver)

    def remove_receiver(self, receiver: SimulationEventReceiver) -> None:
        """Detach ``receiver`` if attached."""
        try:
            self._receivers.remove(receiver)
        except ValueError:
            pass

    def notify_message(self, failed: bool, msg: Optional[Message] = None) -> None:
        """Notify all receivers of a sent/failed message."""
        for er in self._receivers:
            er.update_message(failed, msg)

    def notify_evaluation(
        self, round: int, on_user: bool, evaluation: List[Dict[str, float]]
    ) -> None:
        """Notify all receivers of an evaluation sweep."""
        for er in self._receivers:
            er.update_evaluation(round, on_user, evaluation)

    def notify_timestep(self, t: int) -> None:
        """Notify all receivers of a completed timestep."""
        for er in self._receivers:
            er.update_timestep(t)

    def notify_end(self) -> None:
        """Notify all receivers that the simulation ended."""
        for er in self._receivers:
            er.update_end()


class SimulationReport(SimulationEventReceiver):
    """Accumulates message counts/sizes and per-round mean metrics
    (gossipy/simul.py:180-270)."""

    def __init__(self):
        self.clear()

    def clear(self) -> None:
        """Reset all counters and evaluation logs."""
        self._sent_messages = 0
        self._total_size = 0
        self._failed_messages = 0
        self._global_evaluations: List[Tuple[int, Dict[str, float]]] = []
        self._local_evaluations: List[Tuple[int, Dict[str, float]]] = []

    def update_message(self, failed: bool, msg: Optional[Message] = None) -> None:
        if failed:
            self._failed_messages += 1
        else:
            assert msg is not None, "msg is not set"
            self._sent_messages += 1
            self._total_size += msg.get_size()

    def update_evaluation(
        self, round: int, on_user: bool, evaluation: List[Dict[str, float]]
    ) -> None:
        ev = self._collect_results(evaluation)
        if on_user:
            self._local_evaluations.append((round, ev))
        else:
            self._global_evaluations.append((round, ev))

    def update_end(self) -> None:
        LOG.info("# Sent messages: %d" % self._sent_messages)
        LOG.info("# Failed messages: %d" % self._failed_messages)
        LOG.info("Total size: %d" % self._total_size)

    def update_timestep(self, t: int) -> None:
        pass

    @staticmethod
    def _collect_results(results: List[Dict[str, float]]) -> Dict[str, float]:
        """Mean of each metric over the evaluated nodes."""
        if not results:
            return {}
        return {k: float(np.mean([r[k] for r in results])) for k in results[0]}

    def get_evaluation(self, local: bool = False):
        """The (round, mean-metrics) log — local or global test."""
        return self._local_evaluations if local else self._global_evaluations


class GossipSimulator(SimulationEventSender):
    """Vanilla gossip-learning simulation (gossipy/simul.py:273-503).

    A round is ``delta`` timesteps. Per timestep: timed-out nodes send to a
    random peer (subject to ``drop_prob`` and ``delay``), queued messages
    whose delay elapsed are delivered to online receivers
    (``online_prob``), replies are delivered, and at round end an evaluation
    sweep runs (all nodes, or a ``sampling_eval`` fraction).
    """

    def __init__(
        self,
        nodes: Dict[int, GossipNode],
        data_dispatcher: DataDispatcher,
        delta: int,
        protocol: AntiEntropyProtocol,
        drop_prob: float = 0.0,
        online_prob: float = 1.0,
        delay: Delay = ConstantDelay(0),
        sampling_eval: float = 0.0,
    ):
        super().__init__()
        assert 0 <= drop_prob <= 1, "drop_prob must be in the range [0,1]."
        assert 0 <= online_prob <= 1, "online_prob must be in the range [0,1]."
        assert 0 <= sampling_eval <= 1, "sampling_eval must be in the range [0,1]."
        self.data_dispatcher = data_dispatcher
        self.n_nodes = len(nodes)
        self.delta = delta
        self.protocol = protocol
        self.drop_prob = drop_prob
        self.online_prob = online_prob
        self.delay = delay
        self.sampling_eval = sampling_eval
        self.initialized = False
        self.nodes = nodes

    def init_nodes(self, seed: int = 98765) -> None:
        """Initialize every node's model (gossipy/simul.py:341-355)."""
        self.initialized = True
        for _, node in self.nodes.items():
            node.init_model()

    # -- helpers shared by the three simulators ------------------------------

    def _try_enqueue(self, msg: Message, t: int, queues, strict_drop: bool) -> None:
        """Count the send, coin-flip the drop, and enqueue at ``t + delay``.

        ``strict_drop`` selects the reference's two inconsistent drop tests:
        ``random() >= drop_prob`` for fresh messages vs ``random() > drop_prob``
        for replies (gossipy/simul.py:403,414).
        """
        self.notify_message(False, msg)
        if msg:
            r = np.random.random()
            keep = (r >= self.drop_prob) if strict_drop else (r > self.drop_prob)
            if keep:
                d = self.delay.get(msg)
                queues[t + d].append(msg)
            else:
                self.notify_message(True)

    def _evaluation_sweep(self, t: int) -> None:
        """Round-end evaluation: local test sets + global test set
        (gossipy/simul.py:432-450)."""
        if self.sampling_eval > 0:
            sample = np.random.choice(
                list(self.nodes.keys()),
                max(int(self.n_nodes * self.sampling_eval), 1),
            )
            ev = [self.nodes[i].evaluate() for i in sample if self.nodes[i].has_test()]
        else:
            sample = None
            ev = [n.evaluate() for _, n in self.nodes.items() if n.has_test()]
        if ev:
            self.notify_evaluation(t, True, ev)

        if self.data_dispatcher.has_test():
            eval_set = self.data_dispatcher.get_eval_set()
            if sample is not None:
                ev = [self.nodes[i].evaluate(eval_set) for i in sample]
            else:
                ev = [n.evaluate(eval_set) for _, n in self.nodes.items()]
            if ev:
                self.notify_evaluation(t, False, ev)

    def start(self, n_rounds: int = 100) -> None:
        """Run the simulation for ``n_rounds`` rounds (gossipy/simul.py:366-458)."""
        assert self.initialized, (
            "The simulator is not initialized. Please, call the method 'init_nodes'."
        )
        LOG.info("Simulation started.")
        node_ids = np.arange(self.n_nodes)
        msg_queues = defaultdict(list)
        rep_queues = defaultdict(list)

        try:
            for t in _progress(range(n_rounds * self.delta), "Simulating..."):
                if t % self.delta == 0:
                    np.random.shuffle(node_ids)

                for i in node_ids:
                    node = self.nodes[i]
                    if node.timed_out(t):
                        peer = node.get_peer()
                        if peer is None:
                            # reference quirk: break, not continue
                            # (gossipy/simul.py:397-399)
                            break
                        msg = node.send(t, peer, self.protocol)
                        self._try_enqueue(msg, t, msg_queues, strict_drop=True)

                is_online = np.random.random(self.n_nodes) <= self.online_prob
                for msg in msg_queues[t]:
                    if is_online[msg.receiver]:
                        reply = self.nodes[msg.receiver].receive(t, msg)
                        if reply:
                            self._try_enqueue(reply, t, rep_queues, strict_drop=False)
                    else:
                        self.notify_message(True)
                del msg_queues[t]

                for reply in rep_queues[t]:
                    if is_online[reply.receiver]:
                        self.notify_message(False, reply)
                        # replies to replies are discarded (reference parity)
                        self.nodes[reply.receiver].receive(t, reply)
                    else:
                        self.notify_message(True)
                del rep_queues[t]

                if (t + 1) % self.delta == 0:
                    self._evaluation_sweep(t)
                self.notify_timestep(t)
        except KeyboardInterrupt:
            LOG.warning("Simulation interrupted by user.")

        self.notify_end()

    def save(self, filename: str) -> None:
        """Checkpoint the simulator and the model cache as a two-slot dill
        blob ``{"simul", "cache"}`` (gossipy/simul.py:460-474)."""
        dump = {"simul": self, "cache": CACHE.get_cache()}
        with open(filename, "wb") as f:
            dill.dump(dump, f)

    @classmethod
    def load(cls, filename: str) -> "GossipSimulator":
        """Restore a simulator (and the global CACHE) from :meth:`save`'s
        format (gossipy/simul.py:476-494)."""
        with open(filename, "rb") as f:
            loaded = dill.load(f)
            CACHE.load(loaded["cache"])
            return loaded["simul"]

    def __repr__(self) -> str:
        return str(self)

    def __str__(self) -> str:
        skip = {"nodes", "model_handler_params", "gossip_node_params", "_receivers"}
        attrs = {k: v for k, v in self.__dict__.items() if k not in skip}
        return "%s %s" % (
            self.__class__.__name__,
            json.dumps(attrs, indent=4, sort_keys=True, cls=StringEncoder),
        )


class TokenizedGossipSimulator(GossipSimulator):
    """Token-account flow-controlled gossip (Danner 2018;
    gossipy/simul.py:506-689).

    Timed-out nodes send only with probability ``proactive()`` (else bank a
    token); on a delivered push without a reply, the receiver may spend
    tokens to send a burst of ``reactive(utility)`` extra messages.
    """

    def __init__(
        self,
        nodes: Dict[int, GossipNode],
        data_dispatcher: DataDispatcher,
        token_account: TokenAccount,
        utility_fun: Callable[[ModelHandler, ModelHandler, Message], int],
        delta: int,
        protocol: AntiEntropyProtocol,
        drop_prob: float = 0.0,
        online_prob: float = 1.0,
        delay: Delay = ConstantDelay(0),
        sampling_eval: float = 0.0,
    ):
        super().__init__(
            nodes,
            data_dispatcher,
            delta,
            protocol,
            drop_prob,
            online_prob,
            delay,
            sampling_eval,
        )
        self.utility_fun = utility_fun
        self.token_account_proto = token_account
        self.accounts: Dict[int, TokenAccount] = {}

    def init_nodes(self, seed: int = 98765) -> None:
        super().init_nodes(seed)
        self.accounts = {
            i: deepcopy(self.token_account_proto) for i in range(self.n_nodes)
        }

    def start(self, n_rounds: int = 100) -> None:
        assert self.initialized, (
            "The simulator is not initialized. Please, call the method 'init_nodes'."
        )
        LOG.info("Simulation started.")
        node_ids = np.arange(self.n_nodes)
        msg_queues = defaultdict(list)
        rep_queues = defaultdict(list)

        try:
            for t in _progress(range(n_rounds * self.delta), "Simulating..."):
                if t % self.delta == 0:
                    np.random.shuffle(node_ids)

                for i in node_ids:
                    node = self.nodes[i]
                    if node.timed_out(t):
                        if np.random.random() < self.accounts[i].proactive():
                            peer = node.get_peer()
                            if peer is None:
                                break
                            msg = node.send(t, peer, self.protocol)
                            self._try_enqueue(msg, t, msg_queues, strict_drop=True)
                        else:
                            self.accounts[i].add(1)

                is_online = np.random.random(self.n_nodes) <= self.online_prob
                for msg in msg_queues[t]:
                    if is_online[msg.receiver]:
                        sender_mh = None
                        if msg.value and isinstance(msg.value[0], CacheKey):
                            sender_mh = CACHE[msg.value[0]]
                        receiver = self.nodes[msg.receiver]
                        reply = receiver.receive(t, msg)
                        if reply:
                            self._try_enqueue(reply, t, rep_queues, strict_drop=False)
                        else:
                            # reactive burst, sent by the RECEIVER (bug-fixed
                            # vs gossipy/simul.py:638-641 which reuses the
                            # stale `node` loop variable)
                            utility = self.utility_fun(
                                receiver.model_handler, sender_mh, msg
                            )
                            reaction = self.accounts[msg.receiver].reactive(utility)
                            if reaction:
                                self.accounts[msg.receiver].sub(reaction)
                                for _ in range(reaction):
                                    peer = receiver.get_peer()
                                    if peer is None:
                                        break
                                    extra = receiver.send(t, peer, self.protocol)
                                    self._try_enqueue(
                                        extra, t, msg_queues, strict_drop=True
                                    )
                    else:
                        self.notify_message(True)
                del msg_queues[t]

                for reply in rep_queues[t]:
                    if is_online[reply.receiver]:
                        self.notify_message(False, reply)
                        self.nodes[reply.receiver].receive(t, reply)
                    else:
                        self.notify_message(True)
                del rep_queues[t]

                if (t + 1) % self.delta == 0:
                    self._evaluation_sweep(t)
                self.notify_timestep(t)
        except KeyboardInterrupt:
            LOG.warning("Simulation interrupted by user.")

        self.notify_end()


class All2AllGossipSimulator(GossipSimulator):
    """All-to-all averaging simulation (Koloskova 2020;
    gossipy/simul.py:720-852): every timed-out node pushes its model to
    *all* its peers; merges happen at the receiver's own next timeout with
    the mixing-matrix weights."""

    def __init__(
        self,
        nodes: Dict[int, All2AllGossipNode],
        data_dispatcher: DataDispatcher,
        delta: int,
        protocol: AntiEntropyProtocol,
        drop_prob: float = 0.0,
        online_prob: float = 1.0,
        delay: Delay = ConstantDelay(0),
        sampling_eval: float = 0.0,
    ):
        super().__init__(
            nodes,
            data_dispatcher,
            delta,
            protocol,
            drop_prob,
            online_prob,
            delay,
            sampling_eval,
        )

    def start(self, W_matrix: MixingMatrix, n_rounds: int = 100) -> None:
        """Run for ``n_rounds`` rounds with mixing weights ``W_matrix``
        (gossipy/simul.py:756-852)."""
        assert self.initialized, (
            "The simulator is not initialized. Please, call the method 'init_nodes'."
        )
        LOG.info("Simulation started.")
        node_ids = np.arange(self.n_nodes)
        msg_queues = defaultdict(list)
        rep_queues = defaultdict(list)

        try:
            for t in _progress(range(n_rounds * self.delta), "Simulating..."):
                if t % self.delta == 0:
                    np.random.shuffle(node_ids)

                for i in node_ids:
                    node = self.nodes[i]
                    if node.timed_out(t, W_matrix[i]):
                        for peer in node.get_peers():
                            msg = node.send(t, peer, self.protocol)
                            self._try_enqueue(msg, t, msg_queues, strict_drop=True)

                is_online = np.random.random(self.n_nodes) <= self.online_prob
                for msg in msg_queues[t]:
                    if is_online[msg.receiver]:
                        reply = self.nodes[msg.receiver].receive(t, msg)
                        if reply:
                            self._try_enqueue(reply, t, rep_queues, strict_drop=False)
                    else:
                        self.notify_message(True)
                del msg_queues[t]

                for reply in rep_queues[t]:
                    if is_online[reply.receiver]:
                        self.notify_message(False, reply)
                        self.nodes[reply.receiver].receive(t, reply)
                    else:
                        self.notify_message(True)
                del rep_queues[t]

                if (t + 1) % self.delta == 0:
                    self._evaluation_sweep(t)
                self.notify_timestep(t)
        except KeyboardInterrupt:
            LOG.warning("Simulation interrupted by user.")

        self.notify_end()


class ThroughputTracer(SimulationEventReceiver):
    """Tracing receiver: wall-clock per round and running rounds/sec.

    The reference has no tracing; this receiver hangs off
    the same observer interface the report uses (gossipy/simul.py:37-177)
    and works with both the object layer and the batched engine. Attach
    with ``sim.add_receiver(ThroughputTracer())``; read ``rounds_per_sec``
    / ``round_times`` afterwards, or pass ``log_every`` to emit LOG lines
    while running.
    """

    def __init__(self, delta: Optional[int] = None, log_every: int = 0):
        self._delta = delta
        self._log_every = log_every
        self._t_prev: Optional[float] = None
        self._last_t = -1
        self.round_times: List[float] = []

    def update_message(self, failed: bool, msg: Optional[Message] = None) -> None:
        pass

    def update_evaluation(self, round: int, on_user: bool, evals) -> None:
        pass

    def update_timestep(self, t: int) -> None:
        import time as _time

        if self._delta is None:
            # first two timesteps reveal the cadence (the engine notifies
            # once per round; the object layer once per tick)
            if self._last_t >= 0 and self._delta is None:
                self._delta = t - self._last_t
            self._last_t = t
        now = _time.perf_counter()
        if self._t_prev is not None:
            self.round_times.append(now - self._t_prev)
            if self._log_every and len(self.round_times) % self._log_every == 0:
                LOG.info(
                    "round %d: %.1f rounds/sec",
                    len(self.round_times),
                    self.rounds_per_sec,
                )
        self._t_prev = now

    def update_end(self) -> None:
        self._t_prev = None

    @property
    def rounds_per_sec(self) -> float:
        """Mean throughput over the recorded windows (each window is one
        notify interval: a round on the engine, a tick on the object
        layer)."""
        if not self.round_times:
            return 0.0
        return len(self.round_times) / sum(self.round_times)
