"""Host-side event scheduler for the batched engine.

The object layer discovers events one node at a time inside the timestep
loop (gossipy/simul.py:389-430). The batched engine instead *precomputes*
each round's full event schedule from the :class:`~gossipy_amd.engine.rng.
RandomTape`: which nodes fire at which tick, their peer choices, drop coin
flips, per-message delays, per-tick online masks, and the resulting delivery
lists. The schedule is a pure function of ``(seed, config, round index)`` —
every rank derives the same one, which is what makes node-residency (and GPU
count) invisible to the simulation.

Each tick becomes a :class:`TickPhase` with flat numpy arrays ready to be
shipped to the GPU as kernel arguments:

* ``snap_nodes / snap_slots`` — nodes that snapshot their model this tick
  (the batched replacement of ``ModelHandler.caching``,
  gossipy/model/handler.py:160-176 — an arena row copy instead of a
  ``copy.deepcopy``);
* delivery CSR (``recv_nodes / recv_ptr / del_slots``) — messages delivered
  this tick, grouped by receiver so one workgroup applies one receiver's
  merges+update sequentially while receivers run in parallel;
* ``reply_*`` — the PULL/PUSH_PULL reply sub-phase (same structure).

Snapshot slots are numbered sequentially per round in send order, so the
slot pool is a dense ``[n_messages, D]`` arena and cross-GPU slot exchange
is a gather/`ncclSend` per (src GPU, dst GPU) pair.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from collections import deque
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..core import AntiEntropyProtocol, ConstantDelay, Delay, LinearDelay, UniformDelay
from .rng import Purpose, RandomTape

__all__ = [
    "EngineConfig",
    "TickPhase",
    "RoundSchedule",
    "Scheduler",
    "TokenizedScheduler",
    "All2AllScheduler",
    "CacheNeighScheduler",
    "NativeTokenizedAdapter",
    "NativeSchedulerAdapter",
    "make_scheduler",
]


@dataclass
class EngineConfig:
    """Static simulation parameters shared by every rank."""

    n_nodes: int
    delta: int  #: timesteps per round
    protocol: AntiEntropyProtocol
    model_size: int  #: scalars per model (message size accounting)
    drop_prob: float = 0.0
    online_prob: float = 1.0
    delay: Delay = field(default_factory=lambda: ConstantDelay(0))
    sync: bool = True
    sampling_eval: float = 0.0
    seed: int = 98765
    #: adjacency CSR (None = fully connected); built once by the runner
    peers_indptr: Optional[np.ndarray] = None
    peers_indices: Optional[np.ndarray] = None
    #: >0 = partitioned gossip (PartitioningBasedNode, gossipy/node.py:566-659):
    #: every PUSH/REPLY carries a uniformly drawn partition id
    n_parts: int = 0
    #: sampled gossip (SamplingBasedNode, gossipy/node.py:499-562): every
    #: PUSH/REPLY carries a sample seed (rides the del_pids channel;
    #: mutually exclusive with n_parts)
    sampled: bool = False
    #: pass-through gossip (Giaretta 2019; PassThroughNode,
    #: gossipy/node.py:289-392): each delivery merges normally with
    #: probability min(1, deg_sender/deg_receiver) and otherwise ADOPTS the
    #: received model (PASS). The del_pids channel carries the resolved
    #: coin (0 = normal, 1 = pass). Mutually exclusive with
    #: n_parts/sampled.
    pass_through: bool = False


@dataclass
class TickPhase:
    """All events of one timestep, as flat arrays (see module docstring)."""

    t: int
    # sub-phase A: snapshots of firing nodes (senders)
    snap_nodes: np.ndarray  # int32 [S]
    snap_slots: np.ndarray  # int32 [S]
    # sub-phase B: model deliveries grouped by receiver (CSR)
    recv_nodes: np.ndarray  # int32 [R]   unique receivers this tick
    recv_ptr: np.ndarray  # int32 [R+1]
    del_slots: np.ndarray  # int32 [recv_ptr[-1]]  slot per delivery, in order
    #: writer of each delivered slot (slot ids are recycled, so the owner
    #: must be carried per-delivery — it drives the cross-GPU transfer plan)
    del_owners: np.ndarray  # int32 [recv_ptr[-1]]
    # PUSH_PULL / PULL: deliveries that must also emit a reply snapshot
    # (aligned with del_slots; -1 = no reply)
    reply_slots: np.ndarray  # int32 [recv_ptr[-1]]
    # PULL requests delivered this tick: the receiver snapshots (no merge)
    pull_snap_nodes: np.ndarray  # int32
    pull_snap_slots: np.ndarray  # int32
    # sub-phase C: same-tick REPLY deliveries (delay-0 PUSH_PULL/PULL replies
    # land in the reference's rep_queues[t] and are processed after the main
    # deliveries, gossipy/simul.py:423-430); their slots are written during
    # sub-phase B, so they need their own kernel launch.
    rep_recv_nodes: np.ndarray = None  # int32
    rep_recv_ptr: np.ndarray = None  # int32
    rep_del_slots: np.ndarray = None  # int32
    rep_del_owners: np.ndarray = None  # int32
    # partitioned gossip: partition id per delivery (aligned with del_slots /
    # rep_del_slots; length 0 when cfg.n_parts == 0)
    del_pids: np.ndarray = None  # int32
    rep_pids: np.ndarray = None  # int32
    # all2all weighted merge (Koloskova 2020): nodes merging their
    # accumulated neighbor models *before* this tick's snapshots
    # (gossipy/node.py:833-843). CSR over wm_nodes; wm_self_w = the node's
    # own mixing weight (weights[0]).
    wm_nodes: np.ndarray = None  # int32
    wm_ptr: np.ndarray = None  # int32
    wm_slots: np.ndarray = None  # int32
    wm_weights: np.ndarray = None  # float32 (aligned with wm_slots)
    wm_owners: np.ndarray = None  # int32 (slot writers, for the comm plan)
    wm_self_w: np.ndarray = None  # float32 [len(wm_nodes)]
    # PENS step-1 events (Onoszko 2021): a full candidate cache is scored on
    # the receiver's train data; the top-m are merged (the selection happens
    # on-device — the schedule only lists the candidates)
    pens_nodes: np.ndarray = None  # int32
    pens_ptr: np.ndarray = None  # int32
    pens_slots: np.ndarray = None  # int32 (candidate slots, arrival order)
    pens_owners: np.ndarray = None  # int32 (candidate senders)

    @property
    def n_events(self) -> int:
        n_rep = 0 if self.rep_del_slots is None else len(self.rep_del_slots)
        n_wm = 0 if self.wm_nodes is None else len(self.wm_nodes)
        n_pens = 0 if self.pens_nodes is None else len(self.pens_nodes)
        return (
            len(self.snap_nodes)
            + len(self.del_slots)
            + len(self.pull_snap_nodes)
            + n_rep
            + n_wm
            + n_pens
        )


@dataclass
class RoundSchedule:
    """One round's ticks plus bookkeeping for the report and the comm plan."""

    round_idx: int
    ticks: List[TickPhase]
    n_slots: int  #: size of the snapshot slot pool this round
    slot_owner: np.ndarray  #: int32 [n_slots] — node that writes each slot
    sent_messages: int
    failed_messages: int
    total_size: int
    eval_nodes: Optional[np.ndarray]  #: node sample for the round-end sweep


class Scheduler:
    """Derives :class:`RoundSchedule` objects from the random tape.

    Carry-over state between rounds (in-flight messages whose delay crosses
    the round boundary) is kept internally, so rounds must be generated in
    order — which every rank does identically.
    """

    def __init__(self, cfg: EngineConfig):
        self.cfg = cfg
        self.tape = RandomTape(cfg.seed)
        n = cfg.n_nodes
        # per-node timeout offsets (gossipy/node.py:79): sync U(0, delta),
        # async N(delta, delta/10) clipped to >=1 (the reference can draw <=0
        # and crash on t % 0 — SURVEY.md §2.3 quirk 16)
        g = self.tape.stream(Purpose.TIMEOUT)
        if cfg.sync:
            self.deltas = np.atleast_1d(g.integers(0, cfg.delta, size=n)).astype(
                np.int64
            )
        else:
            self.deltas = np.maximum(
                1, g.normal(cfg.delta, cfg.delta / 10, size=n).astype(np.int64)
            )
        # in-flight messages carried across round boundaries:
        # tick -> list of (receiver, slot, reply_flag, is_pull_request, sender, pid)
        self._pending: Dict[int, List[Tuple[int, int, int, bool, int, int]]] = {}
        # tick -> (sent, failed, size) accounting of replies enqueued there
        self._reply_accounting: Dict[int, Tuple[int, int, int]] = {}
        # deterministic slot allocator: ids persist across rounds (a delayed
        # message keeps its slot until delivered), recycled through a free
        # list once consumed/dropped. This is the arena equivalent of the
        # reference cache's refcounting (gossipy/__init__.py:283-387) —
        # except slots of messages lost to drops/offline receivers are
        # reclaimed instead of leaked (the reference leaks them,
        # SURVEY.md §5 'failure detection').
        self._reuse_q: deque = deque()  # (free_tick, slot), tick-ordered
        self._next_slot = 0
        self.slot_owner = np.zeros(64, dtype=np.int32)
        # effective reuse lag >= one full round: packed launch groups never
        # span a round boundary, so no slot id can recycle into a
        # still-open group (see SLOT_REUSE_LAG doc below). Must match the
        # C++ scheduler (csrc/scheduler.cpp pend_init).
        self.SLOT_REUSE_LAG = max(type(self).SLOT_REUSE_LAG, cfg.delta)

    #: a consumed slot id is only reused SLOT_REUSE_LAG ticks after the tick
    #: that freed it. Adjacent ticks therefore never alias slot ids, which is
    #: what lets the runner fuse conflict-free ticks into single launch
    #: groups (the old LIFO free list re-issued a slot on the very next
    #: tick, making every tick pair conflict). The effective lag is
    #: ``max(32, delta)`` (set per instance in ``__init__``): launch groups
    #: never span a round, so a lag of at least one full round makes
    #: every group structurally alias-free — the packers' touched-set
    #: checks become defense-in-depth instead of the only guarantee.
    #: Costs ~delay-window x firing-rate extra pool rows — negligible
    #: against 288 GB of HBM3E.
    SLOT_REUSE_LAG = 32

    def _alloc_slot(self, owner: int, t: int) -> int:
        q = self._reuse_q
        if q and q[0][0] + self.SLOT_REUSE_LAG <= t:
            s = q.popleft()[1]
        else:
            s = self._next_slot
            self._next_slot += 1
            if s >= len(self.slot_owner):
                self.slot_owner = np.resize(self.slot_owner, 2 * len(self.slot_owner))
        self.slot_owner[s] = owner
        return s

    def _recycle(self, t: int, freed) -> None:
        self._reuse_q.extend((t, s) for s in freed)

    # -- internals -----------------------------------------------------------

    def _firing(self, t: int) -> np.ndarray:
        if self.cfg.sync:
            # static timeout offsets: bucket once, O(1) per tick
            buckets = getattr(self, "_fire_buckets", None)
            if buckets is None:
                buckets = [
                    np.where(self.deltas == ph)[0].astype(np.int64)
                    for ph in range(self.cfg.delta)
                ]
                self._fire_buckets = buckets
            return buckets[t % self.cfg.delta]
        return np.where((t % self.deltas) == 0)[0]

    def _peers_of(self, nodes: np.ndarray, t: int) -> np.ndarray:
        """Peer choice per firing node (uniform over its adjacency row;
        fully-connected fast path excludes self)."""
        cfg = self.cfg
        n = len(nodes)
        if n == 0:
            return np.empty(0, dtype=np.int64)
        g = self.tape.stream(Purpose.PEER, t)
        if cfg.peers_indptr is None:
            draw = np.atleast_1d(g.integers(0, cfg.n_nodes - 1, size=n))
            return draw + (draw >= nodes)  # skip self
        starts = cfg.peers_indptr[nodes]
        degs = cfg.peers_indptr[nodes + 1] - starts
        # nodes with no peers never appear here (runner validates topology)
        offs = np.floor(np.atleast_1d(g.random(n)) * degs).astype(np.int64)
        return cfg.peers_indices[starts + offs]

    def _delays(self, t: int, n: int, sizes: np.ndarray) -> np.ndarray:
        d = self.cfg.delay
        if isinstance(d, ConstantDelay):
            return np.full(n, d._delay, dtype=np.int64)
        if isinstance(d, UniformDelay):
            g = self.tape.stream(Purpose.DELAY, t)
            return np.atleast_1d(g.integers(d._min_delay, d._max_delay + 1, size=n))
        if isinstance(d, LinearDelay):
            return (d._timexunit * sizes).astype(np.int64) + d._overhead
        # custom Delay subclass: fall back to per-message scalar calls
        return np.array([d.get(None) for _ in range(n)], dtype=np.int64)

    # -- public --------------------------------------------------------------

    def next_round(self, r: int) -> RoundSchedule:
        """Build the schedule of round ``r`` (rounds must be consumed in
        order; the in-flight message queue carries over)."""
        cfg = self.cfg
        proto = cfg.protocol
        t0, t1 = r * cfg.delta, (r + 1) * cfg.delta
        sent = failed = total_size = 0
        ticks: List[TickPhase] = []

        for t in range(t0, t1):
            firing = self._firing(t)
            snap_nodes: List[int] = []
            snap_slots: List[int] = []
            freed: List[int] = []  # slots recycled at end of this tick

            # --- sends (sub-phase A of the reference loop,
            #     gossipy/simul.py:393-407)
            n_f = len(firing)
            if n_f:
                peers = self._peers_of(firing, t)
                if cfg.drop_prob <= 0.0:
                    drop_u = np.ones(n_f)
                else:
                    drop_u = self.tape.uniform(Purpose.DROP, t, n_f)
                sizes = np.full(n_f, cfg.model_size if proto != AntiEntropyProtocol.PULL else 1)
                delays = self._delays(t, n_f, sizes)
                # partition ids / sample seeds drawn per send
                # (gossipy/node.py:617,631 / :520-531)
                pids = self._send_extras(t, n_f)
                for j in range(n_f):
                    sender, receiver = int(firing[j]), int(peers[j])
                    is_pull = proto == AntiEntropyProtocol.PULL
                    slot = -1
                    if not is_pull:
                        slot = self._alloc_slot(sender, t)
                        snap_nodes.append(sender)
                        snap_slots.append(slot)
                    sent += 1
                    total_size += int(sizes[j])
                    if drop_u[j] >= cfg.drop_prob:
                        due = t + int(delays[j])
                        wants_reply = proto == AntiEntropyProtocol.PUSH_PULL
                        self._pending.setdefault(due, []).append(
                            (receiver, slot, -2 if wants_reply else -1, is_pull,
                             sender, int(pids[j]))
                        )
                    else:
                        failed += 1
                        if slot >= 0:
                            freed.append(slot)

            # --- deliveries due this tick (sub-phase B,
            #     gossipy/simul.py:409-430). Reply messages generated here are
            #     enqueued for their own delivery tick; replies to replies are
            #     discarded (reference parity).
            if cfg.online_prob >= 1.0:
                online = np.ones(cfg.n_nodes, dtype=bool)
            else:
                online_u = self.tape.uniform(Purpose.ONLINE, t, cfg.n_nodes)
                online = online_u <= cfg.online_prob
            due = self._pending.pop(t, [])
            recv_map: Dict[int, List[Tuple[int, int, int, int]]] = {}
            pull_nodes: List[int] = []
            pull_slots: List[int] = []
            pt_stream = (
                self.tape.stream(Purpose.MISC, t) if cfg.pass_through else None
            )
            for receiver, slot, reply_flag, is_pull, sender, pid in due:
                if not online[receiver]:
                    failed += 1
                    if slot >= 0:
                        freed.append(slot)
                    continue
                if pt_stream is not None and not is_pull:
                    # degree-aware accept coin (gossipy/node.py:380-386)
                    p = min(1.0, self._deg(sender) / max(1, self._deg(receiver)))
                    pid = 0 if float(pt_stream.random()) < p else 1
                if is_pull:
                    # PULL request: receiver snapshots and replies (with a
                    # fresh partition id, gossipy/node.py:651-653)
                    rslot = self._alloc_slot(receiver, t)
                    pull_nodes.append(receiver)
                    pull_slots.append(rslot)
                    rpid = self._reply_pid(t, receiver)
                    if not self._enqueue_reply(t, receiver, sender, rslot, rpid):
                        freed.append(rslot)
                    continue
                rslot = -1
                if reply_flag == -2:  # PUSH_PULL: reply with post-merge model
                    rslot = self._alloc_slot(receiver, t)
                    rpid = self._reply_pid(t, receiver)
                    if not self._enqueue_reply(t, receiver, sender, rslot, rpid):
                        freed.append(rslot)
                recv_map.setdefault(receiver, []).append((slot, rslot, sender, pid))
                freed.append(slot)  # consumed by this delivery

            recv_nodes = np.fromiter(recv_map.keys(), dtype=np.int32, count=len(recv_map))
            recv_ptr = np.zeros(len(recv_map) + 1, dtype=np.int32)
            del_slots: List[int] = []
            del_owners: List[int] = []
            reply_slots: List[int] = []
            del_pids: List[int] = []
            for i, rn in enumerate(recv_nodes):
                pairs = recv_map[int(rn)]
                del_slots.extend(p[0] for p in pairs)
                reply_slots.extend(p[1] for p in pairs)
                del_owners.extend(p[2] for p in pairs)
                del_pids.extend(p[3] for p in pairs)
                recv_ptr[i + 1] = recv_ptr[i] + len(pairs)

            # --- sub-phase C: replies that came due in THIS tick (delay 0).
            # They were enqueued by the loop above into _pending[t]; pop
            # again. Replies to replies do not exist (parity).
            rep_due = self._pending.pop(t, [])
            rep_map: Dict[int, List[Tuple[int, int, int]]] = {}
            for receiver, slot, _rf, _ip, sender, pid in rep_due:
                if not online[receiver]:
                    failed += 1
                    freed.append(slot)
                    continue
                if pt_stream is not None:
                    p = min(1.0, self._deg(sender) / max(1, self._deg(receiver)))
                    pid = 0 if float(pt_stream.random()) < p else 1
                rep_map.setdefault(receiver, []).append((slot, sender, pid))
                freed.append(slot)  # consumed by this reply delivery
            rep_recv = np.fromiter(rep_map.keys(), dtype=np.int32, count=len(rep_map))
            rep_ptr = np.zeros(len(rep_map) + 1, dtype=np.int32)
            rep_slots: List[int] = []
            rep_owners: List[int] = []
            rep_pids: List[int] = []
            for i, rn in enumerate(rep_recv):
                rep_slots.extend(p[0] for p in rep_map[int(rn)])
                rep_owners.extend(p[1] for p in rep_map[int(rn)])
                rep_pids.extend(p[2] for p in rep_map[int(rn)])
                rep_ptr[i + 1] = len(rep_slots)

            phase = TickPhase(
                t=t,
                snap_nodes=np.asarray(snap_nodes, dtype=np.int32),
                snap_slots=np.asarray(snap_slots, dtype=np.int32),
                recv_nodes=recv_nodes,
                recv_ptr=recv_ptr,
                del_slots=np.asarray(del_slots, dtype=np.int32),
                del_owners=np.asarray(del_owners, dtype=np.int32),
                reply_slots=np.asarray(reply_slots, dtype=np.int32),
                pull_snap_nodes=np.asarray(pull_nodes, dtype=np.int32),
                pull_snap_slots=np.asarray(pull_slots, dtype=np.int32),
                rep_recv_nodes=rep_recv,
                rep_recv_ptr=rep_ptr,
                rep_del_slots=np.asarray(rep_slots, dtype=np.int32),
                rep_del_owners=np.asarray(rep_owners, dtype=np.int32),
                del_pids=np.asarray(del_pids, dtype=np.int32),
                rep_pids=np.asarray(rep_pids, dtype=np.int32),
            )
            if phase.n_events:
                ticks.append(phase)

            # count messages the reply bookkeeping generated
            sent_r, failed_r, size_r = self._reply_accounting.pop(t, (0, 0, 0))
            sent += sent_r
            failed += failed_r
            total_size += size_r

            # recycle this tick's consumed/dropped slots (safe: any reuse
            # happens in a later tick's launch, stream-ordered after the
            # consuming kernel)
            self._recycle(t, freed)

        eval_nodes = None
        if cfg.sampling_eval > 0:
            g = self.tape.stream(Purpose.EVAL, t1 - 1)
            k = max(int(cfg.n_nodes * cfg.sampling_eval), 1)
            eval_nodes = np.atleast_1d(g.integers(0, cfg.n_nodes, size=k))

        return RoundSchedule(
            round_idx=r,
            ticks=ticks,
            n_slots=self._next_slot,
            slot_owner=self.slot_owner[: self._next_slot].copy(),
            sent_messages=sent,
            failed_messages=failed,
            total_size=total_size,
            eval_nodes=eval_nodes,
        )

    def _deg(self, node: int) -> int:
        if self.cfg.peers_indptr is None:
            return self.cfg.n_nodes - 1
        return int(
            self.cfg.peers_indptr[node + 1] - self.cfg.peers_indptr[node]
        )

    def _send_extras(self, t: int, n: int) -> np.ndarray:
        """Per-send extra int riding with each message: partition id
        (partitioned gossip), sample seed (sampled gossip), or -1."""
        cfg = self.cfg
        if cfg.n_parts > 0:
            return np.atleast_1d(
                self.tape.stream(Purpose.PART, t).integers(0, cfg.n_parts, size=n)
            )
        if cfg.sampled:
            return np.atleast_1d(
                self.tape.stream(Purpose.SAMPLE, t).integers(0, 2**31, size=n)
            )
        return np.full(n, -1, dtype=np.int64)

    def _reply_pid(self, t: int, replier: int) -> int:
        """Fresh partition id / sample seed for a reply
        (gossipy/node.py:651 / :551-557). Keyed on (t, replier): two
        replies by one node in one tick share the draw."""
        if self.cfg.n_parts > 0:
            return int(
                self.tape.stream(Purpose.PART, t, extra=1 + replier).integers(
                    0, self.cfg.n_parts
                )
            )
        if self.cfg.sampled:
            return int(
                self.tape.stream(Purpose.SAMPLE, t, extra=1 + replier).integers(
                    0, 2**31
                )
            )
        return -1

    def _enqueue_reply(
        self, t: int, replier: int, requester: int, slot: int, pid: int = -1
    ) -> bool:
        """Queue a REPLY message (drop-tested with the reference's
        ``random() > drop_prob`` variant, gossipy/simul.py:414). Returns
        whether the reply was actually enqueued (False = dropped)."""
        g = self.tape.stream(Purpose.DROP, t, extra=1 + replier)
        u = float(g.random())
        sent, failed, size = self._reply_accounting.get(t, (0, 0, 0))
        sent += 1
        size += self.cfg.model_size
        if u > self.cfg.drop_prob:
            gd = self.tape.stream(Purpose.DELAY, t, extra=1 + replier)
            d = self.cfg.delay
            if isinstance(d, ConstantDelay):
                dly = d._delay
            elif isinstance(d, UniformDelay):
                dly = int(gd.integers(d._min_delay, d._max_delay + 1))
            elif isinstance(d, LinearDelay):
                dly = int(d._timexunit * self.cfg.model_size) + d._overhead
            else:
                dly = int(d.get(None))
            # -3 marks a REPLY payload: processed like a plain delivery, but
            # it never triggers a reactive burst (the reference delivers
            # replies outside the reacting loop, gossipy/simul.py:623-648)
            self._pending.setdefault(t + dly, []).append(
                (requester, slot, -3, False, replier, pid)
            )
            self._reply_accounting[t] = (sent, failed, size)
            return True
        failed += 1
        self._reply_accounting[t] = (sent, failed, size)
        return False


class TokenizedScheduler(Scheduler):
    """Token-account flow-controlled schedule (TokenizedGossipSimulator,
    gossipy/simul.py:506-689 — with the receiver, not a stale loop variable,
    sending the reactive burst; see gossipy_amd/simul.py for the same fix).

    Differences from the base schedule:

    * a firing node sends only with probability ``proactive()`` (one tape
      draw per firing node, ascending node order); otherwise banks a token;
    * a delivered *push* (a message that generates no reply) triggers a
      reactive burst from its receiver: ``reactive(utility)`` extra sends,
      with tokens spent, snapshot taken *after* the receiving merge/update
      (the burst-sender snapshot lands in the next same-tick wave's phase A);
    * zero-delay burst messages are delivered in the same tick as additional
      waves (each wave is one more batched deliver launch), matching the
      reference's grow-while-iterating message loop (gossipy/simul.py:
      634-648).

    The utility function is host-side: ``utility_fun(receiver, sender, t)``
    over node ids (the reference's experiments use a constant — e.g.
    main_hegedus_2021.py:57; model-state-dependent utilities stay on the
    object layer).
    """

    def __init__(self, cfg: EngineConfig, token_account, utility_fun=None):
        super().__init__(cfg)
        import copy as _copy
        import inspect as _inspect

        self.accounts = [
            _copy.deepcopy(token_account) for _ in range(cfg.n_nodes)
        ]
        self.utility_fun = utility_fun or (lambda recv, sender, t: 1)
        self._react_takes_u = (
            "u" in _inspect.signature(token_account.reactive).parameters
        )

    def _react(self, node: int, utility: int, u: float) -> int:
        acct = self.accounts[node]
        if self._react_takes_u:
            return acct.reactive(utility, u=u)
        return acct.reactive(utility)

    def next_round(self, r: int) -> RoundSchedule:
        cfg = self.cfg
        proto = cfg.protocol
        t0, t1 = r * cfg.delta, (r + 1) * cfg.delta
        sent = failed = total_size = 0
        ticks: List[TickPhase] = []

        for t in range(t0, t1):
            firing = self._firing(t)
            freed: List[int] = []

            # --- proactive-gated sends (gossipy/simul.py:602-615)
            snap_nodes: List[int] = []
            snap_slots: List[int] = []
            n_f = len(firing)
            if n_f:
                pro_u = self.tape.uniform(Purpose.TOKEN, t, n_f)
                go = [
                    pro_u[j] < self.accounts[int(firing[j])].proactive()
                    for j in range(n_f)
                ]
                senders = firing[np.asarray(go, dtype=bool)]
                for j in range(n_f):
                    if not go[j]:
                        self.accounts[int(firing[j])].add(1)
                n_s = len(senders)
                if n_s:
                    peers = self._peers_of(senders, t)
                    drop_u = self.tape.uniform(Purpose.DROP, t, n_s)
                    sizes = np.full(
                        n_s,
                        cfg.model_size if proto != AntiEntropyProtocol.PULL else 1,
                    )
                    delays = self._delays(t, n_s, sizes)
                    pids = self._send_extras(t, n_s)
                    for j in range(n_s):
                        sender, receiver = int(senders[j]), int(peers[j])
                        is_pull = proto == AntiEntropyProtocol.PULL
                        slot = -1
                        if not is_pull:
                            slot = self._alloc_slot(sender, t)
                            snap_nodes.append(sender)
                            snap_slots.append(slot)
                        sent += 1
                        total_size += int(sizes[j])
                        if drop_u[j] >= cfg.drop_prob:
                            due = t + int(delays[j])
                            wants = proto == AntiEntropyProtocol.PUSH_PULL
                            self._pending.setdefault(due, []).append(
                                (receiver, slot, -2 if wants else -1, is_pull,
                                 sender, int(pids[j]))
                            )
                        else:
                            failed += 1
                            if slot >= 0:
                                freed.append(slot)

            if cfg.online_prob >= 1.0:
                online = np.ones(cfg.n_nodes, dtype=bool)
            else:
                online_u = self.tape.uniform(Purpose.ONLINE, t, cfg.n_nodes)
                online = online_u <= cfg.online_prob

            # --- delivery waves: wave 0 = scheduled messages due at t;
            # each wave's reactive bursts with zero delay feed the next wave
            wave_due = self._pending.pop(t, [])
            #: per-(receiver) sequential tape streams for this tick's bursts
            burst_streams: Dict[Tuple[int, int], object] = {}

            def tick_stream(purpose, node):
                key = (int(purpose), node)
                s = burst_streams.get(key)
                if s is None:
                    s = self.tape.stream(purpose, t, extra=1 + node)
                    burst_streams[key] = s
                return s

            while wave_due or snap_nodes:
                recv_map: Dict[int, List[Tuple[int, int, int, int]]] = {}
                pull_nodes: List[int] = []
                pull_slots: List[int] = []
                next_due: List[Tuple[int, int, int, bool, int, int]] = []
                burst_snap_nodes: List[int] = []
                burst_snap_slots: List[int] = []

                for receiver, slot, reply_flag, is_pull, sender, pid in wave_due:
                    if not online[receiver]:
                        failed += 1
                        if slot >= 0:
                            freed.append(slot)
                        continue
                    if is_pull:
                        rslot = self._alloc_slot(receiver, t)
                        pull_nodes.append(receiver)
                        pull_slots.append(rslot)
                        rpid = self._reply_pid(t, receiver)
                        if not self._enqueue_reply(t, receiver, sender, rslot, rpid):
                            freed.append(rslot)
                        continue
                    rslot = -1
                    if reply_flag == -2:
                        rslot = self._alloc_slot(receiver, t)
                        rpid = self._reply_pid(t, receiver)
                        if not self._enqueue_reply(t, receiver, sender, rslot, rpid):
                            freed.append(rslot)
                    recv_map.setdefault(receiver, []).append((slot, rslot, pid, sender))
                    freed.append(slot)
                    # reactive burst: only deliveries that generate no reply
                    # (gossipy/simul.py:631-648); REPLY deliveries are marked
                    # -3 by _enqueue_reply and excluded
                    if reply_flag == -1:
                        utility = int(self.utility_fun(receiver, sender, t))
                        ru = float(tick_stream(Purpose.TOKEN, receiver).random())
                        reaction = self._react(receiver, utility, ru)
                        if reaction <= 0:
                            continue
                        self.accounts[receiver].sub(reaction)
                        gp = tick_stream(Purpose.PEER, receiver)
                        gd = tick_stream(Purpose.DROP, receiver)
                        gdl = tick_stream(Purpose.DELAY, receiver)
                        for _ in range(reaction):
                            peer = self._burst_peer(receiver, gp)
                            bslot = self._alloc_slot(receiver, t)
                            burst_snap_nodes.append(receiver)
                            burst_snap_slots.append(bslot)
                            if cfg.n_parts > 0:
                                bpid = int(
                                    tick_stream(Purpose.PART, receiver).integers(
                                        0, cfg.n_parts
                                    )
                                )
                            elif cfg.sampled:
                                bpid = int(
                                    tick_stream(Purpose.SAMPLE, receiver).integers(
                                        0, 2**31
                                    )
                                )
                            else:
                                bpid = -1
                            sent += 1
                            total_size += cfg.model_size
                            if float(gd.random()) >= cfg.drop_prob:
                                dly = self._burst_delay(gdl)
                                wants = proto == AntiEntropyProtocol.PUSH_PULL
                                m = (peer, bslot, -2 if wants else -1, False,
                                     receiver, bpid)
                                if dly == 0:
                                    next_due.append(m)
                                else:
                                    self._pending.setdefault(t + dly, []).append(m)
                            else:
                                failed += 1
                                freed.append(bslot)

                # emit this wave's phase
                recv_nodes = np.fromiter(
                    recv_map.keys(), dtype=np.int32, count=len(recv_map)
                )
                recv_ptr = np.zeros(len(recv_map) + 1, dtype=np.int32)
                del_slots: List[int] = []
                del_owners: List[int] = []
                reply_slots: List[int] = []
                del_pids: List[int] = []
                for i, rn in enumerate(recv_nodes):
                    quads = recv_map[int(rn)]
                    del_slots.extend(q[0] for q in quads)
                    reply_slots.extend(q[1] for q in quads)
                    del_pids.extend(q[2] for q in quads)
                    del_owners.extend(q[3] for q in quads)
                    recv_ptr[i + 1] = recv_ptr[i] + len(quads)
                phase = TickPhase(
                    t=t,
                    snap_nodes=np.asarray(snap_nodes, dtype=np.int32),
                    snap_slots=np.asarray(snap_slots, dtype=np.int32),
                    recv_nodes=recv_nodes,
                    recv_ptr=recv_ptr,
                    del_slots=np.asarray(del_slots, dtype=np.int32),
                    del_owners=np.asarray(del_owners, dtype=np.int32),
                    reply_slots=np.asarray(reply_slots, dtype=np.int32),
                    pull_snap_nodes=np.asarray(pull_nodes, dtype=np.int32),
                    pull_snap_slots=np.asarray(pull_slots, dtype=np.int32),
                    del_pids=np.asarray(del_pids, dtype=np.int32),
                    rep_pids=np.zeros(0, dtype=np.int32),
                )
                if phase.n_events:
                    ticks.append(phase)
                # next wave: burst snapshots become phase-A of the next wave
                snap_nodes, snap_slots = burst_snap_nodes, burst_snap_slots
                wave_due = next_due

            # --- same-tick replies (sub-phase C)
            rep_due = self._pending.pop(t, [])
            if rep_due:
                rep_map: Dict[int, List[Tuple[int, int, int]]] = {}
                for receiver, slot, _rf, _ip, sender, pid in rep_due:
                    if not online[receiver]:
                        failed += 1
                        freed.append(slot)
                        continue
                    rep_map.setdefault(receiver, []).append((slot, sender, pid))
                    freed.append(slot)
                rep_recv = np.fromiter(
                    rep_map.keys(), dtype=np.int32, count=len(rep_map)
                )
                rep_ptr = np.zeros(len(rep_map) + 1, dtype=np.int32)
                rep_slots: List[int] = []
                rep_owners: List[int] = []
                rep_pids: List[int] = []
                for i, rn in enumerate(rep_recv):
                    rep_slots.extend(p[0] for p in rep_map[int(rn)])
                    rep_owners.extend(p[1] for p in rep_map[int(rn)])
                    rep_pids.extend(p[2] for p in rep_map[int(rn)])
                    rep_ptr[i + 1] = len(rep_slots)
                ticks.append(
                    TickPhase(
                        t=t,
                        snap_nodes=np.zeros(0, dtype=np.int32),
                        snap_slots=np.zeros(0, dtype=np.int32),
                        recv_nodes=np.zeros(0, dtype=np.int32),
                        recv_ptr=np.zeros(1, dtype=np.int32),
                        del_slots=np.zeros(0, dtype=np.int32),
                        del_owners=np.zeros(0, dtype=np.int32),
                        reply_slots=np.zeros(0, dtype=np.int32),
                        pull_snap_nodes=np.zeros(0, dtype=np.int32),
                        pull_snap_slots=np.zeros(0, dtype=np.int32),
                        rep_recv_nodes=rep_recv,
                        rep_recv_ptr=rep_ptr,
                        rep_del_slots=np.asarray(rep_slots, dtype=np.int32),
                        rep_del_owners=np.asarray(rep_owners, dtype=np.int32),
                        rep_pids=np.asarray(rep_pids, dtype=np.int32),
                        del_pids=np.zeros(0, dtype=np.int32),
                    )
                )

            sent_r, failed_r, size_r = self._reply_accounting.pop(t, (0, 0, 0))
            sent += sent_r
            failed += failed_r
            total_size += size_r
            self._recycle(t, freed)

        eval_nodes = None
        if cfg.sampling_eval > 0:
            g = self.tape.stream(Purpose.EVAL, t1 - 1)
            k = max(int(cfg.n_nodes * cfg.sampling_eval), 1)
            eval_nodes = np.atleast_1d(g.integers(0, cfg.n_nodes, size=k))

        return RoundSchedule(
            round_idx=r,
            ticks=ticks,
            n_slots=self._next_slot,
            slot_owner=self.slot_owner[: self._next_slot].copy(),
            sent_messages=sent,
            failed_messages=failed,
            total_size=total_size,
            eval_nodes=eval_nodes,
        )

    def _burst_peer(self, node: int, gp) -> int:
        cfg = self.cfg
        if cfg.peers_indptr is None:
            draw = int(gp.integers(0, cfg.n_nodes - 1))
            return draw + (1 if draw >= node else 0)
        s = int(cfg.peers_indptr[node])
        deg = int(cfg.peers_indptr[node + 1]) - s
        return int(cfg.peers_indices[s + int(np.floor(float(gp.random()) * deg))])

    def _burst_delay(self, gdl) -> int:
        d = self.cfg.delay
        if isinstance(d, ConstantDelay):
            return d._delay
        if isinstance(d, UniformDelay):
            return int(gdl.integers(d._min_delay, d._max_delay + 1))
        if isinstance(d, LinearDelay):
            return int(d._timexunit * self.cfg.model_size) + d._overhead
        return int(d.get(None))


class All2AllScheduler(Scheduler):
    """Schedule for decentralized weighted averaging (Koloskova 2020;
    All2AllGossipSimulator, gossipy/simul.py:720-852).

    Per tick: every timed-out node first merges the neighbor models it has
    accumulated since its last timeout — a *weighted* k-way merge with
    mixing weights, followed by a local update — and then pushes its
    post-merge model to **all** its peers (gossipy/node.py:833-846). A
    receiver only stores the incoming model (replacing any previous one
    from the same sender, gossipy/node.py:860-869); no learning happens at
    delivery time.

    The mixing weights replicate the reference contract: ``weights[0]``
    scales the node's own model and ``weights[1+j]`` the j-th accumulated
    model *in arrival order* (gossipy/model/handler.py:666-688 applies the
    vector positionally). With fewer arrivals than peers the unused tail is
    dropped — so partial participation shrinks the average exactly as the
    reference does.

    One snapshot slot serves all of a node's peers (the reference caches
    once under one key and pushes the same CacheKey to every peer,
    gossipy/node.py:845-846 + handler.py:160-176), so slots are
    ref-counted: freed when the last referencing delivery is consumed,
    dropped, or replaced.
    """

    def __init__(self, cfg: EngineConfig, mixing=None):
        super().__init__(cfg)
        n = cfg.n_nodes
        if cfg.peers_indptr is None:
            self._degs = np.full(n, n - 1, dtype=np.int64)
        else:
            self._degs = (cfg.peers_indptr[1:] - cfg.peers_indptr[:-1]).astype(
                np.int64
            )
        if mixing is None:
            # UniformMixing (gossipy/core.py:419-434)
            self._W = [
                np.ones(self._degs[i] + 1) / (self._degs[i] + 1) for i in range(n)
            ]
        elif callable(getattr(mixing, "get", None)):
            self._W = [np.asarray(mixing.get(i), dtype=np.float64) for i in range(n)]
        else:
            self._W = [np.asarray(mixing(i), dtype=np.float64) for i in range(n)]
        #: per-node accumulated (sender, slot) in arrival order
        self._acc: List[List[Tuple[int, int]]] = [[] for _ in range(n)]
        #: outstanding references per slot
        self._refs: Dict[int, int] = {}

    def _deref(self, slot: int, freed: List[int]) -> None:
        r = self._refs.get(slot, 0) - 1
        if r <= 0:
            self._refs.pop(slot, None)
            freed.append(slot)
        else:
            self._refs[slot] = r

    def _all_peers(self, node: int) -> np.ndarray:
        cfg = self.cfg
        if cfg.peers_indptr is None:
            return np.concatenate(
                [np.arange(node), np.arange(node + 1, cfg.n_nodes)]
            )
        s = cfg.peers_indptr[node]
        return cfg.peers_indices[s : s + self._degs[node]]

    def next_round(self, r: int) -> RoundSchedule:
        cfg = self.cfg
        t0, t1 = r * cfg.delta, (r + 1) * cfg.delta
        sent = failed = total_size = 0
        ticks: List[TickPhase] = []

        for t in range(t0, t1):
            freed: List[int] = []
            firing = self._firing(t)

            # --- weighted merges of accumulated models (before sends)
            wm_nodes: List[int] = []
            wm_ptr = [0]
            wm_slots: List[int] = []
            wm_weights: List[float] = []
            wm_owners: List[int] = []
            wm_self_w: List[float] = []
            for i in firing:
                acc = self._acc[int(i)]
                if not acc:
                    continue
                W = self._W[int(i)]
                wm_nodes.append(int(i))
                wm_self_w.append(float(W[0]))
                for j, (sender, slot) in enumerate(acc):
                    wm_slots.append(slot)
                    # positional weight; tail beyond the vector repeats the
                    # last entry defensively (cannot happen on static nets)
                    wm_weights.append(float(W[min(1 + j, len(W) - 1)]))
                    wm_owners.append(sender)
                    self._deref(slot, freed)
                wm_ptr.append(len(wm_slots))
                self._acc[int(i)] = []

            # --- broadcast sends: one snapshot slot, one message per peer
            snap_nodes: List[int] = []
            snap_slots: List[int] = []
            n_f = len(firing)
            if n_f:
                drop_stream = self.tape.stream(Purpose.DROP, t)
                delay_stream = self.tape.stream(Purpose.DELAY, t)
                for i in firing:
                    node = int(i)
                    peers = self._all_peers(node)
                    slot = self._alloc_slot(node, t)
                    snap_nodes.append(node)
                    snap_slots.append(slot)
                    refs = 0
                    for peer in peers:
                        sent += 1
                        total_size += cfg.model_size
                        if float(drop_stream.random()) >= cfg.drop_prob:
                            dly = self._wm_delay(delay_stream)
                            self._pending.setdefault(t + dly, []).append(
                                (int(peer), slot, -1, False, node, -1)
                            )
                            refs += 1
                        else:
                            failed += 1
                    if refs == 0:
                        freed.append(slot)
                    else:
                        self._refs[slot] = refs

            # --- deliveries: store into the receiver's accumulator
            if cfg.online_prob >= 1.0:
                online = np.ones(cfg.n_nodes, dtype=bool)
            else:
                online_u = self.tape.uniform(Purpose.ONLINE, t, cfg.n_nodes)
                online = online_u <= cfg.online_prob
            for receiver, slot, _rf, _ip, sender, _pid in self._pending.pop(t, []):
                if not online[receiver]:
                    failed += 1
                    self._deref(slot, freed)
                    continue
                acc = self._acc[receiver]
                for j, (s0, slot0) in enumerate(acc):
                    if s0 == sender:  # replace stale model from this sender
                        self._deref(slot0, freed)
                        acc[j] = (sender, slot)
                        break
                else:
                    acc.append((sender, slot))

            phase = TickPhase(
                t=t,
                snap_nodes=np.asarray(snap_nodes, dtype=np.int32),
                snap_slots=np.asarray(snap_slots, dtype=np.int32),
                recv_nodes=np.zeros(0, dtype=np.int32),
                recv_ptr=np.zeros(1, dtype=np.int32),
                del_slots=np.zeros(0, dtype=np.int32),
                del_owners=np.zeros(0, dtype=np.int32),
                reply_slots=np.zeros(0, dtype=np.int32),
                pull_snap_nodes=np.zeros(0, dtype=np.int32),
                pull_snap_slots=np.zeros(0, dtype=np.int32),
                wm_nodes=np.asarray(wm_nodes, dtype=np.int32),
                wm_ptr=np.asarray(wm_ptr, dtype=np.int32),
                wm_slots=np.asarray(wm_slots, dtype=np.int32),
                wm_weights=np.asarray(wm_weights, dtype=np.float32),
                wm_owners=np.asarray(wm_owners, dtype=np.int32),
                wm_self_w=np.asarray(wm_self_w, dtype=np.float32),
            )
            if phase.n_events:
                ticks.append(phase)
            self._recycle(t, freed)

        eval_nodes = None
        if cfg.sampling_eval > 0:
            g = self.tape.stream(Purpose.EVAL, t1 - 1)
            k = max(int(cfg.n_nodes * cfg.sampling_eval), 1)
            eval_nodes = np.atleast_1d(g.integers(0, cfg.n_nodes, size=k))

        return RoundSchedule(
            round_idx=r,
            ticks=ticks,
            n_slots=self._next_slot,
            slot_owner=self.slot_owner[: self._next_slot].copy(),
            sent_messages=sent,
            failed_messages=failed,
            total_size=total_size,
            eval_nodes=eval_nodes,
        )

    def _wm_delay(self, gdl) -> int:
        d = self.cfg.delay
        if isinstance(d, ConstantDelay):
            return d._delay
        if isinstance(d, UniformDelay):
            return int(gdl.integers(d._min_delay, d._max_delay + 1))
        if isinstance(d, LinearDelay):
            return int(d._timexunit * self.cfg.model_size) + d._overhead
        return int(d.get(None))


class CacheNeighScheduler(Scheduler):
    """Cache-neighborhood gossip (Giaretta 2019; CacheNeighNode,
    gossipy/node.py:395-496).

    A receiver stores each incoming model in a per-sender slot (replacing
    any previous one); when a node times out it first pops a *random*
    cached slot and runs the normal merge+update with it, then snapshots
    and pushes its model. The reference's slot-pick
    (``random.choice(set(...))``, gossipy/node.py:449) crashes as shipped
    (SURVEY.md §2.3 quirk 7); like the object layer, the engine draws
    uniformly over the senders present (sorted order, tape-driven).

    Emitted as two phases per tick: phase 1 delivers the popped-cache
    merges (ordinary delivery CSR — no new kernels), phase 2 carries the
    post-merge snapshots; stores at delivery time are host-side only.
    """

    def __init__(self, cfg: EngineConfig):
        super().__init__(cfg)
        #: per-node cache: sender -> slot (insertion-ordered)
        self._acc: List[Dict[int, int]] = [dict() for _ in range(cfg.n_nodes)]
        self._refs: Dict[int, int] = {}

    def _deref(self, slot: int, freed: List[int]) -> None:
        r = self._refs.get(slot, 1) - 1
        if r <= 0:
            self._refs.pop(slot, None)
            freed.append(slot)
        else:
            self._refs[slot] = r

    def next_round(self, r: int) -> RoundSchedule:
        cfg = self.cfg
        proto = cfg.protocol
        assert proto != AntiEntropyProtocol.PULL, (
            "CacheNeigh PULL requests carry no model; use PUSH or PUSH_PULL"
        )
        t0, t1 = r * cfg.delta, (r + 1) * cfg.delta
        sent = failed = total_size = 0
        ticks: List[TickPhase] = []

        for t in range(t0, t1):
            freed: List[int] = []
            firing = self._firing(t)

            # --- phase 1: pop-a-random-cached-model merges (node.py:442-474)
            m_nodes: List[int] = []
            m_slots: List[int] = []
            m_owners: List[int] = []
            pick = self.tape.stream(Purpose.MISC, t)
            for i in firing:
                acc = self._acc[int(i)]
                if not acc:
                    continue
                senders = sorted(acc)
                kpick = senders[int(pick.integers(0, len(senders)))]
                slot = acc.pop(kpick)
                m_nodes.append(int(i))
                m_slots.append(slot)
                m_owners.append(kpick)
                self._deref(slot, freed)
            if m_nodes:
                ptr = np.arange(len(m_nodes) + 1, dtype=np.int32)
                ticks.append(
                    TickPhase(
                        t=t,
                        snap_nodes=np.zeros(0, dtype=np.int32),
                        snap_slots=np.zeros(0, dtype=np.int32),
                        recv_nodes=np.asarray(m_nodes, dtype=np.int32),
                        recv_ptr=ptr,
                        del_slots=np.asarray(m_slots, dtype=np.int32),
                        del_owners=np.asarray(m_owners, dtype=np.int32),
                        reply_slots=np.full(len(m_nodes), -1, dtype=np.int32),
                        pull_snap_nodes=np.zeros(0, dtype=np.int32),
                        pull_snap_slots=np.zeros(0, dtype=np.int32),
                        del_pids=np.full(len(m_nodes), -1, dtype=np.int32),
                    )
                )

            # --- phase 2: post-merge snapshots + sends (node.py:445-460)
            snap_nodes: List[int] = []
            snap_slots: List[int] = []
            reply_snap_nodes: List[int] = []
            reply_snap_slots: List[int] = []
            n_f = len(firing)
            if n_f:
                peers = self._peers_of(firing, t)
                drop_u = self.tape.uniform(Purpose.DROP, t, n_f)
                sizes = np.full(n_f, cfg.model_size)
                delays = self._delays(t, n_f, sizes)
                for j in range(n_f):
                    sender, receiver = int(firing[j]), int(peers[j])
                    slot = self._alloc_slot(sender, t)
                    snap_nodes.append(sender)
                    snap_slots.append(slot)
                    sent += 1
                    total_size += int(sizes[j])
                    if drop_u[j] >= cfg.drop_prob:
                        wants = proto == AntiEntropyProtocol.PUSH_PULL
                        self._pending.setdefault(t + int(delays[j]), []).append(
                            (receiver, slot, -2 if wants else -1, False, sender, -1)
                        )
                        self._refs[slot] = 1
                    else:
                        failed += 1
                        freed.append(slot)

            # --- deliveries: store into the receiver's per-sender slot
            # (node.py:477-496); PUSH_PULL triggers a reply snapshot of the
            # receiver's CURRENT model (no merge)
            if cfg.online_prob >= 1.0:
                online = np.ones(cfg.n_nodes, dtype=bool)
            else:
                online_u = self.tape.uniform(Purpose.ONLINE, t, cfg.n_nodes)
                online = online_u <= cfg.online_prob
            # loop: zero-delay PUSH_PULL replies land back in _pending[t]
            # and are stored the same tick (the reference's rep_queues[t])
            while True:
                due = self._pending.pop(t, [])
                if not due:
                    break
                for receiver, slot, rf, _ip, sender, _pid in due:
                    if not online[receiver]:
                        failed += 1
                        self._deref(slot, freed)
                        continue
                    if rf == -2:  # reply with own snapshot (no merge)
                        rslot = self._alloc_slot(receiver, t)
                        reply_snap_nodes.append(receiver)
                        reply_snap_slots.append(rslot)
                        if self._enqueue_reply(t, receiver, sender, rslot):
                            self._refs[rslot] = 1
                        else:
                            freed.append(rslot)
                    acc = self._acc[receiver]
                    old = acc.get(sender)
                    if old is not None:
                        self._deref(old, freed)
                    acc[sender] = slot

            phase2 = TickPhase(
                t=t,
                snap_nodes=np.asarray(snap_nodes, dtype=np.int32),
                snap_slots=np.asarray(snap_slots, dtype=np.int32),
                recv_nodes=np.zeros(0, dtype=np.int32),
                recv_ptr=np.zeros(1, dtype=np.int32),
                del_slots=np.zeros(0, dtype=np.int32),
                del_owners=np.zeros(0, dtype=np.int32),
                reply_slots=np.zeros(0, dtype=np.int32),
                pull_snap_nodes=np.asarray(reply_snap_nodes, dtype=np.int32),
                pull_snap_slots=np.asarray(reply_snap_slots, dtype=np.int32),
                del_pids=np.zeros(0, dtype=np.int32),
            )
            if phase2.n_events:
                ticks.append(phase2)

            sent_r, failed_r, size_r = self._reply_accounting.pop(t, (0, 0, 0))
            sent += sent_r
            failed += failed_r
            total_size += size_r
            self._recycle(t, freed)

        eval_nodes = None
        if cfg.sampling_eval > 0:
            g = self.tape.stream(Purpose.EVAL, t1 - 1)
            k = max(int(cfg.n_nodes * cfg.sampling_eval), 1)
            eval_nodes = np.atleast_1d(g.integers(0, cfg.n_nodes, size=k))

        return RoundSchedule(
            round_idx=r,
            ticks=ticks,
            n_slots=self._next_slot,
            slot_owner=self.slot_owner[: self._next_slot].copy(),
            sent_messages=sent,
            failed_messages=failed,
            total_size=total_size,
            eval_nodes=eval_nodes,
        )


class PENSScheduler(Scheduler):
    """PENS — performance-based neighbor selection (Onoszko 2021;
    PENSNode, gossipy/node.py:663-785). PUSH only, MERGE_UPDATE only.

    Step 1 (rounds ``< step1_rounds``): a receiver caches one model slot
    per sender; when the cache reaches ``n_sampled`` the schedule emits a
    *PENS event* listing all candidates — the kernel scores each candidate
    on the receiver's train data, merges the top-``m_top`` (mean incl. own
    model) and counts the winners device-side; the schedule itself never
    needs the accuracies. The node-side ``selected`` counters (how often a
    peer was drawn as a send target, gossipy/node.py:746-749) are pure
    host state.

    At the step boundary the runner reads the device-side winner counts
    and calls :meth:`select_neighbors`; step-2 peer draws are restricted
    to each node's ``best_nodes`` (count > selected * m/n — node.py:
    726-730), falling back to uniform when empty.
    """

    def __init__(self, cfg: EngineConfig, n_sampled: int = 10, m_top: int = 2,
                 step1_rounds: int = 10):
        super().__init__(cfg)
        assert cfg.protocol == AntiEntropyProtocol.PUSH, (
            "PENSNode only supports PUSH (gossipy/node.py:758-763)"
        )
        self.n_sampled = n_sampled
        self.m_top = m_top
        self.step1_rounds = step1_rounds
        #: per-node candidate cache: sender -> slot, arrival-ordered
        self._acc: List[Dict[int, int]] = [dict() for _ in range(cfg.n_nodes)]
        self._refs: Dict[int, int] = {}
        #: host-side target-draw counters (gossipy/node.py:746-749)
        self.selected = np.zeros((cfg.n_nodes, cfg.n_nodes), dtype=np.int64)
        self.best_nodes: Optional[List[np.ndarray]] = None

    def select_neighbors(self, counts: np.ndarray) -> None:
        """``counts[i, j]``: how often j's model made i's top-m (from the
        device counter buffer). Computes each node's ``best_nodes``
        (gossipy/node.py:728-730)."""
        ratio = self.m_top / self.n_sampled
        self.best_nodes = []
        for i in range(self.cfg.n_nodes):
            peers = self._peer_candidates(i)
            best = peers[counts[i, peers] > self.selected[i, peers] * ratio]
            self.best_nodes.append(best.astype(np.int64))

    def _peer_candidates(self, i: int) -> np.ndarray:
        cfg = self.cfg
        if cfg.peers_indptr is None:
            return np.concatenate([np.arange(i), np.arange(i + 1, cfg.n_nodes)])
        s = int(cfg.peers_indptr[i])
        e = int(cfg.peers_indptr[i + 1])
        return np.asarray(cfg.peers_indices[s:e])

    def _deref(self, slot: int, freed: List[int]) -> None:
        r = self._refs.get(slot, 1) - 1
        if r <= 0:
            self._refs.pop(slot, None)
            freed.append(slot)
        else:
            self._refs[slot] = r

    def _peers_of(self, nodes: np.ndarray, t: int) -> np.ndarray:
        """Step-2 peer draws come from best_nodes (node.py:751-755);
        step 1 uses the base draw and bumps the selected counters."""
        in_step2 = self.best_nodes is not None
        if not in_step2:
            peers = super()._peers_of(nodes, t)
            for i, p in zip(nodes, peers):
                self.selected[int(i), int(p)] += 1
            return peers
        g = self.tape.stream(Purpose.PEER, t)
        n = len(nodes)
        if n == 0:
            return np.empty(0, dtype=np.int64)
        u = np.atleast_1d(g.random(n))
        out = np.empty(n, dtype=np.int64)
        for j, i in enumerate(nodes):
            best = self.best_nodes[int(i)]
            cands = best if len(best) else self._peer_candidates(int(i))
            out[j] = int(cands[int(np.floor(u[j] * len(cands)))])
        return out

    def next_round(self, r: int) -> RoundSchedule:
        cfg = self.cfg
        t0, t1 = r * cfg.delta, (r + 1) * cfg.delta
        in_step1 = r < self.step1_rounds
        sent = failed = total_size = 0
        ticks: List[TickPhase] = []

        for t in range(t0, t1):
            freed: List[int] = []
            firing = self._firing(t)

            snap_nodes: List[int] = []
            snap_slots: List[int] = []
            n_f = len(firing)
            if n_f:
                peers = self._peers_of(firing, t)
                if cfg.drop_prob <= 0.0:
                    drop_u = np.ones(n_f)
                else:
                    drop_u = self.tape.uniform(Purpose.DROP, t, n_f)
                delays = self._delays(t, n_f, np.full(n_f, cfg.model_size))
                for j in range(n_f):
                    sender, receiver = int(firing[j]), int(peers[j])
                    slot = self._alloc_slot(sender, t)
                    snap_nodes.append(sender)
                    snap_slots.append(slot)
                    sent += 1
                    total_size += cfg.model_size
                    if drop_u[j] >= cfg.drop_prob:
                        self._pending.setdefault(t + int(delays[j]), []).append(
                            (receiver, slot, -1, False, sender, -1)
                        )
                        self._refs[slot] = 1
                    else:
                        failed += 1
                        freed.append(slot)

            if cfg.online_prob >= 1.0:
                online = np.ones(cfg.n_nodes, dtype=bool)
            else:
                online_u = self.tape.uniform(Purpose.ONLINE, t, cfg.n_nodes)
                online = online_u <= cfg.online_prob

            # deliveries
            pens_nodes: List[int] = []
            pens_ptr = [0]
            pens_slots: List[int] = []
            pens_owners: List[int] = []
            recv_map: Dict[int, List[Tuple[int, int]]] = {}
            for receiver, slot, _rf, _ip, sender, _pid in self._pending.pop(t, []):
                if not online[receiver]:
                    failed += 1
                    self._deref(slot, freed)
                    continue
                if in_step1:
                    acc = self._acc[receiver]
                    old = acc.get(sender)
                    if old is not None:
                        self._deref(old, freed)
                    acc[sender] = slot
                    if len(acc) >= self.n_sampled:
                        # emit the scoring/merge event with every candidate
                        pens_nodes.append(receiver)
                        for snd, sl in acc.items():
                            pens_slots.append(sl)
                            pens_owners.append(snd)
                            self._deref(sl, freed)
                        pens_ptr.append(len(pens_slots))
                        acc.clear()
                else:
                    # step 2: ordinary MERGE_UPDATE delivery
                    recv_map.setdefault(receiver, []).append((slot, sender))
                    self._deref(slot, freed)

            recv_nodes = np.fromiter(recv_map.keys(), dtype=np.int32, count=len(recv_map))
            recv_ptr = np.zeros(len(recv_map) + 1, dtype=np.int32)
            del_slots: List[int] = []
            del_owners: List[int] = []
            for i, rn in enumerate(recv_nodes):
                pairs = recv_map[int(rn)]
                del_slots.extend(p[0] for p in pairs)
                del_owners.extend(p[1] for p in pairs)
                recv_ptr[i + 1] = recv_ptr[i] + len(pairs)

            phase = TickPhase(
                t=t,
                snap_nodes=np.asarray(snap_nodes, dtype=np.int32),
                snap_slots=np.asarray(snap_slots, dtype=np.int32),
                recv_nodes=recv_nodes,
                recv_ptr=recv_ptr,
                del_slots=np.asarray(del_slots, dtype=np.int32),
                del_owners=np.asarray(del_owners, dtype=np.int32),
                reply_slots=np.full(len(del_slots), -1, dtype=np.int32),
                pull_snap_nodes=np.zeros(0, dtype=np.int32),
                pull_snap_slots=np.zeros(0, dtype=np.int32),
                del_pids=np.full(len(del_slots), -1, dtype=np.int32),
                pens_nodes=np.asarray(pens_nodes, dtype=np.int32),
                pens_ptr=np.asarray(pens_ptr, dtype=np.int32),
                pens_slots=np.asarray(pens_slots, dtype=np.int32),
                pens_owners=np.asarray(pens_owners, dtype=np.int32),
            )
            if phase.n_events:
                ticks.append(phase)
            self._recycle(t, freed)

        eval_nodes = None
        if cfg.sampling_eval > 0:
            g = self.tape.stream(Purpose.EVAL, t1 - 1)
            k = max(int(cfg.n_nodes * cfg.sampling_eval), 1)
            eval_nodes = np.atleast_1d(g.integers(0, cfg.n_nodes, size=k))

        return RoundSchedule(
            round_idx=r,
            ticks=ticks,
            n_slots=self._next_slot,
            slot_owner=self.slot_owner[: self._next_slot].copy(),
            sent_messages=sent,
            failed_messages=failed,
            total_size=total_size,
            eval_nodes=eval_nodes,
        )


class NativeSchedulerAdapter:
    """Adapter over the C++ scheduler (``csrc/scheduler.cpp``).

    Produces the same :class:`RoundSchedule`/:class:`TickPhase` objects as
    the Python :class:`Scheduler` (bit-exact — enforced by
    tests/test_native_sched.py) from the native flat per-round arrays, and
    exposes the flat arrays themselves (``last_flat``) for the GPU round
    executor's one-upload-per-round fast path.
    """

    def __init__(self, cfg: EngineConfig):
        from .. import ops

        mod = ops.load_sched()
        if mod is None:
            raise ImportError("_gossip_sched.so not built")
        self.cfg = cfg
        d = cfg.delay
        if isinstance(d, ConstantDelay):
            kind, dmin, dmax, tx, ov = 0, d._delay, d._delay, 0.0, 0
        elif isinstance(d, UniformDelay):
            kind, dmin, dmax, tx, ov = 1, d._min_delay, d._max_delay, 0.0, 0
        elif isinstance(d, LinearDelay):
            kind, dmin, dmax, tx, ov = 2, 0, 0, d._timexunit, d._overhead
        else:
            raise TypeError("custom Delay subclasses need the python Scheduler")
        if cfg.pass_through:
            raise TypeError("pass-through gossip needs the python Scheduler")
        ip = cfg.peers_indptr
        ix = cfg.peers_indices
        self._native = mod.NativeScheduler(
            cfg.n_nodes,
            cfg.delta,
            int(cfg.protocol.value),
            cfg.model_size,
            cfg.drop_prob,
            cfg.online_prob,
            kind,
            dmin,
            dmax,
            tx,
            ov,
            cfg.sync,
            cfg.sampling_eval,
            cfg.seed & 0xFFFFFFFFFFFFFFFF,
            None if ip is None else np.ascontiguousarray(ip, dtype=np.int64),
            None if ix is None else np.ascontiguousarray(ix, dtype=np.int64),
            cfg.n_parts,
            cfg.sampled,
        )
        self.last_flat: Optional[dict] = None

    def next_round(self, r: int) -> RoundSchedule:
        f = self._native.next_round(r)
        self.last_flat = f
        ticks: List[TickPhase] = []
        delta = self.cfg.delta
        t0 = r * delta
        for i in range(delta):
            s0, s1 = int(f["snap_tptr"][i]), int(f["snap_tptr"][i + 1])
            r0, r1 = int(f["recv_tptr"][i]), int(f["recv_tptr"][i + 1])
            p0, p1 = int(f["pull_tptr"][i]), int(f["pull_tptr"][i + 1])
            q0, q1 = int(f["rep_tptr"][i]), int(f["rep_tptr"][i + 1])
            if s0 == s1 and r0 == r1 and p0 == p1 and q0 == q1:
                continue
            d0, d1 = int(f["recv_nptr"][r0]), int(f["recv_nptr"][r1])
            e0, e1 = int(f["rep_nptr"][q0]), int(f["rep_nptr"][q1])
            ticks.append(
                TickPhase(
                    t=t0 + i,
                    snap_nodes=f["snap_nodes"][s0:s1],
                    snap_slots=f["snap_slots"][s0:s1],
                    recv_nodes=f["recv_nodes"][r0:r1],
                    recv_ptr=f["recv_nptr"][r0 : r1 + 1] - d0,
                    del_slots=f["del_slots"][d0:d1],
                    del_owners=f["del_owners"][d0:d1],
                    reply_slots=f["reply_slots"][d0:d1],
                    pull_snap_nodes=f["pull_nodes"][p0:p1],
                    pull_snap_slots=f["pull_slots"][p0:p1],
                    rep_recv_nodes=f["rep_nodes"][q0:q1],
                    rep_recv_ptr=f["rep_nptr"][q0 : q1 + 1] - e0,
                    rep_del_slots=f["rep_slots"][e0:e1],
                    rep_del_owners=f["rep_owners"][e0:e1],
                    del_pids=f["del_pids"][d0:d1],
                    rep_pids=f["rep_pids"][e0:e1],
                )
            )
        return RoundSchedule(
            round_idx=r,
            ticks=ticks,
            n_slots=int(f["n_slots"]),
            slot_owner=None,
            sent_messages=int(f["sent"]),
            failed_messages=int(f["failed"]),
            total_size=int(f["total_size"]),
            eval_nodes=f["eval_nodes"],
        )


class NativeTokenizedAdapter:
    """Adapter over the C++ tokenized scheduler (``NativeTokenizedScheduler``
    in csrc/scheduler.cpp) — bit-exact with :class:`TokenizedScheduler`
    (tests/test_native_sched.py) and emitting flat per-wave launch groups
    for the GPU round executor directly.

    Covers the 5 in-tree token-account strategies and constant utilities
    (the reference experiments use ``utility == 1``,
    main_hegedus_2021.py:57); custom accounts or callable utilities fall
    back to the python scheduler.
    """

    ACCOUNT_KINDS = {
        "PurelyProactiveTokenAccount": 0,
        "PurelyReactiveTokenAccount": 1,
        "SimpleTokenAccount": 2,
        "GeneralizedTokenAccount": 3,
        "RandomizedTokenAccount": 4,
    }

    def __init__(self, cfg: EngineConfig, token_account, utility: int = 1):
        from .. import ops

        mod = ops.load_sched()
        if mod is None or not hasattr(mod, "NativeTokenizedScheduler"):
            raise ImportError("_gossip_sched.so missing NativeTokenizedScheduler")
        kind = self.ACCOUNT_KINDS.get(type(token_account).__name__)
        if kind is None:
            raise TypeError("custom token accounts need the python scheduler")
        if cfg.pass_through:
            raise TypeError("pass-through needs the python scheduler")
        d = cfg.delay
        if isinstance(d, ConstantDelay):
            dk, dmin, dmax, tx, ov = 0, d._delay, d._delay, 0.0, 0
        elif isinstance(d, UniformDelay):
            dk, dmin, dmax, tx, ov = 1, d._min_delay, d._max_delay, 0.0, 0
        elif isinstance(d, LinearDelay):
            dk, dmin, dmax, tx, ov = 2, 0, 0, d._timexunit, d._overhead
        else:
            raise TypeError("custom Delay subclasses need the python Scheduler")
        self.cfg = cfg
        ip, ix = cfg.peers_indptr, cfg.peers_indices
        self._native = mod.NativeTokenizedScheduler(
            cfg.n_nodes,
            cfg.delta,
            int(cfg.protocol.value),
            cfg.model_size,
            cfg.drop_prob,
            cfg.online_prob,
            dk,
            dmin,
            dmax,
            tx,
            ov,
            cfg.sync,
            cfg.sampling_eval,
            cfg.seed & 0xFFFFFFFFFFFFFFFF,
            None if ip is None else np.ascontiguousarray(ip, dtype=np.int64),
            None if ix is None else np.ascontiguousarray(ix, dtype=np.int64),
            cfg.n_parts,
            cfg.sampled,
            kind,
            float(getattr(token_account, "capacity", 0.0)),
            float(getattr(token_account, "reactivity", 0.0)),
            float(getattr(token_account, "k", 0.0)),
            int(utility),
        )
        self.last_flat: Optional[dict] = None

    def token_balances(self):
        return list(self._native.token_balances())

    def next_round_flat(self, r: int) -> RoundSchedule:
        f = self._native.next_round(r)
        self.last_flat = f
        return _flat_round_summary(f, r)

    def set_lean(self, v: bool) -> None:
        self._native.set_lean(v)

    def set_lean(self, v: bool) -> None:
        """Emit only the packed schedule + tick pointers (the single-rank
        fast path consumes nothing else)."""
        self._native.set_lean(v)

    # the per-tick path is not used with this adapter (flat-exec only)
    next_round = next_round_flat


def make_scheduler(cfg: EngineConfig):
    """Native scheduler when available & applicable, else the Python one."""
    try:
        return NativeSchedulerAdapter(cfg)
    except (ImportError, TypeError):
        return Scheduler(cfg)


def _flat_round_summary(f: dict, r: int) -> RoundSchedule:
    return RoundSchedule(
        round_idx=r,
        ticks=[],
        n_slots=int(f["n_slots"]),
        slot_owner=None,
        sent_messages=int(f["sent"]),
        failed_messages=int(f["failed"]),
        total_size=int(f["total_size"]),
        eval_nodes=f["eval_nodes"],
    )


def _adapter_next_round_flat(self, r: int) -> RoundSchedule:
    """Fast-path variant of next_round: skips the TickPhase conversion and
    leaves the flat arrays in ``last_flat`` for the GPU round executor."""
    f = self._native.next_round(r)
    self.last_flat = f
    return _flat_round_summary(f, r)


NativeSchedulerAdapter.next_round_flat = _adapter_next_round_flat


def _adapter_set_lean(self, v: bool) -> None:
    """Emit only the packed schedule + tick pointers (the single-rank fast
    path consumes nothing else — skips the full per-event arrays and the
    multi-rank merge scan)."""
    self._native.set_lean(v)


NativeSchedulerAdapter.set_lean = _adapter_set_lean
