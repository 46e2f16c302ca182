"""The batched gossip runtime: node-sharded, kernel-batched, RCCL-connected.

This is the MI355X replacement of the object layer's per-node event loop
(gossipy/simul.py:366-458). Design:

* N simulated nodes are sharded across ranks in contiguous blocks, one
  process per GPU (``torch.distributed`` over RCCL/xGMI);
* every rank derives the *same* round schedule from the deterministic
  random tape, so there is no control-plane traffic at all — only model
  rows move between GPUs;
* the scheduler packs a round at ENTRY granularity (deliveries to one
  receiver coalesce across ticks into one CSR row, snapshots of receiving
  nodes embed as per-delivery reply writes), so the runner issues a
  handful of hazard-free launch groups per round — each group at most a
  snapshot launch and two deliver launches. Snapshot slots whose writer
  and consumer live on different GPUs travel as ONE grouped
  ``batch_isend_irecv`` transfer per group (RCCL p2p over xGMI) between
  the launches;
* the round-end evaluation sweep runs batched on-device, and only metric
  dicts are gathered to rank 0 (C5).

The public surface mirrors :class:`gossipy_amd.simul.GossipSimulator`
(receivers, report, ``init_nodes``/``start``) so existing observer code
works unchanged.
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from .. import LOG
from ..core import CreateModelMode
from ..simul import SimulationEventSender
from .arena import DataArena, NodeStateArena, SlotPool
from .backend import make_backend
from .metrics import binary_margin_metrics, classification_metrics_shared
from .rng import RandomTape
from .schedule import (
    EngineConfig,
    RoundSchedule,
    Scheduler,
    TickPhase,
    TokenizedScheduler,
    make_scheduler,
)

__all__ = [
    "BatchedGossipSimulator",
    "BatchedTokenizedGossipSimulator",
    "BatchedAll2AllGossipSimulator",
    "BatchedCacheNeighGossipSimulator",
    "BatchedPENSGossipSimulator",
    "RoundTimer",
]


def _csr_gather(ptr: np.ndarray, rows: np.ndarray):
    """Select CSR rows: returns (flat element indices, new local ptr)."""
    ptr = np.asarray(ptr)
    counts = np.diff(ptr)[rows]
    new_ptr = np.zeros(len(counts) + 1, dtype=np.int64)
    np.cumsum(counts, out=new_ptr[1:])
    sel = np.repeat(ptr[rows] - new_ptr[:-1], counts) + np.arange(
        int(new_ptr[-1])
    )
    return sel, new_ptr


class BatchedGossipSimulator(SimulationEventSender):
    """Node-batched gossip simulator for one or many GPUs (or CPU).

    Parameters
    ----------
    cfg : EngineConfig
        Simulation parameters (shared by all ranks).
    spec
        Model-family spec (:mod:`gossipy_amd.engine.models`).
    data : DataArena
        This rank's resident nodes' data shards (+ the global eval set).
    device : torch.device, optional
        Defaults to ``cuda:LOCAL_RANK`` when available, else CPU.
    """

    def __init__(
        self,
        cfg: EngineConfig,
        spec,
        data: DataArena,
        device: Optional[torch.device] = None,
    ):
        super().__init__()
        self.cfg = cfg
        self.spec = spec
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        assert cfg.n_nodes % self.world == 0, (
            "n_nodes must be divisible by the world size"
        )
        self.n_local = cfg.n_nodes // self.world
        self.node_lo = self.rank * self.n_local
        self.node_hi = self.node_lo + self.n_local

        if device is None:
            if torch.cuda.is_available():
                local = int(os.environ.get("LOCAL_RANK", self.rank))
                device = torch.device(f"cuda:{local}")
            else:
                device = torch.device("cpu")
        self.device = device
        if device.type == "cuda":
            torch.cuda.set_device(device)

        self.backend = make_backend(device)
        aw = getattr(spec, "age_width", 1)
        self.state = NodeStateArena(
            self.n_local, spec.D, device, self.node_lo, age_width=aw
        )
        # MF ships only the item block; other families ship the full row
        self.pool = SlotPool(
            getattr(spec, "slot_width", spec.D), device, age_width=aw
        )
        self._snap_off = getattr(spec, "item_off", 0)
        self.data = data
        self.scheduler = make_scheduler(cfg)
        self.initialized = False
        self.rounds_done = 0
        #: subclasses with non-standard phases (wm/pens) opt out of the
        #: flatten-to-round-executor path
        self._flat_schedulable = True
        #: host mirror of per-round slot owners (set per round)
        self._slot_owner: Optional[np.ndarray] = None
        #: single worker thread that drives the (GIL-releasing) C++ round
        #: executors so schedule prefetch overlaps the launch queue
        self._exec_pool = None

    # -- residency helpers ---------------------------------------------------

    def _rank_of(self, nodes: np.ndarray) -> np.ndarray:
        return nodes // self.n_local

    def _is_mine(self, nodes: np.ndarray) -> np.ndarray:
        return (nodes >= self.node_lo) & (nodes < self.node_hi)

    def _to_local_t(self, nodes: np.ndarray) -> torch.Tensor:
        return torch.from_numpy((nodes - self.node_lo).astype(np.int64)).to(self.device)

    # -- setup ---------------------------------------------------------------

    def init_nodes(self, seed: Optional[int] = None) -> None:
        """Initialize all resident models and run the reference's one
        initial local training pass (gossipy/node.py:82-94)."""
        tape = RandomTape(seed if seed is not None else self.cfg.seed)
        self.backend.init_params(self.state, self.spec, tape, self.cfg.n_nodes)
        all_local = torch.arange(self.n_local)
        self.backend.update(self.state, self.data, self.spec, all_local)
        if self.world > 1:
            # collective world-communicator init before any pairwise
            # batch_isend_irecv (lazy NCCL init on a p2p group deadlocks
            # ranks that have no op in it)
            dist.barrier()
        self.initialized = True

    # -- cross-GPU slot exchange ---------------------------------------------

    def _exchange(
        self,
        needed: List[Tuple[int, np.ndarray]],
    ) -> None:
        """Move snapshot slots between ranks.

        ``needed`` is a list of ``(src_rank, dst_rank, slot_ids)`` triples
        (identical on every rank — derived from the shared schedule). Rows
        travel as fp32 ``[n, D+1]`` buffers (last column = age) via grouped
        point-to-point sends, which RCCL maps onto the direct xGMI link of
        each GPU pair.
        """
        self._exchange_finish(self._exchange_start(needed))

    def _exchange_start(self, needed):
        """Issue the grouped P2P ops; returns the in-flight state for
        :meth:`_exchange_finish` (None when there is nothing to move).

        All of this rank's outgoing rows gather into ONE staging buffer
        (one fancy-index per tick, per-destination views feed the isends);
        incoming rows land in one buffer scattered back with one
        index-copy — at 8 GPUs this replaces up to 7+7 small gathers per
        tick with 2."""
        if self.world == 1 or not needed:
            return None
        D = self.pool.slots.shape[1]
        A = getattr(self.spec, "age_width", 1)
        send_parts = []  # (dst, count)
        send_ids = []
        recv_parts = []  # (src, count)
        recv_ids = []
        for src, dst, slot_ids in needed:
            if src == dst:
                continue
            if src == self.rank:
                send_parts.append((dst, len(slot_ids)))
                send_ids.append(slot_ids)
            elif dst == self.rank:
                recv_parts.append((src, len(slot_ids)))
                recv_ids.append(slot_ids)
        ops = []
        send_buf = recv_buf = None
        if send_ids:
            ids = torch.from_numpy(
                np.concatenate(send_ids).astype(np.int64)
            ).to(self.device)
            send_buf = torch.empty(len(ids), D + A, device=self.device)
            send_buf[:, :D] = self.pool.slots[ids]
            send_buf[:, D:] = self.pool.slot_ages[ids].reshape(len(ids), A).float()
            off = 0
            for dst, n in send_parts:
                ops.append(dist.P2POp(dist.isend, send_buf[off : off + n], dst))
                off += n
        if recv_ids:
            total = sum(n for _, n in recv_parts)
            recv_buf = torch.empty(total, D + A, device=self.device)
            off = 0
            for src, n in recv_parts:
                ops.append(dist.P2POp(dist.irecv, recv_buf[off : off + n], src))
                off += n
        works = dist.batch_isend_irecv(ops) if ops else []
        return (works, recv_ids, recv_buf, send_buf, D)

    def _exchange_finish(self, pending) -> None:
        if pending is None:
            return
        works, recv_ids, recv_buf, _send_buf, D = pending
        for w in works:
            w.wait()
        if recv_buf is not None:
            ids = torch.from_numpy(
                np.concatenate(recv_ids).astype(np.int64)
            ).to(self.device)
            self.pool.slots[ids] = recv_buf[:, :D]
            ages = recv_buf[:, D:].int()
            self.pool.slot_ages[ids] = ages.reshape(
                self.pool.slot_ages[ids].shape
            )

    def _plan_exchange(
        self,
        recv_nodes: np.ndarray,
        recv_ptr: np.ndarray,
        slots: np.ndarray,
        owners: np.ndarray,
    ) -> List[Tuple[int, int, np.ndarray]]:
        """(src, dst, slots) transfer plan for one delivery CSR; ``owners``
        is the per-delivery writer node (carried in the phase because slot
        ids are recycled)."""
        if self.world == 1 or len(slots) == 0:
            return []
        src = self._rank_of(owners)
        dst = self._rank_of(np.repeat(recv_nodes, np.diff(recv_ptr)))
        cross = src != dst
        if not cross.any():
            return []
        cs, cd, csl = src[cross], dst[cross], slots[cross]
        # group by (src, dst) and dedupe slots, fully vectorized
        order = np.lexsort((csl, cd, cs))
        cs, cd, csl = cs[order], cd[order], csl[order]
        new_grp = np.empty(len(cs), dtype=bool)
        new_grp[0] = True
        new_grp[1:] = (cs[1:] != cs[:-1]) | (cd[1:] != cd[:-1])
        starts = np.flatnonzero(new_grp)
        ends = np.append(starts[1:], len(cs))
        out = []
        for a, b in zip(starts, ends):
            sl = csl[a:b]
            uniq = np.unique(sl)
            out.append((int(cs[a]), int(cd[a]), uniq.astype(np.int64)))
        return out

    # -- round execution -----------------------------------------------------

    def _deliver_group(self, phase: TickPhase, recv_idx: np.ndarray) -> None:
        """Launch the delivery kernel for the receiver rows ``recv_idx``
        (indices into phase.recv_nodes)."""
        if len(recv_idx) == 0:
            return
        counts = np.diff(phase.recv_ptr)[recv_idx]
        # vectorized gather of each kept receiver's delivery range
        new_ptr = np.zeros(len(counts) + 1, dtype=np.int64)
        np.cumsum(counts, out=new_ptr[1:])
        sel = (
            np.repeat(
                np.asarray(phase.recv_ptr)[recv_idx]
                - new_ptr[:-1],
                counts,
            )
            + np.arange(int(new_ptr[-1]))
        )
        pids = None
        if phase.del_pids is not None and len(phase.del_pids):
            pids = torch.from_numpy(phase.del_pids[sel].astype(np.int64))
        self.backend.deliver(
            self.state,
            self.pool,
            self.data,
            self.spec,
            self._to_local_t(phase.recv_nodes[recv_idx]),
            torch.from_numpy(new_ptr),
            torch.from_numpy(phase.del_slots[sel].astype(np.int64)),
            torch.from_numpy(phase.reply_slots[sel].astype(np.int64)),
            del_pids=pids,
        )

    def _run_round_multi(self, f: dict) -> None:
        """Multi-rank whole-round execution from the flat schedule arrays:
        all per-rank filtering and index math happens ONCE per round
        (vectorized numpy + one device upload); the per-tick loop issues
        only kernel launches and the RCCL slot exchanges. Local-sourced
        deliveries launch while the exchange for remote-sourced ones is in
        flight."""
        dev = self.device
        delta = len(f["snap_tptr"]) - 1
        lo, n_local = self.node_lo, self.n_local

        def _tickify(nodes, tptr):
            """(mask of mine, tick id per kept entry, per-tick ptr)."""
            nodes = np.asarray(nodes)
            mask = (nodes >= lo) & (nodes < lo + n_local)
            tick_of = np.repeat(np.arange(delta), np.diff(tptr))[mask]
            ptr = np.zeros(delta + 1, dtype=np.int64)
            np.cumsum(np.bincount(tick_of, minlength=delta), out=ptr[1:])
            return mask, ptr

        def _dev32(a):
            return torch.from_numpy(np.ascontiguousarray(a, dtype=np.int32)).to(
                dev, non_blocking=True
            )

        # --- snapshots / pull snapshots
        s_mask, s_ptr = _tickify(f["snap_nodes"], f["snap_tptr"])
        snap_nodes_d = _dev32(np.asarray(f["snap_nodes"])[s_mask] - lo)
        snap_slots_d = _dev32(np.asarray(f["snap_slots"])[s_mask])
        p_mask, p_ptr = _tickify(f["pull_nodes"], f["pull_tptr"])
        pull_nodes_d = _dev32(np.asarray(f["pull_nodes"])[p_mask] - lo)
        pull_slots_d = _dev32(np.asarray(f["pull_slots"])[p_mask])

        def _deliver_prep(nodes_key, nptr_key, tptr_key, slots_key, reply_key,
                          pids_key, owners_key):
            """Per-round prep of one delivery CSR family. Returns per-tick
            launch slices for local-sourced and remote-sourced receiver
            groups plus the per-tick exchange plans."""
            recv_nodes = np.asarray(f[nodes_key])
            nptr = np.asarray(f[nptr_key])
            tptr = np.asarray(f[tptr_key])
            dslots = np.asarray(f[slots_key])
            owners = np.asarray(f[owners_key])
            n_rows = len(recv_nodes)
            if n_rows == 0:
                return None
            row_tick = np.repeat(np.arange(delta), np.diff(tptr))
            row_mine = (recv_nodes >= lo) & (recv_nodes < lo + n_local)
            src_rank = owners // n_local
            has_remote = (
                np.maximum.reduceat(src_rank != self.rank, nptr[:-1])
                if len(dslots)
                else np.zeros(n_rows, dtype=bool)
            ).astype(bool)

            # exchange plans, grouped per (tick, src, dst) in one lexsort
            d_tick = np.repeat(row_tick, np.diff(nptr))
            d_dst = np.repeat(recv_nodes // n_local, np.diff(nptr))
            cross = src_rank != d_dst
            plans = [[] for _ in range(delta)]
            if cross.any():
                ct, cs, cd2, csl = (
                    d_tick[cross], src_rank[cross], d_dst[cross], dslots[cross]
                )
                order = np.lexsort((csl, cd2, cs, ct))
                ct, cs, cd2, csl = ct[order], cs[order], cd2[order], csl[order]
                new_g = np.empty(len(ct), dtype=bool)
                new_g[0] = True
                new_g[1:] = (
                    (ct[1:] != ct[:-1]) | (cs[1:] != cs[:-1]) | (cd2[1:] != cd2[:-1])
                )
                starts = np.flatnonzero(new_g)
                ends = np.append(starts[1:], len(ct))
                for a, b in zip(starts, ends):
                    plans[int(ct[a])].append(
                        (int(cs[a]), int(cd2[a]), np.unique(csl[a:b]).astype(np.int64))
                    )

            groups = {}
            for name, rows_mask in (
                ("local", row_mine & ~has_remote),
                ("remote", row_mine & has_remote),
            ):
                rows = np.flatnonzero(rows_mask)
                sel, _ = _csr_gather(nptr, rows)
                # absolute ptr into the gathered arrays
                lens = np.diff(nptr)[rows]
                abs_ptr = np.zeros(len(rows) + 1, dtype=np.int64)
                np.cumsum(lens, out=abs_ptr[1:])
                rt = row_tick[rows]
                row_tptr = np.zeros(delta + 1, dtype=np.int64)
                np.cumsum(np.bincount(rt, minlength=delta), out=row_tptr[1:])
                pids = np.asarray(f[pids_key]) if pids_key else None
                groups[name] = {
                    "nodes": _dev32(recv_nodes[rows] - lo),
                    "ptr": _dev32(abs_ptr),
                    # host mirror: per-tick slicing must not sync the GPU
                    "ptr_host": abs_ptr,
                    "slots": _dev32(dslots[sel]) if len(sel) else _dev32([]),
                    "reply": _dev32(np.asarray(f[reply_key])[sel])
                    if reply_key
                    else None,
                    "pids": _dev32(pids[sel])
                    if pids is not None and len(pids)
                    else None,
                    "row_tptr": row_tptr,
                }
            return {"groups": groups, "plans": plans}

        dmain = _deliver_prep(
            "recv_nodes", "recv_nptr", "recv_tptr", "del_slots",
            "reply_slots", "del_pids", "del_owners",
        )
        drep = _deliver_prep(
            "rep_nodes", "rep_nptr", "rep_tptr", "rep_slots",
            "rep_reply_slots" if len(f.get("rep_reply_slots", ())) else None,
            "rep_pids", "rep_owners",
        )

        def _launch(g, t, reply_default=-1):
            a, b = int(g["row_tptr"][t]), int(g["row_tptr"][t + 1])
            if a == b:
                return
            lo_d = int(g["ptr_host"][a])
            hi_d = int(g["ptr_host"][b])
            reply = (
                g["reply"][lo_d:hi_d]
                if g["reply"] is not None
                else torch.full(
                    (hi_d - lo_d,), reply_default, dtype=torch.int32, device=dev
                )
            )
            self.backend.deliver(
                self.state,
                self.pool,
                self.data,
                self.spec,
                g["nodes"][a:b],
                g["ptr"][a : b + 1] - lo_d,
                g["slots"][:hi_d][lo_d:] if lo_d else g["slots"][:hi_d],
                reply,
                del_pids=g["pids"][lo_d:hi_d] if g["pids"] is not None else None,
            )

        for t in range(delta):
            a, b = int(s_ptr[t]), int(s_ptr[t + 1])
            if b > a:
                self.backend.snapshot(
                    self.state, self.pool, snap_nodes_d[a:b], snap_slots_d[a:b],
                    src_off=self._snap_off,
                )
            if dmain is not None:
                pending = self._exchange_start(dmain["plans"][t])
                _launch(dmain["groups"]["local"], t)
                self._exchange_finish(pending)
                _launch(dmain["groups"]["remote"], t)
            a, b = int(p_ptr[t]), int(p_ptr[t + 1])
            if b > a:
                self.backend.snapshot(
                    self.state, self.pool, pull_nodes_d[a:b], pull_slots_d[a:b],
                    src_off=self._snap_off,
                )
            if drep is not None:
                pending = self._exchange_start(drep["plans"][t])
                _launch(drep["groups"]["local"], t)
                self._exchange_finish(pending)
                _launch(drep["groups"]["remote"], t)

    def _run_round_groups(self, f: dict) -> None:
        """Single-rank whole-round replay of merged/packed launch groups
        through the PYTHON backend (families without a C++ executor —
        torchmod). One backend call-set per group gives the node-batched
        vmap paths their batch: a 100-tick round of 1-receiver ticks
        becomes a handful of ~20-receiver groups."""
        snap_nodes = np.asarray(f["snap_nodes"])
        snap_slots = np.asarray(f["snap_slots"])
        st = np.asarray(f["snap_tptr"])
        recv_nodes = np.asarray(f["recv_nodes"])
        nptr = np.asarray(f["recv_nptr"])
        rt = np.asarray(f["recv_tptr"])
        del_slots = np.asarray(f["del_slots"])
        reply_slots = np.asarray(f["reply_slots"])
        del_pids = np.asarray(f.get("del_pids", ()))
        pull_nodes = np.asarray(f["pull_nodes"])
        pull_slots = np.asarray(f["pull_slots"])
        pt = np.asarray(f["pull_tptr"])
        rep_nodes = np.asarray(f["rep_nodes"])
        rep_nptr = np.asarray(f["rep_nptr"])
        qt = np.asarray(f["rep_tptr"])
        rep_slots = np.asarray(f["rep_slots"])
        rep_pids = np.asarray(f.get("rep_pids", ()))
        rep_reply = np.asarray(f.get("rep_reply_slots", ()))

        def _t64(a):
            return torch.from_numpy(np.ascontiguousarray(a, dtype=np.int64))

        for g in range(len(st) - 1):
            a, b = int(st[g]), int(st[g + 1])
            if b > a:
                self.backend.snapshot(
                    self.state, self.pool,
                    _t64(snap_nodes[a:b]),
                    _t64(snap_slots[a:b]).to(self.device),
                    src_off=self._snap_off,
                )
            r0, r1 = int(rt[g]), int(rt[g + 1])
            if r1 > r0:
                d0, d1 = int(nptr[r0]), int(nptr[r1])
                self.backend.deliver(
                    self.state, self.pool, self.data, self.spec,
                    _t64(recv_nodes[r0:r1]),
                    _t64(nptr[r0 : r1 + 1] - d0),
                    _t64(del_slots[d0:d1]),
                    _t64(reply_slots[d0:d1]),
                    del_pids=_t64(del_pids[d0:d1]) if len(del_pids) else None,
                )
            a, b = int(pt[g]), int(pt[g + 1])
            if b > a:
                self.backend.snapshot(
                    self.state, self.pool,
                    _t64(pull_nodes[a:b]),
                    _t64(pull_slots[a:b]).to(self.device),
                    src_off=self._snap_off,
                )
            q0, q1 = int(qt[g]), int(qt[g + 1])
            if q1 > q0:
                e0, e1 = int(rep_nptr[q0]), int(rep_nptr[q1])
                rr = (
                    _t64(rep_reply[e0:e1])
                    if len(rep_reply)
                    else torch.full((e1 - e0,), -1, dtype=torch.int64)
                )
                self.backend.deliver(
                    self.state, self.pool, self.data, self.spec,
                    _t64(rep_nodes[q0:q1]),
                    _t64(rep_nptr[q0 : q1 + 1] - e0),
                    _t64(rep_slots[e0:e1]),
                    rr,
                    del_pids=_t64(rep_pids[e0:e1]) if len(rep_pids) else None,
                )

    def _run_tick(self, phase: TickPhase) -> None:
        # A: snapshots of firing nodes
        mine = self._is_mine(phase.snap_nodes)
        if mine.any():
            self.backend.snapshot(
                self.state,
                self.pool,
                self._to_local_t(phase.snap_nodes[mine]),
                torch.from_numpy(phase.snap_slots[mine].astype(np.int64)).to(self.device),
                src_off=self._snap_off,
            )

        # B: deliveries (merge + update [+ reply snapshot]) and PULL snapshots.
        # Comm/compute overlap: receivers whose deliveries are all locally
        # sourced launch while the cross-GPU slot exchange is in flight;
        # receivers touching remote slots launch after it completes.
        rmine = self._is_mine(phase.recv_nodes)
        plan = self._plan_exchange(
            phase.recv_nodes, phase.recv_ptr, phase.del_slots, phase.del_owners
        )
        if not plan:
            self._exchange(plan)
            if rmine.any():
                self._deliver_group(phase, np.where(rmine)[0])
        else:
            src_rank = self._rank_of(phase.del_owners)
            has_remote = np.zeros(len(phase.recv_nodes), dtype=bool)
            for i in range(len(phase.recv_nodes)):
                seg = src_rank[phase.recv_ptr[i] : phase.recv_ptr[i + 1]]
                has_remote[i] = (seg != self.rank).any()
            pending = self._exchange_start(plan)
            self._deliver_group(phase, np.where(rmine & ~has_remote)[0])
            self._exchange_finish(pending)
            self._deliver_group(phase, np.where(rmine & has_remote)[0])
        pmine = self._is_mine(phase.pull_snap_nodes)
        if pmine.any():
            self.backend.snapshot(
                self.state,
                self.pool,
                self._to_local_t(phase.pull_snap_nodes[pmine]),
                torch.from_numpy(phase.pull_snap_slots[pmine].astype(np.int64)).to(
                    self.device
                ),
                src_off=self._snap_off,
            )

        # C: same-tick replies
        if phase.rep_del_slots is not None and len(phase.rep_del_slots):
            self._exchange(
                self._plan_exchange(
                    phase.rep_recv_nodes,
                    phase.rep_recv_ptr,
                    phase.rep_del_slots,
                    phase.rep_del_owners,
                )
            )
            cmine = self._is_mine(phase.rep_recv_nodes)
            if cmine.any():
                sel, new_ptr = _csr_gather(
                    phase.rep_recv_ptr, np.where(cmine)[0]
                )
                no_reply = torch.full((len(sel),), -1, dtype=torch.int64)
                rpids = None
                if phase.rep_pids is not None and len(phase.rep_pids):
                    rpids = torch.from_numpy(phase.rep_pids[sel].astype(np.int64))
                self.backend.deliver(
                    self.state,
                    self.pool,
                    self.data,
                    self.spec,
                    self._to_local_t(phase.rep_recv_nodes[cmine]),
                    torch.from_numpy(new_ptr),
                    torch.from_numpy(phase.rep_del_slots[sel].astype(np.int64)),
                    no_reply,
                    del_pids=rpids,
                )

    def _evaluate(self, sched: RoundSchedule, t: int) -> None:
        """Round-end evaluation sweep (gossipy/simul.py:432-450): local test
        shards when present, then the global eval set; metric dicts gathered
        to rank 0."""
        if sched.eval_nodes is not None:
            nodes = np.unique(sched.eval_nodes)
        else:
            nodes = np.arange(self.cfg.n_nodes)
        mine = nodes[self._is_mine(nodes)]
        local_ids = torch.from_numpy((mine - self.node_lo).astype(np.int64))

        if self.spec.family == "kmeans":
            # NMI of each node's clustering of the global eval set
            # (gossipy/model/handler.py:632-636; ClusteringDataHandler's
            # eval set IS the train set, gossipy/data/handler.py:156-161)
            results = []
            if self.data.gx is not None and len(mine):
                from sklearn.metrics import normalized_mutual_info_score as nmi

                assign = self.backend.kmeans_assign(
                    self.state, self.spec, local_ids, self.data.gx
                ).cpu().numpy()
                y_true = self.data.gy.cpu().numpy()
                results = [
                    {"nmi": float(nmi(y_true, assign[r]))}
                    for r in range(assign.shape[0])
                ]
            if self.world > 1:
                gathered = [None] * self.world
                dist.all_gather_object(gathered, results)
                results = [d for part in gathered for d in part]
            if self.rank == 0 and results:
                self.notify_evaluation(t, False, results)
            return

        if self.spec.family == "mf":
            # RecSys eval: per-user RMSE on the held-out ratings; there is
            # no global test set (gossipy/data/__init__.py:550-555)
            results_local = self.backend.mf_rmse(
                self.state, self.data, self.spec, local_ids
            )
            if self.world > 1:
                gathered: List[List[dict]] = [None] * self.world
                dist.all_gather_object(gathered, results_local)
                results_local = [d for part in gathered for d in part]
            if self.rank == 0 and results_local:
                self.notify_evaluation(t, True, results_local)
            return

        # pipelined eval (single rank, global eval set only): enqueue the
        # K13 kernel + async D2H now, collect the metrics one round later —
        # the fetch overlaps the next round's compute instead of syncing
        # the stream mid-loop. Values are identical; report entries keep
        # their own round timestamps and stay chronological.
        if (
            self.world == 1
            and self.data.tx is None
            and self.data.gx is not None
            and os.environ.get("GOSSIPY_NO_EVAL_PIPE") != "1"
            and hasattr(self.backend, "eval_metrics_launch")
            and self.spec.family in ("logreg", "pegasos", "adaline",
                                     "mlp", "torchmod")
        ):
            self._drain_eval(force=False)
            if len(mine):
                h = self.backend.eval_metrics_launch(
                    self.state, self.spec, local_ids, self.data.gx, self.data.gy
                )
                if h is not None:
                    if not hasattr(self, "_eval_pending") or self._eval_pending is None:
                        from collections import deque

                        self._eval_pending = deque()
                    self._eval_pending.append((t, h))
                    return
            else:
                return

        results_global: List[dict] = []
        if self.data.gx is not None and len(mine):
            fast_eval = getattr(self.backend, "eval_metrics_fast", None)
            got = None
            if fast_eval is not None:
                got = fast_eval(
                    self.state, self.spec, local_ids, self.data.gx, self.data.gy
                )
            if got is not None:
                results_global = got
            else:
                scores = self.backend.scores(
                    self.state, self.spec, local_ids, self.data.gx
                )
                gy = self.data.gy
                if self.spec.family in ("pegasos", "adaline"):
                    results_global = binary_margin_metrics(scores[:, :, 0], gy)
                else:
                    results_global = classification_metrics_shared(scores, gy)

        results_local: List[dict] = []
        if self.data.tx is not None and len(mine):
            fast_local = getattr(self.backend, "eval_local_fast", None)
            got_local = None
            if fast_local is not None:
                # one launch: block r scores node r on its own shard
                got_local = fast_local(
                    self.state, self.spec, local_ids,
                    self.data.tx, self.data.ty, self.data.tcounts,
                )
            if got_local is not None:
                results_local = got_local
        if self.data.tx is not None and len(mine) and not results_local:
            # per-node test shards: evaluate each node on its own shard
            for li in local_ids.tolist():
                c = int(self.data.tcounts[li])
                if c == 0:
                    continue
                X = self.data.tx[li, :c]
                yv = self.data.ty[li, :c]
                sc = self.backend.scores(
                    self.state, self.spec, torch.tensor([li]), X
                )
                if self.spec.family in ("pegasos", "adaline"):
                    results_local.extend(binary_margin_metrics(sc[:, :, 0], yv))
                else:
                    results_local.extend(classification_metrics_shared(sc, yv))

        if self.world > 1:
            gathered_g: List[List[dict]] = [None] * self.world
            gathered_l: List[List[dict]] = [None] * self.world
            dist.all_gather_object(gathered_g, results_global)
            dist.all_gather_object(gathered_l, results_local)
            results_global = [d for part in gathered_g for d in part]
            results_local = [d for part in gathered_l for d in part]

        if self.rank == 0:
            if results_local:
                self.notify_evaluation(t, True, results_local)
            if results_global:
                self.notify_evaluation(t, False, results_global)

    def _drain_eval(self, force: bool = True) -> None:
        """Collect in-flight pipelined evaluations. ``force=False`` (the
        per-round call) collects only those whose D2H event has already
        signalled — the host never blocks on the GPU mid-loop; boundaries
        (end of start(), checkpoint save) force-collect everything. At
        most two evaluations ride in flight (the pinned staging is
        double-buffered), so the per-round call force-collects the oldest
        when a third would launch."""
        pend = getattr(self, "_eval_pending", None)
        if not pend:
            return
        while pend:
            t_prev, h = pend[0]
            if not force and len(pend) < 2 and not h[1].query():
                break
            pend.popleft()
            results = self.backend.eval_metrics_collect(h)
            if results:
                self.notify_evaluation(t_prev, False, results)

    def _fast_path_ok(self) -> bool:
        """Single-GPU fast path: native scheduler + HIP round executor —
        one python call per round instead of per-tick dispatch."""
        from .schedule import NativeSchedulerAdapter

        return (
            self.world == 1
            and isinstance(self.scheduler, NativeSchedulerAdapter)
            and self._flat_exec_ok()
        )

    def _flat_exec_ok(self) -> bool:
        """Whether this spec can run through the whole-round C++ executor
        (python-scheduled rounds are flattened into the same format)."""
        return (
            self.world == 1
            and getattr(self.backend, "ext", None) is not None
            and self.spec.family in ("logreg", "pegasos", "adaline", "mlp", "mf")
            and (self.spec.family == "logreg" or getattr(self.spec, "n_parts", 0) == 0)
            and not getattr(self.spec, "pass_through", False)
            and (self.spec.family != "mf" or self.spec.mode == CreateModelMode.MERGE_UPDATE)
        )

    @staticmethod
    def _flatten_phases(ticks) -> dict:
        """Concatenate a python-scheduled round's TickPhases into the flat
        per-launch-group arrays the C++ round executors consume (each phase
        becomes one group; CSR pointers become globally cumulative)."""
        f = {
            "snap_nodes": [], "snap_slots": [], "snap_tptr": [0],
            "recv_nodes": [], "recv_nptr": [0], "recv_tptr": [0],
            "del_slots": [], "reply_slots": [], "del_pids": [],
            "pull_nodes": [], "pull_slots": [], "pull_tptr": [0],
            "rep_nodes": [], "rep_nptr": [0], "rep_tptr": [0],
            "rep_slots": [], "rep_pids": [],
            "del_owners": [], "rep_owners": [],
        }
        d_off = 0
        e_off = 0
        for ph in ticks:
            f["snap_nodes"].append(ph.snap_nodes)
            f["snap_slots"].append(ph.snap_slots)
            f["snap_tptr"].append(f["snap_tptr"][-1] + len(ph.snap_nodes))
            f["recv_nodes"].append(ph.recv_nodes)
            if len(ph.recv_nodes):
                f["recv_nptr"].extend(
                    (np.asarray(ph.recv_ptr[1:]) + d_off).tolist()
                )
            d_off += len(ph.del_slots)
            f["recv_tptr"].append(f["recv_tptr"][-1] + len(ph.recv_nodes))
            f["del_slots"].append(ph.del_slots)
            f["del_owners"].append(ph.del_owners)
            f["reply_slots"].append(ph.reply_slots)
            if ph.del_pids is not None and len(ph.del_pids):
                f["del_pids"].append(ph.del_pids)
            f["pull_nodes"].append(ph.pull_snap_nodes)
            f["pull_slots"].append(ph.pull_snap_slots)
            f["pull_tptr"].append(f["pull_tptr"][-1] + len(ph.pull_snap_nodes))
            rep_n = ph.rep_recv_nodes if ph.rep_recv_nodes is not None else np.zeros(0, np.int32)
            f["rep_nodes"].append(rep_n)
            if len(rep_n):
                f["rep_nptr"].extend(
                    (np.asarray(ph.rep_recv_ptr[1:]) + e_off).tolist()
                )
                e_off += len(ph.rep_del_slots)
                f["rep_slots"].append(ph.rep_del_slots)
                f["rep_owners"].append(ph.rep_del_owners)
                if ph.rep_pids is not None and len(ph.rep_pids):
                    f["rep_pids"].append(ph.rep_pids)
            f["rep_tptr"].append(f["rep_tptr"][-1] + len(rep_n))
        out = {}
        for k, v in f.items():
            if k.endswith("tptr") or k in ("recv_nptr", "rep_nptr"):
                out[k] = np.asarray(v, dtype=np.int32)
            else:
                out[k] = (
                    np.concatenate(v).astype(np.int32)
                    if v else np.zeros(0, dtype=np.int32)
                )
        return out

    def _maybe_merge(
        self, f: dict, pack: bool = True, force_py_pack: bool = False
    ) -> dict:
        """Apply launch-group packing/merging unless disabled
        (``GOSSIPY_NO_MERGE=1`` turns everything off, ``GOSSIPY_NO_PACK=1``
        falls back to tick-level merging). Packed schedules carry the
        owner arrays, so the multi-rank path consumes them too — one
        batched slot exchange per packed group instead of per tick.
        Tracks a cumulative (ticks, groups) counter for the perf probes."""
        if os.environ.get("GOSSIPY_NO_MERGE") == "1":
            return f
        pack = (
            pack
            and os.environ.get("GOSSIPY_NO_PACK") != "1"
            and os.environ.get("GOSSIPY_COOP") != "1"
            and not os.environ.get("GOSSIPY_SB_MAX")
        )
        packed = f.get("packed") if pack else None
        if packed is None and "recv_nodes" not in f:
            # lean flat: only the packed schedule was materialized — any
            # knob that reached here is overridden (loud beats wrong)
            packed = f["packed"]
        if packed is None and pack and (
            force_py_pack or os.environ.get("GOSSIPY_PACK") == "1"
        ):
            # python reference packer (A/B + flats without the C++ packer;
            # forced for the torchmod group-replay path, whose vmap
            # batches live on group width)
            packed = self._pack_flat(f)
        if packed is not None:
            t = getattr(self, "merge_stats", (0, 0))
            self.merge_stats = (
                t[0] + len(f["snap_tptr"]) - 1,
                t[1] + len(packed["snap_tptr"]) - 1,
            )
            return packed
        mb = f.get("merge_bounds")
        if mb is not None and len(mb) - 1 < len(f["snap_tptr"]) - 1:
            # native scheduler pre-computed the group boundaries in C++
            out = dict(f)
            for k in ("snap_tptr", "recv_tptr", "pull_tptr", "rep_tptr"):
                out[k] = f[k][mb]
        elif mb is not None:
            out = f
        else:
            out = self._merge_flat_groups(f)
        t = getattr(self, "merge_stats", (0, 0))
        self.merge_stats = (
            t[0] + len(f["snap_tptr"]) - 1,
            t[1] + len(out["snap_tptr"]) - 1,
        )
        return out

    @staticmethod
    def _merge_flat_groups(f: dict) -> dict:
        """Merge conflict-free adjacent ticks into single launch groups
        (SURVEY.md §7 hard-part 2: "batch independent tick-groups").

        Two ticks can share one snapshot/deliver/pull/reply launch
        quadruple iff no ordering hazard exists between them:

        * node hazard — a later tick may not touch (snapshot, receive)
          any node the group already MUTATES (deliver/reply receivers),
          and may not mutate a node the group touches;
        * slot hazards — a later tick's written slots (snap/pull/reply)
          must not alias anything the group reads or writes (recycled slot
          ids!), and its read slots must not alias slots the group writes
          in the deliver/pull launches (reply and pull writes land after /
          inside launches a merged read could precede).

        Only the four tick-pointer arrays change; the event arrays and the
        global CSRs are untouched, so every round executor consumes the
        merged schedule unmodified.
        """
        delta = len(f["snap_tptr"]) - 1
        if delta <= 1:
            return f
        st, rt = np.asarray(f["snap_tptr"]), np.asarray(f["recv_tptr"])
        pt, qt = np.asarray(f["pull_tptr"]), np.asarray(f["rep_tptr"])
        nptr, rep_nptr = np.asarray(f["recv_nptr"]), np.asarray(f["rep_nptr"])
        recv_nodes = np.asarray(f["recv_nodes"])
        rep_nodes = np.asarray(f["rep_nodes"])
        snap_nodes = np.asarray(f["snap_nodes"])
        pull_nodes = np.asarray(f["pull_nodes"])
        snap_slots = np.asarray(f["snap_slots"])
        pull_slots = np.asarray(f["pull_slots"])
        del_slots = np.asarray(f["del_slots"])
        rep_slots = np.asarray(f["rep_slots"])
        reply_slots = np.asarray(f["reply_slots"])

        def tick_sets(t):
            r0, r1 = rt[t], rt[t + 1]
            q0, q1 = qt[t], qt[t + 1]
            d0, d1 = nptr[r0], nptr[r1]
            e0, e1 = rep_nptr[q0], rep_nptr[q1]
            recv = set(recv_nodes[r0:r1])
            mut = recv | set(rep_nodes[q0:q1])
            pulls = set(pull_nodes[pt[t] : pt[t + 1]])
            touched = mut | pulls | set(snap_nodes[st[t] : st[t + 1]])
            dreads = set(del_slots[d0:d1])
            reads = dreads | set(rep_slots[e0:e1])
            rw = set(x for x in reply_slots[d0:d1] if x >= 0)
            pw = set(pull_slots[pt[t] : pt[t + 1]])
            writes = set(snap_slots[st[t] : st[t + 1]]) | pw | rw
            # reply/pull writes land inside or after the deliver launch, so a
            # merged tick's *deliver* reads must not alias them (its reply
            # reads run in the final launch and are ordered fine)
            late_writes = rw | pw
            return recv, mut, pulls, touched, dreads, reads, writes, late_writes

        bounds = [0]
        (g_recv, g_mut, g_pulls, g_touched,
         g_dreads, g_reads, g_writes, g_late) = tick_sets(0)
        for t in range(1, delta):
            recv, mut, pulls, touched, dreads, reads, writes, late = tick_sets(t)
            ok = (
                # later tick may not read/mutate a node the group mutates
                not (touched & g_mut)
                # deliver launch precedes the pull-snapshot launch: a merged
                # delivery may not hit a node the group pull-snapshots
                and not (recv & g_pulls)
                # recycled slot ids: new writes may not alias live reads/writes
                and not (writes & (g_reads | g_writes))
                # merged deliver reads vs the group's late (reply/pull) writes
                and not (dreads & g_late)
            )
            if ok:
                g_recv |= recv
                g_mut |= mut
                g_pulls |= pulls
                g_touched |= touched
                g_dreads |= dreads
                g_reads |= reads
                g_writes |= writes
                g_late |= late
            else:
                bounds.append(t)
                (g_recv, g_mut, g_pulls, g_touched,
                 g_dreads, g_reads, g_writes, g_late) = (
                    recv, mut, pulls, touched, dreads, reads, writes, late,
                )
        bounds.append(delta)
        if len(bounds) - 1 == delta:
            return f  # nothing merged
        sel = np.asarray(bounds)
        out = dict(f)
        out["snap_tptr"] = st[sel]
        out["recv_tptr"] = rt[sel]
        out["pull_tptr"] = pt[sel]
        out["rep_tptr"] = qt[sel]
        return out

    @staticmethod
    def _pack_flat(f: dict) -> dict:
        """Entry-level launch packing with per-node deferral.

        Every event is placed individually into the open launch group:

        * deliveries to one receiver from MANY ticks coalesce into ONE CSR
          row (the kernel processes a row's deliveries in order, so tick
          order is preserved inside the row);
        * a snapshot of a node that already received in the group embeds
          as the per-delivery reply write (``rslots[j]`` — written right
          after delivery ``j``, exactly the state the reference snapshots
          at the send tick);
        * a delivery whose source slot is produced in the group's own
          deliver launch moves to the second deliver launch (the
          ``rep_*`` arrays with ``rep_reply_slots`` for their embedded
          writes);
        * an event that would need a THIRD level (burst chains: consume a
          second-launch product) **defers its node**: the node's remaining
          events queue up (in global schedule order — causality) and
          replay after the group closes, so one deep chain no longer
          fragments the whole round. Groups close only on recycled-slot
          aliasing, at drain passes, and at round end.

        The flagship's 100-tick round packs into ~4 groups; PUSH_PULL
        rounds into ~7; tokenized burst rounds into ~15. Output uses the
        executors' existing array format (pull arrays come back empty —
        standalone pull-snapshots fold into the snapshot launch, where
        their content is identical)."""
        delta = len(f["snap_tptr"]) - 1
        st, rt, pt, qt = f["snap_tptr"], f["recv_tptr"], f["pull_tptr"], f["rep_tptr"]
        nptr, rep_nptr = f["recv_nptr"], f["rep_nptr"]
        has_pid = len(f.get("del_pids", ())) > 0 or len(f.get("rep_pids", ())) > 0
        has_own = len(f.get("del_owners", ())) > 0 or len(f.get("rep_owners", ())) > 0

        out = {k: [] for k in ("snap_nodes","snap_slots","recv_nodes","del_slots",
            "reply_slots","del_pids","del_owners","rep_nodes","rep_slots",
            "rep_reply_slots","rep_pids","rep_owners")}
        out.update({"snap_tptr":[0],"recv_tptr":[0],"recv_nptr":[0],
                    "rep_tptr":[0],"rep_nptr":[0],"pull_tptr":[0]})

        g_snap = []; rows2 = {}; rows3 = {}; l2 = []; l3 = []; latest = {}
        written = {}; touched = set()
        deferred_nodes = set()
        deferred_list = []   # (node, event) in ORIGINAL schedule order — replay
                             # must preserve cross-node causality (a consumer's
                             # producer always precedes it in schedule order)

        in_replay = [False]

        def close():
            if in_replay[0]:
                flush()
                return
            flush()
            while deferred_list:
                in_replay[0] = True
                dl = deferred_list[:]; deferred_list.clear(); deferred_nodes.clear()
                for n, ev in dl:
                    if ev[0] == "s": place_snap(n, ev[1])
                    else: place_delivery(n, ev[1], ev[2], ev[3], ev[4])
                in_replay[0] = False
                if deferred_list:
                    flush()  # each drain pass lands in a fresh group

        def flush():
            out["snap_nodes"].extend(n for n,_ in g_snap)
            out["snap_slots"].extend(s for _,s in g_snap)
            out["snap_tptr"].append(len(out["snap_nodes"]))
            for node, evs in l2:
                out["recv_nodes"].append(node)
                for s,r_,p,o in evs:
                    out["del_slots"].append(s); out["reply_slots"].append(r_)
                    if has_pid: out["del_pids"].append(p)
                    if has_own: out["del_owners"].append(o)
                out["recv_nptr"].append(len(out["del_slots"]))
            out["recv_tptr"].append(len(out["recv_nodes"]))
            for node, evs in l3:
                out["rep_nodes"].append(node)
                for s,r_,p,o in evs:
                    out["rep_slots"].append(s); out["rep_reply_slots"].append(r_)
                    if has_pid: out["rep_pids"].append(p)
                    if has_own: out["rep_owners"].append(o)
                out["rep_nptr"].append(len(out["rep_slots"]))
            out["rep_tptr"].append(len(out["rep_nodes"]))
            out["pull_tptr"].append(0)
            g_snap.clear(); rows2.clear(); rows3.clear(); l2.clear(); l3.clear()
            latest.clear(); written.clear(); touched.clear()

        def defer(node, ev):
            deferred_nodes.add(node)
            deferred_list.append((node, ev))
            if ev[0] == "s":
                written[ev[1]] = 9; touched.add(ev[1])
            else:
                touched.add(ev[1])
                if ev[2] >= 0: written[ev[2]] = 9; touched.add(ev[2])

        def place_snap(node, slot):
            if node in deferred_nodes: defer(node, ("s", slot)); return
            if slot in touched:
                close()
                if node in deferred_nodes:  # replay re-deferred this node
                    defer(node, ("s", slot)); return
            pos = latest.get(node)
            if pos is None:
                g_snap.append((node, slot)); written[slot] = 1
            else:
                lst = l2 if pos[0]==2 else l3
                evs = lst[pos[1]][1]
                if evs[-1][1] >= 0:
                    defer(node, ("s", slot)); return
                evs[-1] = (evs[-1][0], slot, evs[-1][2], evs[-1][3])
                written[slot] = 2 if pos[0]==2 else 3
            touched.add(slot)

        def place_delivery(node, slot, reply, pid, own):
            if node in deferred_nodes: defer(node, ("d", slot, reply, pid, own)); return
            lvl = written.get(slot, 0)
            if lvl >= 3 or (reply >= 0 and reply in touched):
                defer(node, ("d", slot, reply, pid, own)); return
            pos = latest.get(node)
            if pos is not None and pos[0] == 3:
                l3[pos[1]][1].append((slot, reply, pid, own)); wl = 3
            elif lvl >= 2:
                ri = rows3.get(node)
                if ri is None:
                    rows3[node] = ri = len(l3); l3.append([node, []])
                l3[ri][1].append((slot, reply, pid, own)); latest[node] = (3, ri); wl = 3
            else:
                ri = rows2.get(node)
                if ri is None:
                    rows2[node] = ri = len(l2); l2.append([node, []]); latest[node] = (2, ri)
                l2[ri][1].append((slot, reply, pid, own)); wl = 2
            touched.add(slot)
            if reply >= 0: written[reply] = wl; touched.add(reply)

        sn, ss = f["snap_nodes"], f["snap_slots"]
        rn = f["recv_nodes"]; ds, rs = f["del_slots"], f["reply_slots"]
        dp = f.get("del_pids"); do = f.get("del_owners")
        pn, ps = f["pull_nodes"], f["pull_slots"]
        qn, qs = f["rep_nodes"], f["rep_slots"]
        qp = f.get("rep_pids"); qo = f.get("rep_owners")
        for t in range(delta):
            for i in range(st[t], st[t+1]): place_snap(int(sn[i]), int(ss[i]))
            for r in range(rt[t], rt[t+1]):
                x = int(rn[r])
                for d in range(nptr[r], nptr[r+1]):
                    place_delivery(x, int(ds[d]), int(rs[d]),
                                   int(dp[d]) if has_pid and len(dp) else -1,
                                   int(do[d]) if has_own and len(do) else -1)
            for i in range(pt[t], pt[t+1]): place_snap(int(pn[i]), int(ps[i]))
            for r in range(qt[t], qt[t+1]):
                x = int(qn[r])
                for d in range(rep_nptr[r], rep_nptr[r+1]):
                    place_delivery(x, int(qs[d]), -1,
                                   int(qp[d]) if has_pid and len(qp) else -1,
                                   int(qo[d]) if has_own and len(qo) else -1)
        guard = 0
        while (g_snap or l2 or l3 or deferred_list) and guard < 10000:
            close(); guard += 1
        res = {k: np.asarray(v, dtype=np.int32) for k, v in out.items()}
        res["pull_nodes"] = np.zeros(0, np.int32); res["pull_slots"] = np.zeros(0, np.int32)
        return res

    def _run_round_fast(self, f: dict) -> None:
        """Upload the round's flat event arrays in one H2D copy and hand
        the whole round to the C++ executor (ops/hip round executors)."""
        dev = self.device
        dev_names = (
            "snap_nodes",
            "snap_slots",
            "recv_nodes",
            "recv_nptr",
            "del_slots",
            "reply_slots",
            "pull_nodes",
            "pull_slots",
            "rep_nodes",
            "rep_nptr",
            "rep_slots",
            "del_pids",
            "rep_pids",
            "rep_reply_slots",
        )
        _empty = np.zeros(0, np.int32)
        parts = [
            np.ascontiguousarray(f.get(n, _empty), dtype=np.int32)
            for n in dev_names
        ]
        lens = [len(p) for p in parts]
        host = np.concatenate(parts) if sum(lens) else np.zeros(1, np.int32)
        dbuf = torch.from_numpy(host).to(dev, non_blocking=True)
        views = {}
        off = 0
        for n, l in zip(dev_names, lens):
            views[n] = dbuf[off : off + l]
            off += l
        tp = {
            n: torch.from_numpy(np.ascontiguousarray(f[n], dtype=np.int32))
            for n in ("snap_tptr", "recv_tptr", "pull_tptr", "rep_tptr")
        }
        ext = self.backend.ext
        common = (
            self.state.params,
            self.state.ages,
            self.pool.slots,
            self.pool.slot_ages,
            views["snap_nodes"],
            views["snap_slots"],
            tp["snap_tptr"],
            views["recv_nodes"],
            views["recv_nptr"],
            tp["recv_tptr"],
            views["del_slots"],
            views["reply_slots"],
            views["pull_nodes"],
            views["pull_slots"],
            tp["pull_tptr"],
            views["rep_nodes"],
            views["rep_nptr"],
            tp["rep_tptr"],
            views["rep_slots"],
            self.data.x,
            self.data.y,
            self.data.counts,
        )
        from .backend import _MODE_ID

        spec = self.spec
        if (
            spec.family == "logreg"
            and getattr(spec, "n_parts", 0) == 0
            and getattr(spec, "sample_size", 0) == 0
            and not getattr(spec, "pass_through", False)
            and getattr(self, "_coop_enabled", True)
            and os.environ.get("GOSSIPY_COOP") == "1"
        ):
            # single-launch cooperative round (plain logreg) — measured
            # SLOWER than the stream executor on MI355X (1.68 vs 1.41
            # ms/round at the flagship config: ~200 grid.sync barriers cost
            # more than ~200 stream launch gaps), so opt-in only
            # (GOSSIPY_COOP=1); kept as the measured alternative
            per_tick = []
            for name in ("snap_tptr", "recv_tptr", "pull_tptr", "rep_tptr"):
                dif = np.diff(f[name])
                if len(dif):
                    per_tick.append(int(dif.max()))
            max_batch = max(per_tick) if per_tick else 1
            tpd = {
                n: torch.from_numpy(
                    np.ascontiguousarray(f[n], dtype=np.int32)
                ).to(dev)
                for n in ("snap_tptr", "recv_tptr", "pull_tptr", "rep_tptr")
            }
            try:
                ext.run_round_coop_logreg(
                    self.state.params,
                    self.state.ages,
                    self.pool.slots,
                    self.pool.slot_ages,
                    views["snap_nodes"],
                    views["snap_slots"],
                    tpd["snap_tptr"],
                    views["recv_nodes"],
                    views["recv_nptr"],
                    tpd["recv_tptr"],
                    views["del_slots"],
                    views["reply_slots"],
                    views["pull_nodes"],
                    views["pull_slots"],
                    tpd["pull_tptr"],
                    views["rep_nodes"],
                    views["rep_nptr"],
                    tpd["rep_tptr"],
                    views["rep_slots"],
                    self.data.x,
                    self.data.y,
                    self.data.counts,
                    spec.d_in,
                    spec.n_classes,
                    spec.lr,
                    spec.weight_decay,
                    max(1, spec.local_epochs),
                    spec.batch_size,
                    _MODE_ID[spec.mode],
                    max_batch,
                )
                return
            except RuntimeError as e:
                LOG.warning("cooperative round launch unavailable (%s); "
                            "falling back to the stream executor", e)
                self._coop_enabled = False

        if getattr(spec, "sample_size", 0) > 0:
            c = list(common)
            c.insert(12, views["del_pids"])   # sample seeds ride the pid slot
            c.insert(20, views["rep_pids"])
            ext.run_round_logreg_samp(
                *c,
                spec.samp_count(),
                spec.d_in,
                spec.n_classes,
                spec.lr,
                spec.weight_decay,
                max(1, spec.local_epochs),
                spec.batch_size,
                _MODE_ID[spec.mode],
                views["rep_reply_slots"],
            )
        elif spec.family == "mf":
            ext.run_round_mf(
                *common,
                spec.k,
                spec.n_items,
                spec.reg,
                spec.lr,
                views["rep_reply_slots"],
            )
        elif getattr(spec, "n_parts", 0) > 0:
            perm, pptr, apart = self.backend._part_dev(spec, dev)
            # run_round_logreg_part takes del_pids after reply_slots and
            # rep_pids after rep_slots
            c = list(common)
            c.insert(12, views["del_pids"])   # after reply_slots
            c.insert(20, views["rep_pids"])   # after rep_slots
            ext.run_round_logreg_part(
                *c,
                perm,
                pptr,
                apart,
                spec.n_parts,
                spec.d_in,
                spec.n_classes,
                spec.lr,
                spec.weight_decay,
                max(1, spec.local_epochs),
                spec.batch_size,
                _MODE_ID[spec.mode],
                views["rep_reply_slots"],
            )
        elif spec.family == "logreg":
            ext.run_round_logreg(
                *common,
                spec.d_in,
                spec.n_classes,
                spec.lr,
                spec.weight_decay,
                max(1, spec.local_epochs),
                spec.batch_size,
                _MODE_ID[spec.mode],
                views["rep_reply_slots"],
            )
        elif spec.family == "mlp":
            layout = torch.tensor(
                [x for tup in spec.layer_offsets() for x in tup],
                dtype=torch.int32,
                device=dev,
            )
            ext.run_round_mlp(
                *common,
                layout,
                len(spec.layer_offsets()),
                spec.lr,
                spec.weight_decay,
                max(1, spec.local_epochs),
                spec.batch_size,
                _MODE_ID[spec.mode],
                views["rep_reply_slots"],
            )
        else:
            ext.run_round_linear(
                *common,
                spec.d_in,
                spec.lam if spec.family == "pegasos" else spec.lr,
                1 if spec.family == "pegasos" else 0,
                _MODE_ID[spec.mode],
                views["rep_reply_slots"],
            )

    def start(self, n_rounds: int = 100) -> None:
        """Run ``n_rounds`` rounds."""
        assert self.initialized, "call init_nodes() first"
        fast = self._fast_path_ok()
        if (
            fast
            and hasattr(self.scheduler, "set_lean")
            and os.environ.get("GOSSIPY_NO_MERGE") != "1"
            and os.environ.get("GOSSIPY_NO_PACK") != "1"
            and os.environ.get("GOSSIPY_COOP") != "1"
            and not os.environ.get("GOSSIPY_SB_MAX")
        ):
            # fast path consumes only the packed schedule — skip the full
            # per-event array materialization and the multi-rank merge scan.
            # Any knob that routes around the packed path needs the full
            # arrays, so lean stays off there.
            self.scheduler.set_lean(True)
        # measured SLOWER on the flagship (0.78 vs 0.58 ms/round: the
        # submit/result handoff plus GIL contention with the executor's
        # python prolog outweighs the overlap) — kept as an opt-in A/B
        if (
            fast
            and self._exec_pool is None
            and os.environ.get("GOSSIPY_THREAD") == "1"
        ):
            from concurrent.futures import ThreadPoolExecutor

            self._exec_pool = ThreadPoolExecutor(
                1, thread_name_prefix="gossipy-exec"
            )
        for _ in range(n_rounds):
            r = self.rounds_done
            if fast:
                # the scheduler is stateful (rounds must be generated exactly
                # once, in order), so the prefetch lives on self and is keyed
                # by round index — it survives across start() calls
                pre = getattr(self, "_prefetched", None)
                if pre is not None and pre[0] == r:
                    _, sched, flat = pre
                else:
                    sched = self.scheduler.next_round_flat(r)
                    flat = self._maybe_merge(self.scheduler.last_flat)
                self._prefetched = None
                self.pool.ensure(sched.n_slots)
                # the C++ round executors release the GIL, so a worker
                # thread can drive round r's launch queue while this thread
                # derives round r+1's schedule + merge. The join before
                # _evaluate keeps the stream's enqueue order deterministic.
                fut = None
                if self._exec_pool is not None:
                    fut = self._exec_pool.submit(self._run_round_fast, flat)
                else:
                    self._run_round_fast(flat)
                s2 = self.scheduler.next_round_flat(r + 1)
                self._prefetched = (
                    r + 1, s2, self._maybe_merge(self.scheduler.last_flat)
                )
                if fut is not None:
                    fut.result()
            else:
                sched = self.scheduler.next_round(r)
                self.pool.ensure(sched.n_slots)
                if (
                    self.world > 1
                    and self._flat_schedulable
                    and os.environ.get("GOSSIPY_NO_MULTIFAST") != "1"
                ):
                    # multi-rank: one vectorized per-round prep + upload,
                    # per-group loop of launches and RCCL exchanges only.
                    # Launch-group merging applies here too — the hazard scan
                    # runs on global node/slot ids, so the merged bounds are
                    # identical on every rank, and each group moves all its
                    # cross-rank slots in one batched exchange.
                    flat = getattr(self.scheduler, "last_flat", None)
                    if flat is None:
                        flat = self._flatten_phases(sched.ticks)
                    self._run_round_multi(self._maybe_merge(flat))
                elif (
                    self._flat_schedulable
                    and self._flat_exec_ok()
                    and sched.ticks
                ):
                    # python-scheduled round through the C++ round executor
                    # (tokenized / cache-neigh / python-scheduler fallback)
                    self._run_round_fast(
                        self._maybe_merge(self._flatten_phases(sched.ticks))
                    )
                elif (
                    self.world == 1
                    and self._flat_schedulable
                    and self.spec.family == "torchmod"
                    and sched.ticks
                ):
                    # torchmod has no C++ executor, but its batched python
                    # backend wants LARGE receiver groups (vmap SGD over
                    # all of a group's receivers at once) — replay packed
                    # launch groups instead of 1-2-receiver ticks
                    self._run_round_groups(
                        self._maybe_merge(
                            self._flatten_phases(sched.ticks),
                            force_py_pack=True,
                        )
                    )
                else:
                    for phase in sched.ticks:
                        self._run_tick(phase)
            if self.rank == 0:
                # report accounting comes from the schedule (host-side)
                self.notify_message_counts(sched)
            self._evaluate(sched, (r + 1) * self.cfg.delta - 1)
            self.rounds_done += 1
            self.notify_timestep((r + 1) * self.cfg.delta - 1)
        self._drain_eval()
        self.notify_end()

    def notify_message_counts(self, sched: RoundSchedule) -> None:
        """Feed the schedule's message accounting to the observers through
        the standard update_message interface."""
        for er in self._receivers:
            if hasattr(er, "_sent_messages"):
                er._sent_messages += sched.sent_messages
                er._failed_messages += sched.failed_messages
                er._total_size += sched.total_size

    # -- state access for tests / checkpointing ------------------------------

    def local_params(self) -> torch.Tensor:
        return self.state.params

    def gather_ages(self) -> Optional[torch.Tensor]:
        if self.world == 1:
            return self.state.ages
        out = [torch.empty_like(self.state.ages) for _ in range(self.world)]
        dist.all_gather(out, self.state.ages.contiguous())
        return torch.cat(out, dim=0)

    # -- checkpointing --------------------------------------------------------

    def save(self, filename: str) -> None:
        """Engine-native checkpoint (per rank: ``filename`` should embed the
        rank for multi-GPU runs).

        The object layer keeps the reference's two-slot dill format
        (gossipy/simul.py:460-494); the engine materializes its arenas
        host-side instead of pickling Python object graphs. Scheduler
        carry-over state (in-flight messages, slot free list, token
        accounts) is NOT stored: the schedule is a pure function of
        ``(seed, config, round)``, so :meth:`load` replays rounds
        ``0..rounds_done`` on a fresh scheduler to restore it exactly —
        this is what makes the checkpoint valid for any scheduler
        implementation (python or native C++).
        """
        self._drain_eval()
        import dill

        blob = {
            "cfg": self.cfg,
            "spec": self.spec,
            "rounds_done": self.rounds_done,
            "params": self.state.params.cpu(),
            "ages": self.state.ages.cpu(),
            "pool_slots": self.pool.slots.cpu(),
            "pool_ages": self.pool.slot_ages.cpu(),
            "data": {
                k: (getattr(self.data, k).cpu() if getattr(self.data, k) is not None else None)
                for k in ("x", "y", "counts", "tx", "ty", "tcounts", "gx", "gy")
            },
            "kind": type(self).__name__,
            "extra": self._checkpoint_extra(),
        }
        with open(filename, "wb") as f:
            dill.dump(blob, f)

    def _ctor_fingerprint(self) -> tuple:
        """Identifies the subclass constructor arguments the checkpoint
        blob does NOT carry (token accounts, PENS hyperparameters,
        mixing matrices…). Stored on save and validated on load, so a
        resume with different arguments fails loudly instead of silently
        replaying a different schedule. Subclasses set ``self._ctor_desc``
        in ``__init__``."""
        return getattr(self, "_ctor_desc", ())

    def _checkpoint_extra(self) -> dict:
        return {"ctor": self._ctor_fingerprint()}

    def _restore_extra(self, extra: dict) -> None:
        pass

    def _replay_rounds(self, rounds_done: int, extra: dict) -> None:
        """Replay the deterministic schedule to restore carry-over state
        (in-flight messages, slot allocator, token accounts). Subclasses
        whose schedule depends on device-side results (PENS) override
        this to re-inject the saved results at the right round."""
        for r in range(rounds_done):
            if hasattr(self.scheduler, "next_round_flat"):
                self.scheduler.next_round_flat(r)
            else:
                self.scheduler.next_round(r)

    @classmethod
    def load(cls, filename: str, device: Optional[torch.device] = None, **kw):
        """Restore a checkpoint written by :meth:`save` (same world size).

        Subclass constructor arguments (``token_account``, PENS
        ``n_sampled``/``m_top``/``step1_rounds``, ``mixing``…) are NOT
        stored in the blob — pass them again via ``**kw`` exactly as in
        the original run. A fingerprint of those arguments IS stored, and
        a mismatch raises ``ValueError`` rather than silently replaying a
        different schedule."""
        import dill

        with open(filename, "rb") as f:
            blob = dill.load(f)
        if device is None:
            device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
        d = blob["data"]
        data = DataArena(
            *[d[k].to(device) if d[k] is not None else None for k in ("x", "y", "counts")],
            *[d[k].to(device) if d[k] is not None else None for k in ("tx", "ty", "tcounts")],
            *[d[k].to(device) if d[k] is not None else None for k in ("gx", "gy")],
        )
        sim = cls(blob["cfg"], blob["spec"], data, device=device, **kw)
        saved_ctor = blob["extra"].get("ctor", ())
        if tuple(saved_ctor) != tuple(sim._ctor_fingerprint()):
            raise ValueError(
                "checkpoint constructor mismatch: saved "
                f"{saved_ctor!r} vs supplied {sim._ctor_fingerprint()!r} — "
                "pass the same constructor arguments the saved run used"
            )
        sim.state.params.copy_(blob["params"].to(device))
        sim.state.ages.copy_(blob["ages"].to(device))
        sim.pool.ensure(blob["pool_slots"].shape[0])
        sim.pool.slots[: blob["pool_slots"].shape[0]].copy_(
            blob["pool_slots"].to(device)
        )
        sim.pool.slot_ages[: blob["pool_ages"].shape[0]].copy_(
            blob["pool_ages"].to(device)
        )
        sim._replay_rounds(blob["rounds_done"], blob["extra"])
        sim.rounds_done = blob["rounds_done"]
        sim._restore_extra(blob["extra"])
        sim.initialized = True
        return sim

    def gather_params(self) -> Optional[torch.Tensor]:
        """Full ``[n_nodes, D]`` parameter matrix on rank 0 (None elsewhere)."""
        if self.world == 1:
            return self.state.params
        out = [torch.empty_like(self.state.params) for _ in range(self.world)]
        dist.all_gather(out, self.state.params.contiguous())
        return torch.cat(out, dim=0)


class BatchedTokenizedGossipSimulator(BatchedGossipSimulator):
    """Flow-controlled batched simulator (TokenizedGossipSimulator,
    gossipy/simul.py:506-689, on the batched engine).

    Token accounts and the utility function live host-side in the
    :class:`~gossipy_amd.engine.schedule.TokenizedScheduler` (pure python —
    the schedule's burst waves are data-dependent on the account state, so
    the C++ scheduler is not used); the per-wave batched kernels are the
    same ones the plain simulator launches.

    ``utility_fun(receiver, sender, t) -> int`` is evaluated on node ids
    (the reference's flagship config uses a constant,
    main_hegedus_2021.py:57); utilities that inspect model state belong on
    the object layer.
    """

    def __init__(
        self,
        cfg: EngineConfig,
        spec,
        data: DataArena,
        token_account,
        utility_fun=None,
        device: Optional[torch.device] = None,
    ):
        super().__init__(cfg, spec, data, device=device)
        from .schedule import NativeTokenizedAdapter

        self._ctor_desc = (
            "tokenized",
            type(token_account).__name__,
            tuple(sorted(
                (k, v) for k, v in vars(token_account).items()
                if isinstance(v, (int, float, str, bool))
            )),
            utility_fun
            if (utility_fun is None or isinstance(utility_fun, int))
            else "callable:" + getattr(utility_fun, "__name__", "?"),
        )
        self.scheduler = None
        if (utility_fun is None or isinstance(utility_fun, int)) and self._flat_exec_ok():
            try:
                self.scheduler = NativeTokenizedAdapter(
                    cfg,
                    token_account,
                    1 if utility_fun is None else utility_fun,
                )
            except (ImportError, TypeError):
                self.scheduler = None
        if self.scheduler is None:
            self.scheduler = TokenizedScheduler(cfg, token_account, utility_fun)

    def _fast_path_ok(self) -> bool:
        from .schedule import NativeTokenizedAdapter

        # native tokenized schedule feeds the whole-round executor; the
        # python scheduler's waves go through the flatten path instead
        return isinstance(self.scheduler, NativeTokenizedAdapter) and self._flat_exec_ok()

    @property
    def accounts(self):
        from types import SimpleNamespace

        from .schedule import NativeTokenizedAdapter

        if isinstance(self.scheduler, NativeTokenizedAdapter):
            return [
                SimpleNamespace(n_tokens=b)
                for b in self.scheduler.token_balances()
            ]
        return self.scheduler.accounts


class BatchedAll2AllGossipSimulator(BatchedGossipSimulator):
    """Decentralized weighted averaging on the batched engine
    (All2AllGossipSimulator, gossipy/simul.py:720-852 + All2AllGossipNode,
    gossipy/node.py:789-869).

    Per tick: timed-out nodes first k-way weighted-merge the models
    accumulated from their peers (wmerge kernel + per-family update), then
    snapshot and broadcast the post-merge model to every peer. Deliveries
    are pure host-side bookkeeping (accumulation); no kernel runs at
    delivery time.

    ``mixing`` is a :class:`~gossipy_amd.core.MixingMatrix` (or any
    ``get(node) -> weights`` object / callable); ``None`` = UniformMixing.
    """

    def __init__(
        self,
        cfg: EngineConfig,
        spec,
        data: DataArena,
        mixing=None,
        device: Optional[torch.device] = None,
    ):
        super().__init__(cfg, spec, data, device=device)
        from ..core import CreateModelMode

        assert spec.mode == CreateModelMode.MERGE_UPDATE, (
            "all2all engine runs WeightedTMH MERGE_UPDATE semantics"
        )
        from .schedule import All2AllScheduler

        self._ctor_desc = (
            "all2all", type(mixing).__name__ if mixing is not None else None
        )
        self.scheduler = All2AllScheduler(cfg, mixing)
        self._flat_schedulable = False

    def _fast_path_ok(self) -> bool:
        return False

    def _run_tick(self, phase: TickPhase) -> None:
        # weighted merges come BEFORE this tick's snapshots: the reference
        # merges inside timed_out() and sends afterwards
        # (gossipy/simul.py:789-801)
        if phase.wm_nodes is not None and len(phase.wm_nodes):
            self._exchange(
                self._plan_exchange(
                    phase.wm_nodes, phase.wm_ptr, phase.wm_slots, phase.wm_owners
                )
            )
            mine = self._is_mine(phase.wm_nodes)
            if mine.any():
                sel, new_ptr = _csr_gather(phase.wm_ptr, np.where(mine)[0])
                self.backend.deliver_weighted(
                    self.state,
                    self.pool,
                    self.data,
                    self.spec,
                    self._to_local_t(phase.wm_nodes[mine]),
                    torch.from_numpy(new_ptr),
                    torch.from_numpy(phase.wm_slots[sel].astype(np.int64)),
                    torch.from_numpy(phase.wm_weights[sel].astype(np.float32)),
                    torch.from_numpy(phase.wm_self_w[mine].astype(np.float32)),
                )
        smine = self._is_mine(phase.snap_nodes)
        if smine.any():
            self.backend.snapshot(
                self.state,
                self.pool,
                self._to_local_t(phase.snap_nodes[smine]),
                torch.from_numpy(phase.snap_slots[smine].astype(np.int64)).to(
                    self.device
                ),
            )


class BatchedCacheNeighGossipSimulator(BatchedGossipSimulator):
    """Cache-neighborhood gossip on the batched engine (CacheNeighNode,
    gossipy/node.py:395-496): deliveries only store the model; a timed-out
    node merges one randomly chosen cached neighbor model before sending.
    Scheduling is host-side (CacheNeighScheduler); the kernels are the
    ordinary per-family tick kernels."""

    def __init__(
        self,
        cfg: EngineConfig,
        spec,
        data: DataArena,
        device: Optional[torch.device] = None,
    ):
        super().__init__(cfg, spec, data, device=device)
        from .schedule import CacheNeighScheduler

        self.scheduler = CacheNeighScheduler(cfg)

    def _fast_path_ok(self) -> bool:
        return False


class BatchedPENSGossipSimulator(BatchedGossipSimulator):
    """PENS on the batched engine (PENSNode, gossipy/node.py:663-785).

    Step-1 scoring/merging runs as one kernel per tick (tick_pens) with
    winner counts accumulated device-side; the only host<->device
    dependency is ONE counts read at the step boundary, where
    ``best_nodes`` is computed and the scheduler's step-2 peer draws are
    restricted to it. PUSH + MERGE_UPDATE; logreg scores candidates
    in-kernel (tick_pens), torchmod (the Onoszko CIFAR10Net CNN) scores
    them with a node-batched vmap forward.
    """

    def __init__(
        self,
        cfg: EngineConfig,
        spec,
        data: DataArena,
        n_sampled: int = 10,
        m_top: int = 2,
        step1_rounds: int = 10,
        device: Optional[torch.device] = None,
    ):
        super().__init__(cfg, spec, data, device=device)
        from ..core import CreateModelMode
        from .schedule import PENSScheduler

        assert spec.mode == CreateModelMode.MERGE_UPDATE, (
            "PENSNode can only be used with MERGE_UPDATE mode."
        )
        self._ctor_desc = ("pens", n_sampled, m_top, step1_rounds)
        self.scheduler = PENSScheduler(cfg, n_sampled, m_top, step1_rounds)
        self.m_top = m_top
        #: device-side winner counters [n_local, n_nodes]
        self.counts = torch.zeros(
            self.n_local, cfg.n_nodes, dtype=torch.int32, device=self.device
        )

    def _checkpoint_extra(self) -> dict:
        extra = super()._checkpoint_extra()
        extra["pens_counts"] = self.counts.cpu()
        extra["pens_selected"] = self.scheduler.selected.copy()
        extra["pens_best_nodes"] = (
            None
            if self.scheduler.best_nodes is None
            else [b.copy() for b in self.scheduler.best_nodes]
        )
        return extra

    def _replay_rounds(self, rounds_done: int, extra: dict) -> None:
        # the PENS schedule is data-dependent: step-2 peer draws read
        # best_nodes, which came from device-side winner counts. Replay
        # step-1 rounds with best_nodes=None (as the saved run did), then
        # re-inject the SAVED best_nodes before replaying step-2 rounds —
        # otherwise the replay takes the step-1 draw path and diverges.
        boundary = min(rounds_done, self.scheduler.step1_rounds)
        for r in range(boundary):
            self.scheduler.next_round(r)
        if rounds_done > self.scheduler.step1_rounds or (
            rounds_done == self.scheduler.step1_rounds
            and extra.get("pens_best_nodes") is not None
        ):
            saved = extra.get("pens_best_nodes")
            if saved is None:
                raise ValueError(
                    "PENS checkpoint past the step-1 boundary has no saved "
                    "best_nodes — blob predates PENS checkpoint support"
                )
            self.scheduler.best_nodes = [
                np.asarray(b, dtype=np.int64) for b in saved
            ]
            for r in range(boundary, rounds_done):
                self.scheduler.next_round(r)

    def _restore_extra(self, extra: dict) -> None:
        if "pens_counts" not in extra:
            raise ValueError(
                "blob predates PENS checkpoint support (no pens_counts)"
            )
        self.counts.copy_(extra["pens_counts"].to(self.device))
        self.scheduler.selected = np.asarray(
            extra["pens_selected"], dtype=np.int64
        )
        saved = extra.get("pens_best_nodes")
        self.scheduler.best_nodes = (
            None if saved is None
            else [np.asarray(b, dtype=np.int64) for b in saved]
        )

    def _fast_path_ok(self) -> bool:
        return False

    def _run_tick(self, phase: TickPhase) -> None:
        super()._run_tick(phase)
        if phase.pens_nodes is not None and len(phase.pens_nodes):
            self._exchange(
                self._plan_exchange(
                    phase.pens_nodes, phase.pens_ptr, phase.pens_slots,
                    phase.pens_owners,
                )
            )
            mine = self._is_mine(phase.pens_nodes)
            if mine.any():
                sel, new_ptr = _csr_gather(phase.pens_ptr, np.where(mine)[0])
                self.backend.deliver_pens(
                    self.state,
                    self.pool,
                    self.data,
                    self.spec,
                    self._to_local_t(phase.pens_nodes[mine]),
                    torch.from_numpy(new_ptr),
                    torch.from_numpy(phase.pens_slots[sel].astype(np.int64)),
                    torch.from_numpy(phase.pens_owners[sel].astype(np.int64)),
                    self.counts,
                    self.m_top,
                )

    def _select_neighbors(self) -> None:
        """Step boundary: read the device counters, assemble the full
        [n_nodes, n_nodes] matrix on every rank, and fix step-2 topology."""
        local = self.counts.cpu().numpy()
        if self.world > 1:
            gathered = [
                torch.empty_like(self.counts) for _ in range(self.world)
            ]
            dist.all_gather(gathered, self.counts.contiguous())
            full = torch.cat(gathered, dim=0).cpu().numpy()
        else:
            full = local
        self.scheduler.select_neighbors(full.astype(np.int64))

    def start(self, n_rounds: int = 100) -> None:
        assert self.initialized, "call init_nodes() first"
        for _ in range(n_rounds):
            r = self.rounds_done
            if (
                r >= self.scheduler.step1_rounds
                and self.scheduler.best_nodes is None
            ):
                self._select_neighbors()
            sched = self.scheduler.next_round(r)
            self.pool.ensure(sched.n_slots)
            for phase in sched.ticks:
                self._run_tick(phase)
            if self.rank == 0:
                self.notify_message_counts(sched)
            self._evaluate(sched, (r + 1) * self.cfg.delta - 1)
            self.rounds_done += 1
            self.notify_timestep((r + 1) * self.cfg.delta - 1)
        self._drain_eval()
        self.notify_end()


class RoundTimer:
    """Opt-in GPU round timing via HIP events (SURVEY.md §5's tracing gap).

    ``with RoundTimer(sim) as rt: sim.start(...)`` brackets every round's
    device work with ``torch.cuda.Event`` pairs (hipEvent under ROCm) and
    reports per-round GPU milliseconds — kernel + comm time as the GPU saw
    it, independent of host overlap. On CPU devices it falls back to
    wall-clock.
    """

    def __init__(self, sim: BatchedGossipSimulator):
        self.sim = sim
        self.gpu_ms: List[float] = []
        self._orig = None

    def __enter__(self):
        sim = self.sim
        orig = sim._run_round_fast if hasattr(sim, "_run_round_fast") else None
        timer = self
        use_cuda = sim.device.type == "cuda"
        orig_start = sim.start

        def timed_start(n_rounds: int = 100):
            import time as _t

            for _ in range(n_rounds):
                if use_cuda:
                    e0 = torch.cuda.Event(enable_timing=True)
                    e1 = torch.cuda.Event(enable_timing=True)
                    e0.record()
                    orig_start(n_rounds=1)
                    e1.record()
                    e1.synchronize()
                    timer.gpu_ms.append(e0.elapsed_time(e1))
                else:
                    t0 = _t.perf_counter()
                    orig_start(n_rounds=1)
                    timer.gpu_ms.append(1000.0 * (_t.perf_counter() - t0))

        self._orig = orig_start
        sim.start = timed_start
        return self

    def __exit__(self, *exc):
        self.sim.start = self._orig
        return False

    @property
    def mean_ms(self) -> float:
        return sum(self.gpu_ms) / max(1, len(self.gpu_ms))
