"""Packed device arenas for the batched engine.

The reference keeps each node's model in a Python object and "transports"
it by deep-copying the whole handler into a global cache on every send
(gossipy/model/handler.py:160-176 — measured at ~36% of its runtime,
SURVEY.md §6). The engine replaces both with flat HBM arenas:

* :class:`NodeStateArena` — one fp32 row per *resident* node (params) plus
  an int32 age vector (``n_updates``);
* :class:`SlotPool` — the per-round snapshot pool: one row per sent
  message, written by the sender's snapshot kernel and read by the
  receiver's merge kernel. A "send" is a row copy on-device; a cross-GPU
  send is an RCCL transfer of that row. Ref-count semantics of the
  reference cache collapse to round-scoped slot lifetime (every slot is
  written once and consumed within its delivery schedule).
* :class:`DataArena` — per-node train/test shards padded to a fixed
  ``S_max`` stride so batched kernels index with one multiply.

All tensors live on the runner's device (``cuda:local_rank`` on MI355X —
288 GB HBM3E makes padding a non-issue at any supported node count).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

__all__ = ["NodeStateArena", "SlotPool", "DataArena"]


class NodeStateArena:
    """Parameters + ages of this rank's resident nodes.

    ``node_lo/node_hi`` give the global-id range owned by this rank
    (contiguous block residency); global index ``g`` maps to row
    ``g - node_lo``.
    """

    def __init__(
        self,
        n_local: int,
        D: int,
        device: torch.device,
        node_lo: int = 0,
        age_width: int = 1,
    ):
        self.params = torch.zeros(n_local, D, device=device, dtype=torch.float32)
        # ``age_width > 1``: per-partition age vectors (PartitionedTMH's
        # ``n_updates`` array, gossipy/model/handler.py:475); else scalar age
        if age_width > 1:
            self.ages = torch.zeros(n_local, age_width, device=device, dtype=torch.int32)
        else:
            self.ages = torch.zeros(n_local, device=device, dtype=torch.int32)
        self.age_width = age_width
        self.node_lo = node_lo
        self.n_local = n_local
        self.D = D

    def to_local(self, global_ids: torch.Tensor) -> torch.Tensor:
        return global_ids - self.node_lo


class SlotPool:
    """Per-round snapshot slot pool (grown geometrically, never shrunk)."""

    def __init__(
        self, D: int, device: torch.device, capacity: int = 1024, age_width: int = 1
    ):
        self.D = D
        self.device = device
        self.age_width = age_width
        self.slots = torch.zeros(capacity, D, device=device, dtype=torch.float32)
        self.slot_ages = self._new_ages(capacity)

    def _new_ages(self, cap: int) -> torch.Tensor:
        if self.age_width > 1:
            return torch.zeros(
                cap, self.age_width, device=self.device, dtype=torch.int32
            )
        return torch.zeros(cap, device=self.device, dtype=torch.int32)

    def ensure(self, n_slots: int) -> None:
        if n_slots > self.slots.shape[0]:
            cap = max(n_slots, 2 * self.slots.shape[0])
            old_s, old_a = self.slots, self.slot_ages
            self.slots = torch.zeros(
                cap, self.D, device=self.device, dtype=torch.float32
            )
            self.slot_ages = self._new_ages(cap)
            # slots of in-flight (delayed) messages survive the growth;
            # the copy is stream-ordered after any kernel still reading them
            self.slots[: old_s.shape[0]].copy_(old_s)
            self.slot_ages[: old_a.shape[0]].copy_(old_a)


class DataArena:
    """Padded per-node data shards.

    ``x``: ``[n_local, S_max, d]``, ``y``: ``[n_local, S_max]`` (float;
    class index or ±1 target by model family), ``counts``: ``[n_local]``.
    A second triple holds the local *test* shards when per-user eval is on,
    and ``gx/gy`` the shared global test set.
    """

    def __init__(
        self,
        x: torch.Tensor,
        y: torch.Tensor,
        counts: torch.Tensor,
        tx: Optional[torch.Tensor] = None,
        ty: Optional[torch.Tensor] = None,
        tcounts: Optional[torch.Tensor] = None,
        gx: Optional[torch.Tensor] = None,
        gy: Optional[torch.Tensor] = None,
    ):
        self.x, self.y, self.counts = x, y, counts
        self.tx, self.ty, self.tcounts = tx, ty, tcounts
        self.gx, self.gy = gx, gy

    @property
    def S_max(self) -> int:
        return self.x.shape[1]

    @staticmethod
    def from_shards(
        shards: list,
        device: torch.device,
        test_shards: Optional[list] = None,
        global_eval: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
    ) -> "DataArena":
        """Pack a list of ``(X_i, y_i)`` shards into padded arenas."""

        def pack(sh):
            n = len(sh)
            d = sh[0][0].shape[1]
            smax = max(int(x.shape[0]) for x, _ in sh)
            X = torch.zeros(n, smax, d, dtype=torch.float32)
            Y = torch.zeros(n, smax, dtype=torch.float32)
            C = torch.zeros(n, dtype=torch.int32)
            for i, (x, y) in enumerate(sh):
                c = int(x.shape[0])
                X[i, :c] = x.float()
                Y[i, :c] = y.float().reshape(-1)
                C[i] = c
            return X.to(device), Y.to(device), C.to(device)

        x, y, c = pack(shards)
        tx = ty = tc = None
        if test_shards is not None:
            tx, ty, tc = pack(test_shards)
        gx = gy = None
        if global_eval is not None:
            gx = global_eval[0].float().to(device)
            gy = global_eval[1].float().reshape(-1).to(device)
        return DataArena(x, y, c, tx, ty, tc, gx, gy)
