"""Model-family specs for the batched engine.

Each spec describes one model family's flat parameter layout and
hyperparameters. The engine stores every node's model as one row of a packed
``[n_nodes, D]`` fp32 arena; the spec defines what the D scalars mean and
which batched kernel trains them:

* ``PegasosSpec`` / ``AdaLineSpec`` — flat weight vector ``w[d]``
  (gossipy/model/nn.py:116-143); trained by the K1/K2 per-sample kernels
  (SURVEY.md §2.4).
* ``LogRegSpec`` — ``W[k,d]`` row-major followed by ``b[k]``
  (gossipy/model/nn.py:147-174, trained as in
  gossipy/model/handler.py:235-258); fused forward/backward/SGD kernel
  (K3/K4). The reference's LogisticRegression applies a sigmoid and then
  CrossEntropyLoss *on the sigmoid outputs* — the kernels replicate that
  exact composition.
* ``MLPSpec`` — hidden layers with ReLU, linear head
  (gossipy/model/nn.py:67-113); layer blocks packed back-to-back.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

import numpy as np

from ..core import CreateModelMode

__all__ = ["PegasosSpec", "AdaLineSpec", "LogRegSpec", "MLPSpec", "MFSpec", "KMeansSpec", "TorchModuleSpec"]


def _fortran_to_arena(layers: List[Tuple[int, int, int, int]], D: int) -> np.ndarray:
    """Permutation mapping *reference partition order* -> arena offset.

    The reference's :class:`TorchModelPartition` splits the model into
    contiguous ranges of the Fortran-flat (dim-0 fastest) concatenation of
    its parameter tensors (gossipy/model/sampling.py:144-198). The arena
    packs each ``W`` row-major. ``perm[g]`` is the arena offset of global
    F-flat position ``g``, so partition ``p`` is ``perm[lo:hi]`` for the
    p-th equal split ``[lo, hi)``.

    ``layers``: per-layer ``(w_off, b_off, fin, fout)`` arena offsets.
    """
    perm = np.empty(D, dtype=np.int32)
    g = 0
    for (w_off, b_off, fin, fout) in layers:
        n_w = fin * fout
        idx = np.arange(n_w)
        # F-order over shape (fout, fin): position i -> (row i%fout, col i//fout)
        perm[g : g + n_w] = w_off + (idx % fout) * fin + idx // fout
        g += n_w
        perm[g : g + fout] = b_off + np.arange(fout)
        g += fout
    assert g == D
    return perm


class _PartitionMixin:
    """Partition cover helpers shared by partitionable specs.

    Active when ``n_parts > 0``. The cover replicates the reference's
    equal-size F-flat split: partition ``p`` owns ``mu`` scalars (``mu+1``
    for the first ``D % n_parts``).
    """

    def part_ptr(self) -> np.ndarray:
        """``[P+1]`` int32 boundaries of the partition split."""
        P = self.n_parts
        mu, rem = divmod(self.D, P)
        sizes = np.full(P, mu, dtype=np.int32)
        sizes[:rem] += 1
        return np.concatenate([[0], np.cumsum(sizes)]).astype(np.int32)

    def part_perm(self) -> np.ndarray:
        """``[D]`` int32: arena offset of each F-flat position (partition
        ``p`` = ``part_perm()[ptr[p]:ptr[p+1]]``)."""
        return _fortran_to_arena(self._arena_layers(), self.D)

    def arena_part(self) -> np.ndarray:
        """``[D]`` int32: partition id owning each *arena* offset (the
        inverse view, used by the gradient age-rescale,
        gossipy/model/handler.py:514-520)."""
        perm = self.part_perm()
        ptr = self.part_ptr()
        out = np.empty(self.D, dtype=np.int32)
        for p in range(self.n_parts):
            out[perm[ptr[p] : ptr[p + 1]]] = p
        return out

    @property
    def age_width(self) -> int:
        return max(1, self.n_parts)


@dataclass
class PegasosSpec:
    """Pegasos SVM: per-sample hinge updates with ``lr = 1/(t*lam)``
    (gossipy/model/handler.py:416-423). Labels in {-1, +1}."""

    d_in: int
    lam: float = 0.01  #: the handler's ``learning_rate`` (lambda)
    mode: CreateModelMode = CreateModelMode.MERGE_UPDATE
    #: pass-through gossip (PassThroughNode): deliveries may resolve to PASS
    pass_through: bool = False

    family = "pegasos"
    n_parts = 0
    age_width = 1

    @property
    def D(self) -> int:
        return self.d_in


@dataclass
class AdaLineSpec:
    """AdaLine delta rule ``w += lr*(y - <w,x>)*x``
    (gossipy/model/handler.py:364-368). Labels in {-1, +1}."""

    d_in: int
    lr: float = 0.01
    mode: CreateModelMode = CreateModelMode.MERGE_UPDATE
    pass_through: bool = False

    family = "adaline"
    n_parts = 0
    age_width = 1

    @property
    def D(self) -> int:
        return self.d_in


@dataclass
class LogRegSpec(_PartitionMixin):
    """Logistic regression ``softmax-CE(sigmoid(W x + b))`` with plain SGD.

    Parameter row layout: ``W`` (k*d, row-major) then ``b`` (k).
    ``n_parts > 0`` switches the family to partitioned gossip
    (PartitionedTMH semantics: per-partition ages, age-weighted partition
    merge, gradient age-rescale — gossipy/model/handler.py:455-525).
    """

    d_in: int
    n_classes: int = 2
    lr: float = 0.1
    weight_decay: float = 0.0
    local_epochs: int = 1
    batch_size: int = 32  #: 0 = full-batch
    mode: CreateModelMode = CreateModelMode.MERGE_UPDATE
    n_parts: int = 0
    #: >0 = sampled gossip (SamplingBasedNode, gossipy/node.py:499-562):
    #: each delivery merges only a random ``sample_size`` fraction of the
    #: coordinates. Engine-native sampling: uniform over the flat parameter
    #: row with replacement — the same distribution as the reference's
    #: numel-weighted layer multinomial + per-dim uniform draw
    #: (gossipy/model/sampling.py:37-72), derived from a per-delivery tape
    #: seed instead of global RNG. Mutually exclusive with ``n_parts``.
    sample_size: float = 0.0
    #: pass-through gossip (PassThroughNode): deliveries may resolve to PASS
    pass_through: bool = False

    family = "logreg"

    def samp_count(self) -> int:
        """Coordinates per sampled merge (gossipy/model/sampling.py:57)."""
        return max(1, int(round(self.sample_size * self.D)))

    @property
    def D(self) -> int:
        return self.n_classes * self.d_in + self.n_classes

    def _arena_layers(self):
        kd = self.n_classes * self.d_in
        return [(0, kd, self.d_in, self.n_classes)]


@dataclass
class MLPSpec(_PartitionMixin):
    """MLP with ReLU hidden layers and a linear head
    (gossipy/model/nn.py:67-113); CrossEntropyLoss on the raw head outputs.

    Parameter row layout: per layer ``W_i`` (out_i * in_i, row-major) then
    ``b_i`` (out_i), layers packed in order.
    """

    d_in: int
    n_classes: int
    hidden: Tuple[int, ...] = (100,)
    lr: float = 0.1
    weight_decay: float = 0.0
    local_epochs: int = 1
    batch_size: int = 32
    mode: CreateModelMode = CreateModelMode.MERGE_UPDATE
    n_parts: int = 0
    pass_through: bool = False

    family = "mlp"

    def _arena_layers(self):
        return self.layer_offsets()

    @property
    def dims(self) -> Tuple[int, ...]:
        return (self.d_in, *self.hidden, self.n_classes)

    @property
    def D(self) -> int:
        dims = self.dims
        return sum(dims[i + 1] * dims[i] + dims[i + 1] for i in range(len(dims) - 1))

    def layer_offsets(self):
        """Per-layer ``(w_off, b_off, in, out)`` tuples into the flat row."""
        dims = self.dims
        off = 0
        out = []
        for i in range(len(dims) - 1):
            w_off = off
            b_off = off + dims[i + 1] * dims[i]
            off = b_off + dims[i + 1]
            out.append((w_off, b_off, dims[i], dims[i + 1]))
        return out


@dataclass
class MFSpec:
    """Low-rank matrix-factorization recommender (MFModelHandler,
    gossipy/model/handler.py:528-576).

    Arena row layout: ``X`` (k user factors), ``b`` (1), ``Y``
    (n_items*k, row-major), ``c`` (n_items). Messages carry only the item
    block ``(Y, c)`` — the merge touches nothing else
    (gossipy/model/handler.py:562-568) — so ``slot_width`` is the item
    block size, not ``D``. Quirk parity: the merge divides by ``2*(w1+w2)``
    (the reference's extra factor of two, handler.py:566-567) and does NOT
    update ``n_updates``; ages start at 1 (handler.py:540).

    Only MERGE_UPDATE is supported on the engine: UPDATE-style adoption
    would need the sender's *user* factors, which the item-block transport
    deliberately omits (use the object layer for those modes).
    """

    k: int  #: latent dimension (``dim``)
    n_items: int
    reg: float = 0.1  #: ``lam_reg``
    lr: float = 0.001
    r_min: int = 1
    r_max: int = 5
    mode: CreateModelMode = CreateModelMode.MERGE_UPDATE

    family = "mf"
    n_parts = 0
    sample_size = 0.0
    age_width = 1

    @property
    def D(self) -> int:
        return self.k + 1 + self.n_items * self.k + self.n_items

    @property
    def item_off(self) -> int:
        return self.k + 1

    @property
    def slot_width(self) -> int:
        return self.n_items * self.k + self.n_items


@dataclass
class KMeansSpec:
    """Online k-means with EMA centroid updates (KMeansHandler,
    gossipy/model/handler.py:579-639).

    Arena row layout: centroids ``C`` (k*dim, row-major). Quirk parity: the
    per-call update assigns every sample to its nearest centroid and applies
    ``C[idx] = (1-alpha)*C[idx] + alpha*x`` with torch indexed-assignment
    semantics — duplicate assignments collapse to the LAST sample per
    centroid (handler.py:613-615); ``n_updates`` bumps once per update call,
    and merges never touch it.

    ``matching="hungarian"`` (bug-fixed column-index matching, see
    gossipy_amd/model/handler.py — the reference's own hungarian path is
    a no-op identity permutation, handler.py:629-630) runs on BOTH
    backends: the torch loop, and on the engine a wave-batched device
    cdist + host assignment + batched matched-mean feeding the HIP EMA
    update (HIPBackend._deliver_kmeans_hungarian). ``naive`` is the
    behavior the reference actually ships and stays fully in-kernel.
    """

    k: int
    dim: int
    alpha: float = 0.1
    matching: str = "naive"
    mode: CreateModelMode = CreateModelMode.UPDATE

    family = "kmeans"
    n_parts = 0
    sample_size = 0.0
    age_width = 1

    @property
    def D(self) -> int:
        return self.k * self.dim


def _bmm_convs(module) -> None:
    """Rewrite every plain (groups=1, 2-d) ``nn.Conv2d`` in ``module`` to
    compute via im2col (``unfold``) + ``matmul`` instead of MIOpen.

    The node-batched engine paths run the module under ``torch.func.vmap``
    with per-node weights, which MIOpen sees as a grouped conv with
    ``groups = vmap-batch`` — a distinct solver problem per batch size,
    and on gfx950 the grouped solvers fall back to slow kernels (measured:
    the Onoszko eval forward alone cost ~96 ms/round). unfold+matmul maps
    the same math onto rocBLAS strided-batched GEMMs: no solver search,
    no grouped-conv penalty, fully vmappable, autograd = more GEMMs.
    Parameters and state-dict keys are untouched."""
    import types

    import torch
    import torch.nn.functional as F

    def bmm_forward(self, x):
        kh, kw = self.kernel_size
        cols = F.unfold(
            x, self.kernel_size, dilation=self.dilation,
            padding=self.padding, stride=self.stride,
        )  # [N, C*kh*kw, L]
        out = self.weight.reshape(self.out_channels, -1) @ cols
        if self.bias is not None:
            out = out + self.bias[:, None]
        ph, pw = self.padding if isinstance(self.padding, tuple) else (
            self.padding, self.padding
        )
        dh, dw = self.dilation
        sh, sw = self.stride
        ho = (x.shape[-2] + 2 * ph - dh * (kh - 1) - 1) // sh + 1
        wo = (x.shape[-1] + 2 * pw - dw * (kw - 1) - 1) // sw + 1
        return out.reshape(*out.shape[:-2], self.out_channels, ho, wo)

    for sub in module.modules():
        if (
            type(sub) is torch.nn.Conv2d
            and sub.groups == 1
            and sub.padding_mode == "zeros"
            and not isinstance(sub.padding, str)
        ):
            sub.forward = types.MethodType(bmm_forward, sub)


class TorchModuleSpec:
    """Engine family for arbitrary ``nn.Module`` architectures (the CNN
    path of SURVEY.md §7 step 7 — Onoszko 2021's CIFAR10Net,
    main_onoszko_2021.py:31-60 — and any user model).

    The node population's parameters still live in the packed ``[n, D]``
    arena (flattened in ``parameters()`` order, each tensor C-contiguous),
    so snapshots, merges and cross-GPU traffic use the same HIP kernels and
    RCCL paths as every other family. The local SGD step runs through torch
    autograd on a per-node view of the arena row — convolutions therefore
    execute on MIOpen, which is the measured-right choice for 3x3 convs
    ("conv via MIOpen or an im2col+MFMA kernel — decide by
    measurement").

    ``module_factory`` must build a fresh module (identical architecture)
    each call; inputs may be N-d (the DataArena stores them flattened and
    ``input_shape`` restores them).
    """

    family = "torchmod"
    n_parts = 0
    sample_size = 0.0
    pass_through = False
    age_width = 1

    def __init__(
        self,
        module_factory,
        input_shape,
        lr: float = 0.1,
        weight_decay: float = 0.0,
        local_epochs: int = 1,
        batch_size: int = 32,
        mode: CreateModelMode = CreateModelMode.MERGE_UPDATE,
    ):
        self.module_factory = module_factory
        self.input_shape = tuple(input_shape)
        self.lr = lr
        self.weight_decay = weight_decay
        self.local_epochs = local_epochs
        self.batch_size = batch_size
        self.mode = mode
        proto = module_factory()
        self._names = [n for n, _ in proto.named_parameters()]
        self._shapes = [tuple(p.shape) for p in proto.parameters()]
        self._numels = [int(np.prod(s)) for s in self._shapes]
        self._D = int(sum(self._numels))

    def __getstate__(self):
        # checkpoints dill the spec; the lazily-built template module
        # (possibly holding device tensors) is derived state — drop it
        st = dict(self.__dict__)
        st.pop("_template", None)
        return st

    def __setstate__(self, st):
        self.__dict__.update(st)

    def param_layout(self):
        """``(name, shape, offset, numel)`` per parameter tensor, in
        ``parameters()`` order — the arena row layout."""
        out = []
        off = 0
        for name, shape, n in zip(self._names, self._shapes, self._numels):
            out.append((name, shape, off, n))
            off += n
        return out

    @property
    def D(self) -> int:
        return self._D

    @property
    def d_in(self) -> int:
        return int(np.prod(self.input_shape))

    def template(self):
        """A module instance for row_to/row_from round-trips and for the
        batched vmap paths. Plain ``nn.Conv2d`` forwards are rewritten to
        unfold+matmul (see :func:`_bmm_convs`) unless ``GOSSIPY_MIOPEN_CONV=1``."""
        if not hasattr(self, "_template"):
            self._template = self.module_factory()
            import os

            if os.environ.get("GOSSIPY_BMM_CONV") == "1":
                # opt-in A/B: measured SLOWER than MIOpen grouped conv
                # under vmap on gfx950 (unfold's vmap rule materializes
                # im2col per example) — kept for comparison runs
                _bmm_convs(self._template)
        return self._template

    def load_row(self, module, row) -> None:
        """Copy an arena row into a module's parameters (in place)."""
        import torch

        off = 0
        with torch.no_grad():
            for p, n in zip(module.parameters(), self._numels):
                p.copy_(row[off : off + n].view(p.shape))
                off += n

    def store_row(self, module, row) -> None:
        import torch

        off = 0
        with torch.no_grad():
            for p, n in zip(module.parameters(), self._numels):
                row[off : off + n] = p.reshape(-1)
                off += n

    def init_row(self, generator) -> "np.ndarray":
        """Xavier-uniform weights / zero biases per the reference's intent
        (main_onoszko_2021.py:38-44), drawn from the tape stream."""
        out = np.empty(self._D, dtype=np.float32)
        off = 0
        for shape, n in zip(self._shapes, self._numels):
            if len(shape) >= 2:
                fan_out = shape[0] * int(np.prod(shape[2:]))
                fan_in = shape[1] * int(np.prod(shape[2:]))
                bound = float(np.sqrt(6.0 / (fan_in + fan_out)))
                out[off : off + n] = generator.uniform(-bound, bound, size=n)
            else:
                out[off : off + n] = 0.0
            off += n
        return out
