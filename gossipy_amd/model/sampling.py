"""Model compression ops: random parameter sampling and deterministic
partitioning (parity: gossipy/model/sampling.py).

Both exist to shrink gossip payloads. In the batched engine these become
indexed-merge kernels: the partition cover is precomputed once as contiguous
ranges over the packed parameter arena (Fortran-flat order, see
:meth:`TorchModelPartition.flat_ranges`), so the partition merge is a
segment-offset variant of the full merge kernel rather than a gather/scatter.
"""

from __future__ import annotations

from collections import Counter
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
from numpy.random import choice
from torch import LongTensor
from torch.nn import ParameterList

from .. import LOG
from . import TorchModel

__all__ = ["TorchModelSampling", "TorchModelPartition"]


class TorchModelSampling:
    """Random subset sampling of a model's parameters (static methods only).

    Parity: gossipy/model/sampling.py:27-107 — layers are drawn from a
    multinomial weighted by parameter count, then per-dimension indices are
    drawn uniformly *with replacement* (so a sample may contain duplicates,
    exactly like the reference).
    """

    @classmethod
    def sample(
        cls, size: float, net: TorchModel
    ) -> Dict[int, Optional[Tuple[LongTensor, ...]]]:
        """Draw a ``size`` fraction of the parameters of ``net``.

        Returns a dict ``layer index -> tuple of per-dim index tensors`` (or
        ``None`` for layers with no sampled coordinates).
        """
        assert 0 < size <= 1, "size must be in the range (0, 1]."
        if size >= 0.9:
            LOG.warning(
                "You are using a high sample size (=%.2f) which can impact "
                "the performance without much advantage in terms of saved bandwith."
                % size
            )

        plist = ParameterList(net.parameters())
        probs = np.array([t.numel() for t in plist], dtype="float")
        probs /= probs.sum()
        sample_size = max(1, int(round(size * net.get_size())))
        per_layer = Counter(choice(len(plist), size=sample_size, p=probs))
        samples: Dict[int, Optional[Tuple[LongTensor, ...]]] = {
            i: None for i in range(len(plist))
        }
        for layer, count in per_layer.items():
            shape = tuple(plist[layer].size())
            samples[layer] = tuple(
                LongTensor(list(choice(dim, size=count))) for dim in shape
            )
        return samples

    @classmethod
    def merge(
        cls,
        sample: Dict[int, Optional[Tuple[LongTensor, ...]]],
        net1: TorchModel,
        net2: TorchModel,
        reduce: str = "mean",
    ) -> None:
        """In-place combine of ``net2`` into ``net1`` on the sampled
        coordinates only (gossipy/model/sampling.py:75-107)."""
        assert str(net1) == str(net2), "net1 and net2 must have the same architecture."
        assert reduce in {"mean", "sum"}, "reduce must be either 'sum' or 'mean'."

        plist1 = ParameterList(net1.parameters())
        plist2 = ParameterList(net2.parameters())
        assert len(plist1) == len(sample), (
            "The provided sample is incompatible with the network."
        )
        div = 2 if reduce == "mean" else 1
        with torch.no_grad():
            for i in range(len(plist1)):
                ids = sample[i]
                if ids is not None:
                    plist1[i][ids] = (plist1[i][ids] + plist2[i][ids]) / div


class TorchModelPartition:
    """Deterministic equal-size split of a model's parameters into
    ``n_parts`` partitions.

    Semantics parity with gossipy/model/sampling.py:110-198: the reference
    walks each tensor's dim-0 fastest with the remaining dims held fixed,
    i.e. it assigns **contiguous ranges in Fortran (column-major) flat order**
    across the concatenation of all parameter tensors; partition ``p`` owns
    ``mu`` elements (``mu+1`` for the first ``size % n`` partitions). This
    implementation computes those ranges directly and materializes the same
    per-layer index tuples.
    """

    def __init__(self, net_proto: TorchModel, n_parts: int):
        self._check(net_proto)
        self.str_arch = str(net_proto)
        net_size = net_proto.get_size()
        self.n_parts = min(n_parts, net_size)
        self._layer_shapes = [tuple(p.shape) for p in net_proto.parameters()]
        self.partitions = self._build(net_size)

    @staticmethod
    def _check(net: TorchModel) -> None:
        for t in net.parameters():
            if t.dim() > 3:
                raise TypeError(
                    "Partitioning is only supported for neural "
                    "networks with at most 3D layers."
                )

    def flat_ranges(self) -> List[Tuple[int, int]]:
        """Per-partition ``[start, end)`` ranges in the global Fortran-flat
        parameter order — the layout the batched engine packs its arena in,
        making each partition a contiguous slab."""
        total = sum(int(np.prod(s)) if s else 1 for s in self._layer_shapes)
        mu, rem = divmod(total, self.n_parts)
        ranges = []
        start = 0
        for p in range(self.n_parts):
            length = mu + (1 if p < rem else 0)
            ranges.append((start, start + length))
            start += length
        return ranges

    def _build(self, net_size: int) -> Dict[int, Dict[int, Optional[Tuple[LongTensor, ...]]]]:
        # global Fortran-flat offset of each layer
        layer_sizes = [int(np.prod(s)) if s else 1 for s in self._layer_shapes]
        layer_offsets = np.cumsum([0] + layer_sizes)
        parts: Dict[int, Dict[int, Optional[Tuple[LongTensor, ...]]]] = {}
        for p, (lo, hi) in enumerate(self.flat_ranges()):
            per_layer: Dict[int, Optional[Tuple[LongTensor, ...]]] = {}
            for li, shape in enumerate(self._layer_shapes):
                a = max(lo, int(layer_offsets[li]))
                b = min(hi, int(layer_offsets[li + 1]))
                if a >= b:
                    per_layer[li] = None
                    continue
                flat = np.arange(a, b) - int(layer_offsets[li])
                idx = np.unravel_index(flat, shape, order="F")
                per_layer[li] = tuple(torch.LongTensor(d) for d in idx)
            parts[p] = per_layer
        return parts

    def _partition(self, net_size: int):
        """Reference-private alias (gossipy/model/sampling.py:144-198):
        builds the per-partition index cover — same output as the
        range-based :meth:`_build`."""
        return self._build(net_size)

    def merge(
        self,
        id_part: int,
        net1: TorchModel,
        net2: TorchModel,
        weights: Optional[Tuple[int, int]] = None,
    ) -> None:
        """Weighted in-place merge of one partition of ``net2`` into ``net1``
        (gossipy/model/sampling.py:201-234)."""
        assert str(net1) == self.str_arch, "net1 is not compatible."
        assert str(net2) == self.str_arch, "net2 is not compatible."

        id_part = id_part % self.n_parts
        plist1 = ParameterList(net1.parameters())
        plist2 = ParameterList(net2.parameters())
        w = weights if (weights is not None and weights != (0, 0)) else (1, 1)
        mul1, mul2 = w[0] / sum(w), w[1] / sum(w)
        with torch.no_grad():
            for i in range(len(plist1)):
                ids = self.partitions[id_part][i]
                if ids is not None:
                    plist1[i][ids] = mul1 * plist1[i][ids] + mul2 * plist2[i][ids]
