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

from dataclasses import dataclass, field
from typing import Tuple

from ..core import CreateModelMode

__all__ = ["PegasosSpec", "AdaLineSpec", "LogRegSpec", "MLPSpec"]


@dataclass
class PegasosSpec:
    """Pegasos SVM: per-sample hinge updates with ``lr = 1/(t*lam)``
    (gossipy/model/handler.py:416-423). Labels in {-1, +1}."""

    d_in: int
    lam: float = 0.01  #: the handler's ``learning_rate`` (lambda)
    mode: CreateModelMode = CreateModelMode.MERGE_UPDATE

    family = "pegasos"

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

    family = "adaline"

    @property
    def D(self) -> int:
        return self.d_in


@dataclass
class LogRegSpec:
    """Logistic regression ``softmax-CE(sigmoid(W x + b))`` with plain SGD.

    Parameter row layout: ``W`` (k*d, row-major) then ``b`` (k).
    """

    d_in: int
    n_classes: int = 2
    lr: float = 0.1
    weight_decay: float = 0.0
    local_epochs: int = 1
    batch_size: int = 32  #: 0 = full-batch
    mode: CreateModelMode = CreateModelMode.MERGE_UPDATE

    family = "logreg"

    @property
    def D(self) -> int:
        return self.n_classes * self.d_in + self.n_classes


@dataclass
class MLPSpec:
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

    family = "mlp"

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
