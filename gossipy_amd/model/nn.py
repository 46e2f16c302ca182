"""Built-in model families (parity: gossipy/model/nn.py).

These are the object-layer definitions; the batched engine executes the same
math over packed parameter arenas (see ``gossipy_amd/ops``), with each model
family mapped to a node-batched HIP kernel.
"""

from __future__ import annotations

from collections import OrderedDict
from typing import Tuple

import torch
from torch.nn import Linear, Module, Sequential
from torch.nn.init import xavier_uniform_
from torch.nn.modules.activation import ReLU, Sigmoid

from . import TorchModel

__all__ = [
    "TorchPerceptron",
    "TorchMLP",
    "AdaLine",
    "LogisticRegression",
    "LinearRegression",
]


class TorchPerceptron(TorchModel):
    """Single-neuron perceptron with a configurable output activation
    (gossipy/model/nn.py:26-64)."""

    def __init__(self, dim: int, activation: type = Sigmoid, bias: bool = True):
        super().__init__()
        self.input_dim = dim
        self.model = Sequential(
            OrderedDict(
                {"linear": Linear(dim, 1, bias=bias), "sigmoid": activation()}
            )
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.model(x)

    def init_weights(self) -> None:
        xavier_uniform_(self.model._modules["linear"].weight)

    def __str__(self) -> str:
        return "TorchPerceptron(size=%d)\n%s" % (self.get_size(), self.model)


class TorchMLP(TorchModel):
    """MLP with uniform hidden activation and a linear output layer
    (gossipy/model/nn.py:67-113)."""

    def __init__(
        self,
        input_dim: int,
        output_dim: int,
        hidden_dims: Tuple[int, ...] = (100,),
        activation: type = ReLU,
    ):
        super().__init__()
        dims = [input_dim] + list(hidden_dims)
        layers: "OrderedDict[str, Module]" = OrderedDict()
        for i in range(len(dims) - 1):
            layers["linear_%d" % (i + 1)] = Linear(dims[i], dims[i + 1])
            layers["activ_%d" % (i + 1)] = activation()
        layers["linear_%d" % len(dims)] = Linear(dims[-1], output_dim)
        self.model = Sequential(layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.model(x)

    def init_weights(self) -> None:
        for m in self.model.modules():
            if isinstance(m, Linear):
                xavier_uniform_(m.weight)

    def __str__(self) -> str:
        return "%s(size=%d)\n%s" % (self.__class__.__name__, self.get_size(), self.model)


class AdaLine(TorchModel):
    """Single weight vector with linear response; trained by hand-written
    delta-rule / Pegasos updates, never autograd (gossipy/model/nn.py:116-143).

    In the batched engine this family maps to the rank-1 update kernels
    (``pegasos_update`` / ``adaline_update`` in ``ops/hip/gossip_kernels.hip``).
    """

    def __init__(self, dim: int):
        super().__init__()
        self.input_dim = dim
        self.model = torch.nn.Parameter(torch.zeros(dim), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.model @ x.T

    def get_size(self) -> int:
        return self.input_dim

    def init_weights(self) -> None:
        pass


class LogisticRegression(TorchModel):
    """``sigmoid(Linear(in, out))`` (gossipy/model/nn.py:147-174).

    The flagship benchmark model; maps to the fused forward/backward/SGD
    LogReg kernel in the batched engine.
    """

    def __init__(self, input_dim: int, output_dim: int):
        super().__init__()
        self.model = torch.nn.Linear(input_dim, output_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.sigmoid(self.model(x))

    def init_weights(self) -> None:
        pass

    def __str__(self) -> str:
        return "LogisticRegression(in_size=%d, out_size=%d)" % (
            self.model.in_features,
            self.model.out_features,
        )


class LinearRegression(TorchModel):
    """Plain linear layer (gossipy/model/nn.py:176-198)."""

    def __init__(self, input_dim: int, output_dim: int):
        super().__init__()
        self.model = torch.nn.Linear(input_dim, output_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.model(x)

    def init_weights(self) -> None:
        pass

    def __str__(self) -> str:
        return "LinearRegression(in_size=%d, out_size=%d)" % (
            self.model.in_features,
            self.model.out_features,
        )
