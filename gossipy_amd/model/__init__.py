"""Model abstractions: :class:`TorchModel` wraps a torch module with size
accounting and weight-init hooks (parity: gossipy/model/__init__.py:22-74)."""

from __future__ import annotations

from abc import ABC, abstractmethod

import torch
from torch.nn.modules.container import ParameterList

from .. import Sizeable

__all__ = ["TorchModel"]


class TorchModel(torch.nn.Module, Sizeable, ABC):
    """A torch module that knows its parameter count and how to re-init itself.

    ``get_size()`` (the parameter count) feeds message-size accounting; the
    batched engine uses it to size per-node arena slabs.
    """

    @abstractmethod
    def init_weights(self, *args, **kwargs) -> None:
        """(Re-)initialize the model weights."""
        raise NotImplementedError

    def get_size(self) -> int:
        """Total number of parameters."""
        return sum(p.numel() for p in self.parameters())

    def get_params_list(self) -> ParameterList:
        """Parameters as a :class:`torch.nn.ParameterList`
        (gossipy/model/__init__.py:65-74)."""
        return ParameterList(self.parameters())

    def __repr__(self) -> str:
        return str(self)

    def __str__(self) -> str:
        return "%s(size=%d)" % (self.__class__.__name__, self.get_size())

    def _get_n_params(self) -> int:
        """Reference-private alias (gossipy/model/__init__.py): parameter
        count — same value as :meth:`get_size`."""
        return sum(p.numel() for p in self.parameters())
