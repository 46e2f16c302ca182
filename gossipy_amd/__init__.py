"""gossipy_amd — an MI355X-native gossip-learning / decentralized-FL framework.

This package re-creates the full capability surface of the reference simulator
(gossipy: ``GossipSimulator`` / ``GossipNode`` / ``ModelHandler`` / ``DataDispatcher``,
anti-entropy protocols, token-account flow control, model sampling/partitioning,
non-IID partitioners, observer-based reporting, dill checkpointing) with a
from-scratch design built for one 8xMI355X box:

* an **object layer** (this package's top-level modules) that is API-compatible
  with the reference and runs anywhere (CPU included), and
* a **batched engine** (:mod:`gossipy_amd.engine`) that shards the simulated
  nodes across GPUs and executes every per-node compute site as a single
  node-batched CDNA4 HIP kernel against packed HBM parameter arenas, with
  cross-GPU gossip traffic carried by RCCL point-to-point over xGMI.

Reference parity notes cite the reference as ``file:line`` (e.g.
``gossipy/__init__.py:283-387`` for the Cache contract).
"""

from __future__ import annotations

import logging
import random
from abc import ABC, abstractmethod
from typing import Any, Dict, Tuple

import numpy as np
import torch

__version__ = "0.1.0"

__all__ = [
    "LOG",
    "CACHE",
    "set_seed",
    "CacheKey",
    "CacheItem",
    "Sizeable",
    "Cache",
    "GlobalSettings",
]


class Singleton(type):
    """Metaclass giving each class a single shared instance."""

    _instances: Dict[type, Any] = {}

    def __call__(cls, *args, **kwargs):
        if cls not in cls._instances:
            cls._instances[cls] = super().__call__(*args, **kwargs)
        return cls._instances[cls]


class GlobalSettings(metaclass=Singleton):
    """Global device selection (parity with gossipy/__init__.py:46-91).

    The object layer defaults to CPU like the reference; the batched engine
    manages its own devices/streams and ignores this setting.
    """

    _device = "cpu"

    def auto_device(self) -> torch.device:
        """Pick ``cuda`` (ROCm HIP under torch) when available, else ``cpu``."""
        self._device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
        return self._device

    def set_device(self, device_name: str) -> torch.device:
        """Set the device by name; ``"auto"`` defers to :meth:`auto_device`."""
        if device_name == "auto":
            return self.auto_device()
        self._device = torch.device(device_name)
        return self._device

    def get_device(self):
        """Return the currently selected device."""
        return self._device


class _OncePerMessageFilter(logging.Filter):
    """Suppress messages already emitted once (parity: DuplicateFilter,
    gossipy/__init__.py:94-103)."""

    def __init__(self):
        super().__init__()
        self._seen = set()

    def filter(self, record) -> bool:
        first = record.msg not in self._seen
        self._seen.add(record.msg)
        return first


def _make_logger() -> logging.Logger:
    logger = logging.getLogger("gossipy_amd")
    if not logger.handlers:
        try:
            from rich.logging import RichHandler

            handler: logging.Handler = RichHandler()
        except Exception:  # pragma: no cover - rich is present in the target env
            handler = logging.StreamHandler()
        handler.setFormatter(logging.Formatter("%(message)s"))
        logger.addHandler(handler)
        logger.setLevel(logging.INFO)
        logger.propagate = False
    logger.addFilter(_OncePerMessageFilter())
    return logger


LOG = _make_logger()
"""Framework logger with duplicate-message suppression."""


def set_seed(seed: int = 0) -> None:
    """Seed ``random``, ``numpy`` and ``torch`` (parity: gossipy/__init__.py:118-131)."""
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)


class Sizeable(ABC):
    """Interface for objects whose size (in atomic scalars) can be measured.

    Parity: gossipy/__init__.py:134-156. Sizes feed message-size accounting
    and the :class:`~gossipy_amd.core.LinearDelay` bandwidth model.
    """

    @abstractmethod
    def get_size(self) -> int:
        """Number of atomic scalars contained in the object."""
        raise NotImplementedError


def _size_of(value: Any, strict: bool) -> int:
    """Scalar-count of an arbitrary payload element."""
    if isinstance(value, Sizeable):
        return value.get_size()
    if isinstance(value, (float, int, bool)):
        return 1
    if strict:
        raise TypeError("Cannot compute the size of the payload!")
    LOG.warning("Impossible to compute the size of %s. Set to 0." % (value,))
    return 0


class CacheKey(Sizeable):
    """Hashable key identifying a cached model snapshot.

    The canonical key is ``(owner, n_updates)`` — see
    ``gossipy/model/handler.py:160-176``. ``get_size`` dereferences the global
    :data:`CACHE` so a message carrying only the key is accounted with the
    size of the model it points at (gossipy/__init__.py:177-183).
    """

    __slots__ = ("key",)

    def __init__(self, *args):
        self.key: Tuple[Any, ...] = tuple(args)

    def get(self) -> Tuple[Any, ...]:
        """Return the raw key tuple."""
        return self.key

    def get_size(self) -> int:
        val = CACHE[self]
        return _size_of(val, strict=False)

    def __repr__(self):
        return str(self.key)

    def __hash__(self):
        return hash(self.key)

    def __eq__(self, other: Any) -> bool:
        return isinstance(other, CacheKey) and self.key == other.key

    def __ne__(self, other: Any) -> bool:
        return not self.__eq__(other)


class CacheItem(Sizeable):
    """Reference-counted slot in the :class:`Cache`.

    Parity: gossipy/__init__.py:200-280 (created with one reference; a pop
    decrements; unreferenced items are evicted by the cache).
    """

    __slots__ = ("_value", "_refs")

    def __init__(self, value: Any):
        self._value = value
        self._refs = 1

    def add_ref(self) -> None:
        """Add one reference."""
        self._refs += 1

    def del_ref(self) -> Any:
        """Drop one reference and return the value."""
        self._refs -= 1
        return self._value

    def is_referenced(self) -> bool:
        """Whether at least one reference is outstanding."""
        return self._refs > 0

    def get(self) -> Any:
        """Return the stored value without touching the refcount."""
        return self._value

    def get_size(self) -> int:
        if isinstance(self._value, (tuple, list)):
            total = sum(_size_of(v, strict=False) for v in self._value if v is not None)
            return max(total, 1)
        return _size_of(self._value, strict=False)

    def __repr__(self):
        return repr(self._value)

    def __str__(self) -> str:
        return f"CacheItem({self._value})"


class Cache:
    """In-memory ref-counted store for in-flight model snapshots.

    This is the simulator's "wire": a message carries a :class:`CacheKey`
    and the receiver pops the snapshot from here. Contract parity with
    gossipy/__init__.py:283-387: ``push`` on an existing key adds a
    reference instead of overwriting; ``pop`` dereferences and auto-evicts
    at refcount zero; ``load``/``get_cache`` expose the raw dict for
    checkpointing.

    In the batched engine the same refcount semantics are applied to arena
    slots instead of Python objects (see :mod:`gossipy_amd.engine.arena`).
    """

    def __init__(self):
        self._cache: Dict[CacheKey, CacheItem] = {}

    def push(self, key: CacheKey, value: Any) -> None:
        """Store ``value`` under ``key`` (or add a reference if present)."""
        slot = self._cache.get(key)
        if slot is None:
            self._cache[key] = CacheItem(value)
        else:
            slot.add_ref()

    def pop(self, key: CacheKey) -> Any:
        """Retrieve the value for ``key``, dropping one reference.

        Returns ``None`` for unknown keys. The item is evicted when its
        refcount reaches zero.
        """
        slot = self._cache.get(key)
        if slot is None:
            return None
        value = slot.del_ref()
        if not slot.is_referenced():
            del self._cache[key]
        return value

    def clear(self) -> None:
        """Drop every cached item (a previously exported ``get_cache`` dict
        stays valid — the store is replaced, not emptied in place)."""
        self._cache = {}

    def __getitem__(self, key: CacheKey) -> Any:
        slot = self._cache.get(key)
        return slot.get() if slot is not None else None

    def load(self, cache_dict: Dict[CacheKey, Any]) -> None:
        """Replace the cache content (checkpoint restore)."""
        self._cache = cache_dict

    def get_cache(self) -> Dict[CacheKey, Any]:
        """Return the raw cache dict (checkpoint save)."""
        return self._cache

    def __len__(self) -> int:
        return len(self._cache)

    def __repr__(self):
        return str(self._cache)


CACHE = Cache()
"""Global model cache — the in-process transport for exchanged models."""


#: reference-name alias (gossipy/__init__.py:94-115)
DuplicateFilter = _OncePerMessageFilter

# make the layer modules reachable as package attributes, like the
# reference's ``import gossipy; gossipy.node`` usage
from . import core  # noqa: E402,F401
from . import data  # noqa: E402,F401
from . import flow_control  # noqa: E402,F401
from . import model  # noqa: E402,F401
from . import node  # noqa: E402,F401
from . import simul  # noqa: E402,F401
from . import utils  # noqa: E402,F401
