"""Token-account flow control (Danner 2018 / Hegedus 2021).

Parity layer for the reference's ``gossipy/flow_control.py:22-236``. Token
accounts are pure host-side control plane in the batched engine too: the
per-node balances live in a numpy array there, but the strategy math is the
same as these classes.
"""

from __future__ import annotations

from abc import ABC, abstractmethod

from numpy.random import binomial

__all__ = [
    "TokenAccount",
    "PurelyProactiveTokenAccount",
    "PurelyReactiveTokenAccount",
    "SimpleTokenAccount",
    "GeneralizedTokenAccount",
    "RandomizedTokenAccount",
]


class TokenAccount(ABC):
    """A token balance plus a (proactive, reactive) sending strategy.

    ``proactive()`` returns the probability of sending when the node times
    out; ``reactive(utility)`` returns how many messages to send in reaction
    to an incoming message (gossipy/flow_control.py:22-82).
    """

    def __init__(self):
        self.n_tokens = 0

    def add(self, n: int = 1) -> None:
        """Add ``n`` tokens."""
        self.n_tokens += n

    def sub(self, n: int = 1) -> None:
        """Remove ``n`` tokens (floored at zero)."""
        self.n_tokens = max(0, self.n_tokens - n)

    @abstractmethod
    def proactive(self) -> float:
        """Probability of proactively sending on timeout."""
        raise NotImplementedError

    @abstractmethod
    def reactive(self, utility: int) -> int:
        """Number of reaction messages for a received message of ``utility``."""
        raise NotImplementedError


class PurelyProactiveTokenAccount(TokenAccount):
    """Always send on timeout, never react — plain push gossip
    (gossipy/flow_control.py:85-102)."""

    def proactive(self) -> float:
        return 1

    def reactive(self, utility: int) -> int:
        return 0


class PurelyReactiveTokenAccount(TokenAccount):
    """Never proactive; every received message triggers ``utility * k`` sends
    (gossipy/flow_control.py:105-127)."""

    def __init__(self, k: int = 1):
        super().__init__()
        self.k = k

    def proactive(self) -> float:
        return 0

    def reactive(self, utility: int) -> int:
        return int(utility * self.k)


class SimpleTokenAccount(TokenAccount):
    """Proactive iff the balance reached capacity; reactive iff any token
    (gossipy/flow_control.py:130-154)."""

    def __init__(self, C: int = 1):
        super().__init__()
        assert C >= 1, "The capacity C must be strictly positive."
        self.capacity = C

    def proactive(self) -> float:
        return int(self.n_tokens >= self.capacity)

    def reactive(self, utility: int) -> int:
        return int(self.n_tokens > 0)


class GeneralizedTokenAccount(SimpleTokenAccount):
    """Reactive count ``floor((A-1+a)/A)``, halved without utility
    (gossipy/flow_control.py:157-189)."""

    def __init__(self, C: int, A: int):
        super().__init__(C)
        assert A >= 1, "The reactivity A must be positive."
        assert A <= C, "The capacity C must be >= the reactivity A."
        self.reactivity = A

    def reactive(self, utility: int) -> int:
        num = self.reactivity + self.n_tokens - 1
        return int(num / self.reactivity if utility > 0 else num / (2 * self.reactivity))


class RandomizedTokenAccount(GeneralizedTokenAccount):
    """Linear-ramp proactive probability and randomized-rounding reactive
    count (gossipy/flow_control.py:192-236)."""

    def proactive(self) -> float:
        if self.n_tokens < self.reactivity - 1:
            return 0
        if self.n_tokens <= self.capacity:
            return (self.n_tokens - self.reactivity + 1) / (
                self.capacity - self.reactivity + 1
            )
        return 1

    def reactive(self, utility: int, u: float = None) -> int:
        """``u``: optional uniform in [0,1) for the randomized rounding —
        the batched engine passes a tape draw so reactions are deterministic
        and residency-invariant; ``None`` falls back to numpy's global RNG
        (object-layer behavior, gossipy/flow_control.py:232-236)."""
        if utility > 0:
            r = self.n_tokens / self.reactivity
            frac = r - int(r)
            rounded = (u < frac) if u is not None else binomial(1, frac)
            return int(r) + int(rounded)  # randomized rounding
        return 0
