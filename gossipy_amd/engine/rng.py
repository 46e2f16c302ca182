"""Counter-based random tape for the batched engine.

The object layer consumes numpy's *global* RNG in node-iteration order
(gossipy/simul.py:390-409), which makes results depend on how nodes are
scheduled. The batched engine instead derives every random decision from a
Philox counter keyed on ``(seed, purpose, timestep)``, so the full event
schedule — peer choices, drop/online coin flips, delays, timeout offsets,
evaluation samples — is a pure function of the seed. Every rank replays the
same tape, which makes the schedule *identical for any GPU count and any
node->GPU residency map*: a 1-GPU run and an 8-GPU run execute the same
simulation event-for-event (SURVEY.md §7 hard-part 3).
"""

from __future__ import annotations

from enum import IntEnum

import numpy as np

__all__ = ["Purpose", "RandomTape"]


class Purpose(IntEnum):
    """Stream id for each kind of random decision in a round."""

    TIMEOUT = 0  #: per-node timeout offsets (simulation setup)
    PEER = 1  #: peer choice of firing nodes
    DROP = 2  #: per-message drop coin flip
    ONLINE = 3  #: per-node per-tick online mask
    DELAY = 4  #: per-message delay draw
    EVAL = 5  #: evaluation node sampling
    INIT = 6  #: model init
    DATA = 7  #: data shuffling / minibatch permutations
    TOKEN = 8  #: token-account proactive coin flips
    MISC = 9


class RandomTape:
    """Deterministic per-(purpose, timestep) random streams.

    Each call materializes a fresh ``np.random.Generator`` seeded by the
    Philox key ``(seed, purpose, t)`` — cheap (µs) and stateless, so any rank
    can draw any slice of the tape in any order and get identical values.
    """

    def __init__(self, seed: int):
        self.seed = int(seed)

    def stream(self, purpose: Purpose, t: int = 0, extra: int = 0) -> np.random.Generator:
        """The generator for ``(purpose, t, extra)`` (Philox 2x64 key)."""
        k0 = (self.seed << 8) ^ int(purpose)
        k1 = (int(t) << 20) ^ int(extra)
        key = np.array([k0 & 0xFFFFFFFFFFFFFFFF, k1 & 0xFFFFFFFFFFFFFFFF], dtype=np.uint64)
        return np.random.Generator(np.random.Philox(key=key))

    # -- convenience draws used by the scheduler -----------------------------

    def uniform_ints(
        self, purpose: Purpose, t: int, n: int, low: int, high: int
    ) -> np.ndarray:
        """``n`` ints in ``[low, high)``."""
        return self.stream(purpose, t).integers(low, high, size=n)

    def uniform(self, purpose: Purpose, t: int, n: int) -> np.ndarray:
        """``n`` floats in ``[0, 1)``."""
        return self.stream(purpose, t).random(n)
