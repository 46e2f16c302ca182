"""Counter-based random tape for the batched engine.

The object layer consumes numpy's *global* RNG in node-iteration order
(gossipy/simul.py:390-409), which makes results depend on how nodes are
scheduled. The batched engine instead derives every random decision from a
counter-based splitmix64 stream keyed on ``(seed, purpose, t, extra)``, so
the full event schedule — peer choices, drop/online coin flips, delays,
timeout offsets, evaluation samples — is a pure function of the seed. Every
rank replays the same tape, making the schedule *identical for any GPU
count and any node->GPU residency map* (SURVEY.md §7 hard-part 3).

The generator is splitmix64 (public-domain finalizer constants), chosen
over numpy's Philox because the exact draw sequence must be reproducible
bit-for-bit by the native C++ scheduler (``csrc/scheduler.cpp``) — both
implementations share these few lines of integer arithmetic, which numpy
replicates exactly with uint64 wraparound ops.
"""

from __future__ import annotations

from enum import IntEnum

import numpy as np

__all__ = ["Purpose", "RandomTape", "TapeStream"]

_GOLDEN = np.uint64(0x9E3779B97F4A7C15)
_MIX1 = np.uint64(0xBF58476D1CE4E5B9)
_MIX2 = np.uint64(0x94D049BB133111EB)
_U53_INV = 1.0 / 9007199254740992.0  # 2^-53
_M64 = (1 << 64) - 1


def _splitmix64_int(x: int) -> int:
    """Scalar splitmix64 on python ints — bit-identical to the vectorized
    numpy version but ~10x faster for single draws (numpy scalar uint64
    ops dominate the tokenized scheduler otherwise)."""
    x = (x + 0x9E3779B97F4A7C15) & _M64
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & _M64
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & _M64
    return x ^ (x >> 31)


def _splitmix64(x: np.ndarray) -> np.ndarray:
    """Vectorized splitmix64 finalizer (uint64 in, uint64 out)."""
    x = (x + _GOLDEN).astype(np.uint64)
    x = ((x ^ (x >> np.uint64(30))) * _MIX1).astype(np.uint64)
    x = ((x ^ (x >> np.uint64(27))) * _MIX2).astype(np.uint64)
    return (x ^ (x >> np.uint64(31))).astype(np.uint64)


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
    PART = 10  #: partition-id draws for partitioned gossip (Hegedus 2021)
    SAMPLE = 11  #: parameter-sample seeds for sampled gossip (Hegedus 2021)


class TapeStream:
    """Sequential draws from one keyed stream: draw ``i`` is
    ``splitmix64(key + i)``; the instance keeps a running offset."""

    __slots__ = ("key", "_off")

    def __init__(self, key: int):
        self.key = np.uint64(key)
        self._off = 0

    def _raw(self, n: int) -> np.ndarray:
        idx = np.arange(self._off, self._off + n, dtype=np.uint64)
        self._off += n
        return _splitmix64(self.key + idx)

    def _raw1(self) -> int:
        v = _splitmix64_int((int(self.key) + self._off) & _M64)
        self._off += 1
        return v

    def random(self, n: int = 1):
        """``n`` float64 uniforms in [0, 1) (53-bit)."""
        if n == 1:
            return float(self._raw1() >> 11) * _U53_INV
        u = (self._raw(n) >> np.uint64(11)).astype(np.float64) * _U53_INV
        return u

    def integers(self, low: int, high: int, size: int = 1):
        """``size`` ints in [low, high) (scaled-double method — tiny bias,
        chosen for trivial cross-language reproducibility)."""
        if size == 1:
            u = float(self._raw1() >> 11) * _U53_INV
            return low + int(np.floor(u * (high - low)))
        u = (self._raw(size) >> np.uint64(11)).astype(np.float64) * _U53_INV
        v = low + np.floor(u * (high - low)).astype(np.int64)
        return v

    def uniform(self, low: float, high: float, size) -> np.ndarray:
        """Uniform floats in [low, high); ``size`` may be a tuple."""
        n = int(np.prod(size))
        u = (self._raw(n) >> np.uint64(11)).astype(np.float64) * _U53_INV
        return (low + u * (high - low)).reshape(size)

    def normal(self, mu: float, sigma: float, size: int) -> np.ndarray:
        """Box–Muller normals (pairs of uniforms; cos branch only)."""
        u = (self._raw(2 * size) >> np.uint64(11)).astype(np.float64) * _U53_INV
        u1 = np.maximum(u[0::2], 1e-300)
        u2 = u[1::2]
        z = np.sqrt(-2.0 * np.log(u1)) * np.cos(2.0 * np.pi * u2)
        return mu + sigma * z


def sample_indices(seed: int, count: int, D: int) -> np.ndarray:
    """``count`` coordinate indices in [0, D) with replacement for a sampled
    merge (K6): draw ``j`` is ``splitmix64(seed + j) % D``. The HIP kernel
    computes the identical sequence device-side (gossip_kernels.hip,
    samp_merge)."""
    raw = _splitmix64(np.uint64(seed) + np.arange(count, dtype=np.uint64))
    return (raw % np.uint64(D)).astype(np.int64)


class RandomTape:
    """Deterministic per-(purpose, timestep, extra) streams."""

    def __init__(self, seed: int):
        self.seed = int(seed) & 0xFFFFFFFFFFFFFFFF

    def stream_key(self, purpose: Purpose, t: int = 0, extra: int = 0) -> int:
        k = _splitmix64_int(self.seed ^ int(purpose))
        k = _splitmix64_int(k ^ int(t))
        k = _splitmix64_int(k ^ int(extra))
        return k

    def stream(self, purpose: Purpose, t: int = 0, extra: int = 0) -> TapeStream:
        return TapeStream(self.stream_key(purpose, t, extra))

    # -- convenience draws used by the scheduler -----------------------------

    def uniform_ints(
        self, purpose: Purpose, t: int, n: int, low: int, high: int
    ) -> np.ndarray:
        """``n`` ints in ``[low, high)``."""
        return np.atleast_1d(self.stream(purpose, t).integers(low, high, size=n))

    def uniform(self, purpose: Purpose, t: int, n: int) -> np.ndarray:
        """``n`` floats in ``[0, 1)``."""
        return np.atleast_1d(self.stream(purpose, t).random(n))
