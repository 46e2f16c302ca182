"""Property-based tests (hypothesis): tape invariants and slot-allocator
safety under arbitrary protocol/fault settings."""

import numpy as np
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from gossipy_amd.core import AntiEntropyProtocol, UniformDelay
from gossipy_amd.engine import EngineConfig, Purpose, RandomTape, Scheduler


@given(seed=st.integers(0, 2**63), purpose=st.sampled_from(list(Purpose)),
       t=st.integers(0, 10**6), n=st.integers(1, 64))
@settings(max_examples=50, deadline=None)
def test_tape_draws_in_range_and_reproducible(seed, purpose, t, n):
    tape = RandomTape(seed)
    a = tape.stream(purpose, t).random(n)
    b = tape.stream(purpose, t).random(n)
    a = np.atleast_1d(a)
    assert np.array_equal(a, np.atleast_1d(b))
    assert ((a >= 0) & (a < 1)).all()


@given(seed=st.integers(0, 2**31), lo=st.integers(-100, 100),
       width=st.integers(1, 1000), n=st.integers(1, 32))
@settings(max_examples=50, deadline=None)
def test_tape_integers_bounds(seed, lo, width, n):
    v = np.atleast_1d(RandomTape(seed).stream(Purpose.MISC, 3).integers(
        lo, lo + width, size=n))
    assert ((v >= lo) & (v < lo + width)).all()


@given(
    proto=st.sampled_from([AntiEntropyProtocol.PUSH,
                           AntiEntropyProtocol.PULL,
                           AntiEntropyProtocol.PUSH_PULL]),
    drop=st.floats(0, 0.9),
    online=st.floats(0.1, 1.0),
    dmax=st.integers(0, 25),
    seed=st.integers(0, 2**31),
)
@settings(max_examples=25, deadline=None)
def test_slot_allocator_never_double_frees(proto, drop, online, dmax, seed):
    """Across arbitrary drop/online/delay settings, every slot is freed at
    most once per allocation and the pool high-water stays bounded by the
    live-message count, never the cumulative message count."""
    cfg = EngineConfig(
        n_nodes=40, delta=8, protocol=proto, model_size=10,
        drop_prob=drop, online_prob=online,
        delay=UniformDelay(0, dmax), seed=seed,
    )
    s = Scheduler(cfg)
    for r in range(6):
        sched = s.next_round(r)
        # reuse queue must never contain duplicates (double free)
        q = [slot for _, slot in s._reuse_q]
        assert len(q) == len(set(q))
        # high-water bounded: in-flight + the reuse-lag window's worth of
        # retired slots (slots sit out SLOT_REUSE_LAG ticks before reuse)
        lag = Scheduler.SLOT_REUSE_LAG
        assert sched.n_slots <= 40 * (dmax + 2 + lag) * 2 + 64
    # delivered + failed == sent accounting closes over rounds with no
    # in-flight messages remaining after a drain round
