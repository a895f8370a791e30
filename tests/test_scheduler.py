"""Unit tests for the DBS partition solver + scheduler state machine."""

import numpy as np
import pytest

from dynamic_load_balance_distributeddnn_amd.scheduler import (
    DBSScheduler, solve_partition)


def test_exact_sum_always():
    rng = np.random.default_rng(0)
    for _ in range(200):
        n = int(rng.integers(2, 9))
        B = int(rng.integers(n, 2048))
        times = rng.uniform(0.1, 10.0, n)
        fracs = rng.uniform(0.05, 1.0, n)
        fracs /= fracs.sum()
        out = solve_partition(times, fracs, B)
        assert out.sum() == B
        assert (out >= 1).all()


def test_proportional_to_speed():
    # rank 1 twice as slow per sample -> gets ~half the samples
    times = np.array([1.0, 2.0])
    fracs = np.array([0.5, 0.5])
    out = solve_partition(times, fracs, 512)
    assert out.sum() == 512
    # continuous target is (2/3, 1/3) of 512 = (341.3, 170.7)
    assert out[0] in (341, 342)
    assert out[1] in (170, 171)


def test_speed_estimate_uses_share():
    # rank 0 had 3x the samples and took 3x the time -> equal speed
    times = np.array([3.0, 1.0])
    fracs = np.array([0.75, 0.25])
    out = solve_partition(times, fracs, 100)
    assert abs(out[0] - out[1]) <= 1


def test_min_per_rank_floor():
    times = np.array([1e-3, 1e3, 1e3, 1e3])
    fracs = np.full(4, 0.25)
    out = solve_partition(times, fracs, 8)
    assert out.sum() == 8
    assert (out >= 1).all()


def test_deterministic_across_calls():
    times = np.array([1.1, 0.9, 1.3])
    fracs = np.array([0.4, 0.3, 0.3])
    a = solve_partition(times, fracs, 333)
    b = solve_partition(times, fracs, 333)
    assert (a == b).all()


def test_degenerate_times_fall_back_equal():
    out = solve_partition(np.zeros(4), np.full(4, 0.25), 64)
    assert (out == 16).all()


def test_too_small_batch_raises():
    with pytest.raises(ValueError):
        solve_partition(np.ones(8), np.full(8, 0.125), 4)


def test_scheduler_feedback_shifts_away_from_straggler():
    sched = DBSScheduler(world_size=4, global_batch=512)
    assert sched.batches.sum() == 512
    start = sched.batches.copy()
    # rank 3 consistently 2x slower
    for _ in range(5):
        times = np.array([1.0, 1.0, 1.0, 2.0])
        sched.step(times)
    assert sched.batches.sum() == 512
    assert sched.batches[3] < start[3]
    assert sched.batches[0] > start[0]
    # weights follow exact batch shares
    np.testing.assert_allclose(sched.weights.sum(), 1.0)


def test_scheduler_disabled_keeps_partition():
    sched = DBSScheduler(world_size=4, global_batch=100, enabled=False)
    before = sched.batches.copy()
    sched.step(np.array([1.0, 5.0, 1.0, 1.0]))
    assert (sched.batches == before).all()
    assert before.sum() == 100  # 25 each


# ------------------------------------------------------- property tests
try:
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=200, deadline=None)
    @given(
        times=st.lists(st.floats(min_value=1e-3, max_value=1e3,
                                 allow_nan=False), min_size=1, max_size=16),
        batch=st.integers(min_value=1, max_value=4096),
    )
    def test_solve_partition_properties(times, batch):
        """Exact sum, floor respected (when feasible), deterministic,
        replicated-decision safe for ANY positive time vector."""
        from dynamic_load_balance_distributeddnn_amd.scheduler import \
            solve_partition

        n = len(times)
        t = np.asarray(times)
        frac = np.full(n, 1.0 / n)
        if batch < n:  # documented contract: every rank needs >= 1 sample
            with pytest.raises(ValueError):
                solve_partition(t, frac, batch)
            return
        out = solve_partition(t, frac, batch)
        assert out.sum() == batch
        assert out.dtype.kind == "i"
        assert (out >= 1).all()
        # determinism: same inputs -> same outputs (replicated decision)
        again = solve_partition(t.copy(), frac.copy(), batch)
        assert (out == again).all()

    @settings(max_examples=100, deadline=None)
    @given(
        ratio=st.floats(min_value=1.5, max_value=50.0),
        batch=st.integers(min_value=8, max_value=2048),
    )
    def test_slower_rank_gets_fewer(ratio, batch):
        from dynamic_load_balance_distributeddnn_amd.scheduler import \
            solve_partition

        t = np.array([1.0, float(ratio)])
        out = solve_partition(t, np.array([0.5, 0.5]), batch)
        assert out.sum() == batch
        assert out[1] <= out[0]
except ImportError:  # pragma: no cover - hypothesis always in this image
    pass
