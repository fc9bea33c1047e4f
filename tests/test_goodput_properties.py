"""Property-based goodput-model tests (hypothesis).

Deepens the invariant coverage of tests/test_goodput.py (reference
strategy: goodput_test.py's evaluate/optimize invariants) by sampling
model parameters and configurations instead of fixing them:

- optimize() returns configs that respect max_batch_size, the local
  batch-size bounds, and accumulation settings, and whose goodput
  equals evaluate() at that config (self-consistency);
- efficiency is in (0, 1] and non-increasing in batch size;
- throughput never increases when the batch grows at fixed replicas.
"""

import numpy as np
from hypothesis import given, settings, strategies as st

from adaptdl_amd.goodput import GoodputFunction, GradParams, PerfParams

# Parameter magnitudes bracketing the realistic fitted values
# (BASELINE.md's PerfParams(0.121, 0.00568, 0.0236, 0.00634, 0.0118,
# 0.00317, 1.14) oracle).
_pos = st.floats(min_value=1e-4, max_value=0.5, allow_nan=False)
_gamma = st.floats(min_value=1.0, max_value=2.0, allow_nan=False)
_grad = st.floats(min_value=1e-5, max_value=0.1, allow_nan=False)


def _goodput_fn(params, grad, init_bsz=128):
    perf = PerfParams(*params)
    return GoodputFunction(perf, GradParams(*grad), init_bsz)


@settings(max_examples=40, deadline=None)
@given(params=st.tuples(_pos, _pos, _pos, _pos, _pos, _pos, _gamma),
       grad=st.tuples(_grad, _grad),
       replicas=st.integers(min_value=1, max_value=8),
       max_bsz=st.integers(min_value=256, max_value=8192))
def test_optimize_respects_constraints(params, grad, replicas, max_bsz):
    fn = _goodput_fn(params, grad)
    goodput, atomic_bsz, accum_steps = fn.optimize(
        1, replicas, max_batch_size=max_bsz,
        atomic_bsz_range=(32, 1024), accumulation=True)
    # reference semantics: atomic_bsz = ceil(target / replicas /
    # (accum+1)), so the realized total may exceed the cap by the
    # ceiling slack (< one sample per replica per accum group).
    total = replicas * atomic_bsz * (accum_steps + 1)
    slack = replicas * (accum_steps + 1) - 1
    assert total <= max(max_bsz, 128) + slack
    assert 32 <= atomic_bsz <= 1024
    assert accum_steps >= 0
    # self-consistency: reported goodput == evaluate at that config
    ref = fn.evaluate(1, replicas, atomic_bsz, accum_steps)
    assert np.isclose(goodput, ref, rtol=1e-6)


@settings(max_examples=40, deadline=None)
@given(params=st.tuples(_pos, _pos, _pos, _pos, _pos, _pos, _gamma),
       grad=st.tuples(_grad, _grad))
def test_efficiency_monotone_in_batch(params, grad):
    fn = _goodput_fn(params, grad)
    sizes = np.array([128, 256, 512, 1024, 2048, 4096, 8192])
    eff = fn.efficiency(sizes)
    assert np.all(eff > 0) and np.all(eff <= 1.0 + 1e-9)
    assert np.all(np.diff(eff) <= 1e-9)    # non-increasing


@settings(max_examples=40, deadline=None)
@given(params=st.tuples(_pos, _pos, _pos, _pos, _pos, _pos, _gamma),
       replicas=st.integers(min_value=1, max_value=8))
def test_throughput_decreasing_in_atomic_bsz_time(params, replicas):
    """Larger atomic batches can only slow a single step down (the
    throughput model's accum/optim time is affine increasing in bsz),
    so samples/s per step-time unit must stay finite and positive and
    step TIME must be non-decreasing."""
    fn = _goodput_fn(params, (1e-3, 1e-3))
    bszs = np.array([128, 256, 512, 1024])
    thr = fn.throughput(1, replicas, bszs, np.zeros_like(bszs))
    assert np.all(np.isfinite(thr)) and np.all(thr > 0)
    step_time = replicas * bszs / thr
    assert np.all(np.diff(step_time) >= -1e-9)


@settings(max_examples=25, deadline=None)
@given(params=st.tuples(_pos, _pos, _pos, _pos, _pos, _pos, _gamma),
       grad=st.tuples(_grad, _grad))
def test_optimize_beats_grid(params, grad):
    """optimize() must find a config at least as good as a coarse grid
    of valid alternatives, up to its 50-point geomspace discretization
    (off-sample grid points can win by up to ~10%)."""
    fn = _goodput_fn(params, grad)
    best, atomic_bsz, accum = fn.optimize(
        1, 4, max_batch_size=4096, atomic_bsz_range=(32, 1024),
        accumulation=True)
    for bsz in (32, 64, 128, 256, 512, 1024):
        for acc in (0, 1, 3):
            if 4 * bsz * (acc + 1) > 4096 or 4 * bsz * (acc + 1) < 128:
                continue
            alt = fn.evaluate(1, 4, bsz, acc)
            assert best >= alt * 0.9


# ---- fit_perf_params optimistic-freeze properties:
# the degenerate single-config profiles are exactly where a bad
# extrapolation would steer the 8-GPU replica choice, so pin the
# freeze semantics down.

from adaptdl_amd.goodput import fit_perf_params  # noqa: E402


def _times(params, nodes, replicas, bsz):
    """Synthesize observed (accum, optim) step times from a true model."""
    fn = _goodput_fn(params, (1e-3, 1e-3))
    accum = params[0] + params[1] * bsz
    total = bsz * replicas / fn.throughput(nodes, replicas, bsz, 0)
    return accum, total


@settings(max_examples=25, deadline=None)
@given(alpha=st.floats(1e-3, 0.2), beta=st.floats(1e-5, 1e-3),
       bsz=st.integers(32, 1024))
def test_fit_single_bsz_freezes_optimistically(alpha, beta, bsz):
    """One observed atomic size at 1x1: alpha_c pins to half the mean
    accum time, so the model stays optimistic about raising the batch
    size (per-sample time falls) instead of extrapolating wildly."""
    t = alpha + beta * bsz
    fitted = fit_perf_params([1, 1], [1, 1], [bsz, bsz],
                             [t, t], [t * 1.05, t * 1.05])
    assert np.isclose(fitted.alpha_c, t / 2, rtol=1e-6)
    # The fit must reproduce the observed config closely (no wild
    # extrapolation baseline)...
    pred = fitted.alpha_c + fitted.beta_c * bsz
    assert abs(pred - t) / t < 0.15
    # ...and with alpha_c pinned to half the step, doubling the batch
    # is predicted sublinear relative to the fit's own step time.
    assert fitted.alpha_c + fitted.beta_c * 2 * bsz < 2 * pred + 1e-12
    assert np.all(np.isfinite(np.asarray(fitted)))


@settings(max_examples=25, deadline=None)
@given(alpha=st.floats(1e-3, 0.2), beta=st.floats(1e-5, 1e-3))
def test_fit_no_multireplica_data_predicts_scaleup_speedup(alpha, beta):
    """Profiles from a single replica only: the network terms freeze at
    ~zero, so the fitted model must predict a real speedup from adding
    replicas (the optimistic direction that makes the allocator try)."""
    bszs = np.array([64, 128, 256, 512])
    accum = alpha + beta * bszs
    optim = accum * 1.1
    fitted = fit_perf_params(np.ones(4), np.ones(4), bszs, accum, optim)
    fn = GoodputFunction(fitted, GradParams(1e-3, 1e-3), 128)
    t1 = fn.throughput(1, 1, 128, 0)
    t2 = fn.throughput(1, 2, 128, 0)
    assert t2 > 1.5 * t1
    # Inter-node prior: at least 10% worse than intra-node.
    assert fitted.alpha_n >= fitted.alpha_r * 1.1 - 1e-12
    assert fitted.beta_n >= fitted.beta_r * 1.1 - 1e-12


def test_fit_no_retrogression_data_freezes_beta():
    """No profile with replicas > 2: the per-replica retrogression
    slopes stay pinned (no invented penalty for scaling out)."""
    nodes = np.array([1, 1, 1])
    replicas = np.array([1, 2, 2])
    bszs = np.array([128, 128, 256])
    accum = 0.05 + 1e-4 * bszs
    optim = accum + 0.01
    fitted = fit_perf_params(nodes, replicas, bszs, accum, optim)
    assert fitted.beta_n <= 2e-8 * 1.1 + 1e-12
    assert fitted.beta_r <= 1e-8 + 1e-12
    assert np.all(np.isfinite(np.asarray(fitted)))


def test_fit_degenerate_identical_rows():
    """All-identical observations must not produce NaNs or a model that
    predicts slower-with-more-replicas at the observed point."""
    n = 4
    fitted = fit_perf_params(np.ones(n), np.ones(n),
                             np.full(n, 128), np.full(n, 0.05),
                             np.full(n, 0.06))
    assert np.all(np.isfinite(np.asarray(fitted)))
    fn = GoodputFunction(fitted, GradParams(1e-3, 1e-3), 128)
    assert fn.throughput(1, 2, 128, 0) >= fn.throughput(1, 1, 128, 0)
