import numpy as np
import pytest

from adaptdl_amd.goodput import (GoodputFunction, GradParams, PerfParams,
                                 fit_perf_params, _obj_and_grad)

# Realistic fitted parameters from a profiling run, used as an oracle by the
# reference test suite (reference: sched/.../pollux_test.py:33-38).
PERF_PARAMS = PerfParams(0.121, 0.00568, 0.0236, 0.00634,
                         0.0118, 0.00317, 1.14)
GRAD_PARAMS = GradParams(sqr=0.00136, var=0.000502)


@pytest.fixture
def fn():
    return GoodputFunction(PERF_PARAMS, GRAD_PARAMS, 128)


def test_evaluate_positive_and_scalar(fn):
    val = fn.evaluate(1, 1, 128, 0)
    assert np.isscalar(val) or val.shape == ()
    assert val > 0


def test_throughput_monotonic_in_bsz(fn):
    bszs = np.array([128, 256, 512, 1024])
    thr = fn.throughput(1, 1, bszs, 0)
    assert np.all(np.diff(thr) > 0)


def test_efficiency_decreasing_in_bsz(fn):
    bszs = np.array([128, 256, 512, 1024, 4096])
    eff = fn.efficiency(bszs)
    assert np.all(np.diff(eff) < 0)
    assert np.all(eff <= 1.0 + 1e-9)


def test_evaluate_equals_throughput_times_efficiency(fn):
    for r in (1, 2, 4, 8):
        v = fn.evaluate(1, r, 128, 0)
        t = fn.throughput(1, r, 128, 0)
        e = fn.efficiency(r * 128)
        assert np.isclose(v, t * e)


def test_optimize_respects_bounds(fn):
    goodput, atomic_bsz, accum_steps = fn.optimize(
        1, 4, max_batch_size=4096, atomic_bsz_range=(32, 1024))
    assert 32 <= atomic_bsz <= 1024
    assert accum_steps == 0
    assert goodput > 0


def test_optimize_accumulation(fn):
    goodput, atomic_bsz, accum_steps = fn.optimize(
        1, 1, max_batch_size=4096, atomic_bsz_range=(32, 128),
        accumulation=True)
    total = atomic_bsz * (accum_steps + 1)
    assert accum_steps >= 1 or total == 128
    assert atomic_bsz <= 128


def test_optimize_single_replica_accum_floor(fn):
    # With one replica and a scaled-up batch, accum_steps must be >= 1.
    _, atomic_bsz, accum_steps = fn.optimize(
        1, 1, max_batch_size=4096, atomic_bsz_range=(32, 4096),
        accumulation=True)
    if atomic_bsz * (accum_steps + 1) > 128:
        assert accum_steps >= 1


def test_optimize_vectorized_matches_scalar(fn):
    nodes = np.ones(4, dtype=int)
    replicas = np.array([1, 2, 4, 8])
    g_vec, bsz_vec, acc_vec = fn.optimize(
        nodes, replicas, max_batch_size=4096, atomic_bsz_range=(32, 1024))
    for i, r in enumerate(replicas):
        g, b, a = fn.optimize(1, int(r), max_batch_size=4096,
                              atomic_bsz_range=(32, 1024))
        assert np.isclose(g, g_vec[i])
        assert b == bsz_vec[i]
        assert a == acc_vec[i]


def test_goodput_increases_with_replicas(fn):
    vals = [fn.optimize(1, r, max_batch_size=4096,
                        atomic_bsz_range=(32, 1024))[0]
            for r in (1, 2, 4, 8)]
    assert np.all(np.diff(vals) > 0)


def test_analytic_gradient_matches_finite_differences():
    rng = np.random.RandomState(0)
    num_nodes = np.array([1, 1, 1, 1, 2, 2], dtype=float)
    num_replicas = np.array([1, 2, 4, 8, 8, 16], dtype=float)
    atomic_bsz = np.array([128, 128, 256, 256, 512, 512], dtype=float)
    accum_t = np.abs(rng.rand(6)) + 0.1
    optim_t = accum_t + np.abs(rng.rand(6)) * 0.1
    for _ in range(5):
        x = np.concatenate([np.abs(rng.rand(6)) * 0.2 + 0.01,
                            [1.0 + rng.rand() * 5]])
        obj, grad = _obj_and_grad(x, num_nodes, num_replicas, atomic_bsz,
                                  accum_t, optim_t)
        eps = 1e-7
        for j in range(7):
            xp = x.copy()
            xp[j] += eps
            op, _ = _obj_and_grad(xp, num_nodes, num_replicas, atomic_bsz,
                                  accum_t, optim_t)
            fd = (op - obj) / eps
            assert abs(fd - grad[j]) < 1e-4 * max(1.0, abs(fd)), \
                "param {}: fd={} analytic={}".format(j, fd, grad[j])


def _model_times(params, num_nodes, num_replicas, atomic_bsz):
    from adaptdl_amd.goodput import (_predict_accum_time,
                                     _predict_network_time,
                                     _predict_log_optim_time)
    accum = _predict_accum_time(params, atomic_bsz)
    net = _predict_network_time(params, num_nodes, num_replicas)
    optim = np.exp(_predict_log_optim_time(params, accum, net))
    return accum, optim


def test_fit_recovers_model():
    # Generate data from a known model with small noise; the fit must
    # reproduce the model's *predictions* (parameters may trade off).
    rng = np.random.RandomState(1)
    true = PERF_PARAMS
    grid = [(1, 1, 128), (1, 2, 128), (1, 4, 128), (1, 8, 128),
            (1, 2, 256), (1, 4, 256), (1, 8, 512), (1, 8, 1024)]
    num_nodes = np.array([g[0] for g in grid], dtype=float)
    num_replicas = np.array([g[1] for g in grid], dtype=float)
    atomic_bsz = np.array([g[2] for g in grid], dtype=float)
    accum_t, optim_t = _model_times(true, num_nodes, num_replicas, atomic_bsz)
    noise = 1.0 + 0.01 * rng.randn(len(grid))
    fitted = fit_perf_params(num_nodes, num_replicas, atomic_bsz,
                             accum_t * noise, optim_t * noise)
    fa, fo = _model_times(fitted, num_nodes, num_replicas, atomic_bsz)
    assert np.allclose(fa, accum_t, rtol=0.15)
    assert np.allclose(fo, optim_t, rtol=0.15)


def test_fit_single_bsz_freezes_alpha_c():
    num_nodes = np.ones(4)
    num_replicas = np.array([1, 2, 4, 8], dtype=float)
    atomic_bsz = np.full(4, 128.0)
    accum_t = np.full(4, 0.85)
    optim_t = np.array([0.9, 1.0, 1.1, 1.2])
    fitted = fit_perf_params(num_nodes, num_replicas, atomic_bsz,
                             accum_t, optim_t)
    assert np.isclose(fitted.alpha_c, 0.425)
    # No multi-node data: inter-node params obey the 1.1x prior.
    assert fitted.alpha_n >= fitted.alpha_r * 1.1 - 1e-12
