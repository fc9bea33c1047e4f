"""Gradient-noise-scale numerics against hand-computed oracles.

The engine's bucketed statistics must reproduce the reference definitions:
  local_sqr  = sum over replicas+microbatches of ||g_microbatch||^2
  total_sqr  = || mean gradient ||^2
  grad_sqr   = (count * total - local) / (count - 1)
  grad_var   = (local - total) * scale / (count - 1)
(reference: adaptdl/torch/gradient_noise_scale.py:242-273)
"""

import numpy as np
import pytest
import torch

import adaptdl_amd.collective as collective
import adaptdl_amd.env as env

from conftest import elastic_multiprocessing


class _FakeADP:
    """Minimal stand-in for AdaptiveDataParallel in unit tests."""
    require_backward_grad_sync = True

    def _after_sync(self):
        pass


def _make_model_and_gns(world=1, lr=0.1):
    from adaptdl_amd.torch.gradient_noise_scale import GradientNoiseScale
    torch.manual_seed(0)
    model = torch.nn.Linear(4, 1, bias=False)
    optim = torch.optim.SGD(model.parameters(), lr=lr)
    adp = _FakeADP()
    gns = GradientNoiseScale(adp, optim, num_replicas=world)
    return model, optim, adp, gns


def test_engine_single_replica_differenced():
    model, optim, adp, gns = _make_model_and_gns(world=1)
    w = list(model.parameters())[0]
    xs = [torch.tensor([[1.0, 2.0, 0.5, -1.0]]),
          torch.tensor([[0.5, -1.0, 2.0, 1.0]])]
    grads = []
    for x in xs:
        out = model(x).sum()
        out.backward()
        g = w.grad.detach().clone().numpy().ravel()
        grads.append(g)
        gns.reset_accumulation()
    # For y = w.x, dL/dw = x; each step's grad == x.
    for g, x in zip(grads, xs):
        assert np.allclose(g, x.numpy().ravel())
    # After two steps the differenced estimator has updated:
    g1, g2 = grads
    local = (np.sum(g1 ** 2) + np.sum(g2 ** 2)) / 2
    total = np.sum(((g1 + g2) / 2) ** 2)
    grad_sqr = 2 * total - local
    grad_var = (local - total) * 2
    assert gns._state["biased"]
    assert np.isclose(gns._state["sqr_avg"][0], grad_sqr, rtol=1e-6)
    assert np.isclose(gns._state["var_avg"][0], grad_var, rtol=1e-6)


def test_engine_accumulation_stats():
    model, optim, adp, gns = _make_model_and_gns(world=1)
    w = list(model.parameters())[0]
    xs = [torch.tensor([[1.0, 2.0, 0.5, -1.0]]),
          torch.tensor([[0.5, -1.0, 2.0, 1.0]]),
          torch.tensor([[2.0, 0.0, -1.0, 0.5]])]
    # Three microbatches accumulated into one step.
    for i, x in enumerate(xs):
        gns.engine.require_sync = (i == len(xs) - 1)
        adp.require_backward_grad_sync = gns.engine.require_sync
        model(x).sum().backward()
    gs = [x.numpy().ravel() for x in xs]
    count = 3
    local = sum(np.sum(g ** 2) for g in gs) / count
    mean = sum(gs) / count
    total = np.sum(mean ** 2)
    # Gradient must equal the mean of microbatch gradients.
    assert np.allclose(w.grad.detach().numpy().ravel(), mean, rtol=1e-6)
    scale = gns.accum_scale * count
    grad_sqr = (count * total - local) / (count - 1)
    grad_var = (local - total) * scale / (count - 1)
    assert not gns._state["biased"]
    assert np.isclose(gns._state["sqr_avg"][0], grad_sqr, rtol=1e-5)
    assert np.isclose(gns._state["var_avg"][0], grad_var, rtol=1e-5)
    assert gns.should_zero_grad
    gns.reset_accumulation()
    assert np.allclose(w.grad.detach().numpy(), 0.0)


@elastic_multiprocessing
def _run_gns_two_replicas():
    import torch.distributed
    from adaptdl_amd.torch.gradient_noise_scale import GradientNoiseScale
    collective.initialize()
    if env.num_restarts() == 0:
        collective.teardown()
        return 2
    torch.distributed.init_process_group(
        "gloo", init_method="tcp://127.0.0.1:{}".format(
            collective.broadcast(_free_port())),
        world_size=env.num_replicas(), rank=env.replica_rank())
    torch.manual_seed(0)
    model = torch.nn.Linear(4, 1, bias=False)
    optim = torch.optim.SGD(model.parameters(), lr=0.1)
    adp = _FakeADP()
    gns = GradientNoiseScale(adp, optim)
    w = list(model.parameters())[0]
    data = {0: torch.tensor([[1.0, 2.0, 0.5, -1.0]]),
            1: torch.tensor([[0.5, -1.0, 2.0, 1.0]])}
    x = data[env.replica_rank()]
    model(x).sum().backward()
    g0 = data[0].numpy().ravel()
    g1 = data[1].numpy().ravel()
    mean = (g0 + g1) / 2
    # Synchronized gradient must be the replica mean.
    assert np.allclose(w.grad.detach().numpy().ravel(), mean, atol=1e-6)
    count = 2
    local = (np.sum(g0 ** 2) + np.sum(g1 ** 2)) / count
    total = np.sum(mean ** 2)
    grad_sqr = count * total - local
    grad_var = (local - total) * 2
    assert np.isclose(gns._state["sqr_avg"][0], grad_sqr, rtol=1e-5)
    assert np.isclose(gns._state["var_avg"][0], grad_var, rtol=1e-5)
    torch.distributed.destroy_process_group()
    collective.teardown()
    return 0


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_gns_two_replicas():
    _run_gns_two_replicas()


def test_gain_formula():
    model, optim, adp, gns = _make_model_and_gns(world=1)
    gns._state["sqr_avg"] = np.array([0.5])
    gns._state["var_avg"] = np.array([2.0])
    expected = (2.0 + 0.5) / (2.0 / 4 + 0.5)
    assert np.isclose(gns.gain(4.0), expected)


def test_nan_gradient_skipped():
    model, optim, adp, gns = _make_model_and_gns(world=1)
    sqr_before = gns._state["sqr_avg"].copy()
    x = torch.tensor([[float("nan"), 1.0, 1.0, 1.0]])
    model(x).sum().backward()
    assert np.allclose(gns._state["sqr_avg"], sqr_before)


def test_precond_sqsum_dev_cpu_fallback_matches():
    """Device-scalar preconditioned sqsum: CPU fallback math matches the
    host-scalar variant in both modes."""
    from adaptdl_amd import ops as _ops
    torch.manual_seed(3)
    g = torch.randn(1001)
    v = torch.rand(1001)
    beta2, eps, step = 0.999, 1e-8, 17
    pc = torch.zeros(4)
    _ops.set_precond_scalars(pc, beta2, eps, step)
    out_dev = torch.zeros((), dtype=torch.float64)
    _ops.precond_sqsum_dev(g, v, pc, out_dev)
    out_ref = torch.zeros((), dtype=torch.float64)
    _ops.precond_sqsum(g, v, beta2, eps, step, out_ref)
    assert torch.allclose(out_dev, out_ref, rtol=1e-6)
    # Warmup gate -> identity preconditioner -> plain sqsum.
    _ops.set_precond_scalars(pc, beta2, eps, 1)
    out_id = torch.zeros((), dtype=torch.float64)
    _ops.precond_sqsum_dev(g, v, pc, out_id)
    assert torch.allclose(out_id, g.double().pow(2).sum(), rtol=1e-12)


def test_adam_precond_stats_one_launch_per_bucket(tmp_ckpt_env):
    """With FusedAdam, the Adam-preconditioned statistics pass must use
    the whole-bucket device-scalar kernel (O(buckets) launches), not the
    per-segment fallback (VERDICT r1 task 7)."""
    from unittest import mock
    import adaptdl_amd.torch as adl
    from adaptdl_amd import ops as _ops

    if not collective.initialized():
        collective.initialize(master_addr="127.0.0.1")
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))  # 4 params
    optim = adl.FusedAdam(model.parameters(), lr=1e-3)
    adp = adl.AdaptiveDataParallel(model, optim, name="adam-batched")
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(torch.randn(64, 8),
                                       torch.randint(0, 4, (64,))),
        batch_size=16)

    dev_calls = []
    seg_calls = []
    real_dev = _ops.precond_sqsum_dev
    real_seg = _ops.precond_sqsum
    with mock.patch.object(_ops, "precond_sqsum_dev",
                           side_effect=lambda *a: (dev_calls.append(1),
                                                   real_dev(*a))), \
         mock.patch.object(_ops, "precond_sqsum",
                           side_effect=lambda *a: (seg_calls.append(1),
                                                   real_seg(*a))):
        # Re-import the names inside gradient_noise_scale? Not needed:
        # it calls through the ops module attribute.
        for _epoch in adl.remaining_epochs_until(2):
            for x, y in loader:
                optim.zero_grad()
                torch.nn.functional.cross_entropy(adp(x), y).backward()
                optim.step()
    n_buckets = len(adp.gns.engine.buckets)
    assert n_buckets >= 1
    # After the first optimizer step the flat Adam state exists, so all
    # statistics launches take the one-per-bucket fast path.
    assert len(dev_calls) > 0
    assert not seg_calls, "per-segment fallback used with FusedAdam"
    collective.teardown()


import pytest as _pytest


@_pytest.mark.parametrize("lp_dtype", [torch.bfloat16, torch.float16])
def test_bf16_parameter_model_trains(tmp_ckpt_env, lp_dtype):
    """True-low-precision-parameter model through the engine:
    statistics run on bf16/fp16 buckets; fused
    optimizers refuse clearly."""
    import pytest
    import adaptdl_amd.torch as adl

    if not collective.initialized():
        collective.initialize(master_addr="127.0.0.1")
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4)).to(lp_dtype)
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    adp = adl.AdaptiveDataParallel(
        model, optim,
        name="lp-model-{}".format(str(lp_dtype).split(".")[-1]))
    assert all(b.flat.dtype == lp_dtype
               for b in adp.gns.engine.buckets)
    xs = torch.randn(64, 8).to(lp_dtype)
    ys = torch.randint(0, 4, (64,))
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(xs, ys), batch_size=16)
    for _epoch in adl.remaining_epochs_until(2):
        for x, y in loader:
            optim.zero_grad()
            loss = torch.nn.functional.cross_entropy(
                adp(x).float(), y)
            loss.backward()
            optim.step()
    assert np.isfinite(adp.gns.sqr_avg())
    assert np.isfinite(adp.gns.var_avg())
    for p in model.parameters():
        assert torch.isfinite(p).all()

    # Fused optimizers state their fp32 requirement up front.
    model2 = torch.nn.Linear(4, 2).to(lp_dtype)
    optim2 = adl.FusedSGD(model2.parameters(), lr=0.1)
    with pytest.raises(ValueError, match="float32"):
        adl.AdaptiveDataParallel(model2, optim2, name="bf16-fused")
    collective.teardown()
