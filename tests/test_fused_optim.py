"""Fused flat-bucket optimizers: numerics vs torch optimizers.

Attached through AdaptiveDataParallel, FusedSGD/FusedAdamW flatten
parameters into the gradient-bucket layout and step once per bucket;
training must be numerically identical (CPU fallback path here; the
GPU kernels are covered by tests/test_gpu_kernels.py numerics).
"""

import numpy as np
import pytest
import torch

import adaptdl_amd.checkpoint
import adaptdl_amd.collective
import adaptdl_amd.env
from conftest import elastic_multiprocessing


def _model(seed):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Conv2d(3, 8, 3, padding=1, bias=False),
        torch.nn.BatchNorm2d(8),
        torch.nn.ReLU(),
        torch.nn.Flatten(),
        torch.nn.Linear(8 * 16, 4))


@elastic_multiprocessing
def _run_fused_vs_plain(fused_cls, plain_cls, kwargs):
    import adaptdl_amd.torch as adl
    adaptdl_amd.collective.initialize("127.0.0.1")
    from adaptdl_amd.torch import data as _data
    from adaptdl_amd.torch import epoch as _epoch

    results = []
    for variant in ("plain", "fused"):
        model = _model(11)
        if variant == "plain":
            optim = plain_cls(model.parameters(), **kwargs)
        else:
            optim = fused_cls(model.parameters(), **kwargs)
        adp = adl.AdaptiveDataParallel(model, optim,
                                       name="fo-" + variant)
        xs = torch.randn(32, 3, 4, 4)
        ys = torch.randint(0, 4, (32,))
        loader = adl.AdaptiveDataLoader(
            torch.utils.data.TensorDataset(xs, ys), batch_size=8)
        for epoch in adl.remaining_epochs_until(2):
            for x, y in loader:
                optim.zero_grad()
                torch.nn.functional.cross_entropy(adp(x), y).backward()
                optim.step()
        results.append({k: v.detach().clone()
                        for k, v in model.state_dict().items()})
        adp.gns.engine.detach()
        _data.AdaptiveDataLoaderHelper._current = None
        _data.AdaptiveDataLoaderHelper._training = None
        _data.AdaptiveDataLoaderHelper._position.clear()
        if _epoch._EPOCH_STATE is not None:
            _epoch._EPOCH_STATE.finished_epochs = 0
            _epoch._EPOCH_STATE.current_epoch = None

    for k in results[0]:
        assert torch.allclose(results[0][k], results[1][k],
                              atol=1e-5), k
    adaptdl_amd.collective.teardown()
    return 0


def test_fused_sgd_matches_torch():
    from adaptdl_amd.torch.optim import FusedSGD
    _run_fused_vs_plain(FusedSGD, torch.optim.SGD,
                        dict(lr=0.05, momentum=0.9, weight_decay=1e-4))


def test_fused_adamw_matches_torch():
    from adaptdl_amd.torch.optim import FusedAdamW
    _run_fused_vs_plain(FusedAdamW, torch.optim.AdamW,
                        dict(lr=1e-3, weight_decay=0.01))


def test_fused_adam_matches_torch():
    from adaptdl_amd.torch.optim import FusedAdam
    _run_fused_vs_plain(FusedAdam, torch.optim.Adam,
                        dict(lr=1e-3))


@elastic_multiprocessing
def _run_fused_checkpoint_roundtrip():
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.optim import FusedSGD
    adaptdl_amd.collective.initialize("127.0.0.1")
    model = _model(13)
    optim = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim, name="fo-ckpt")
    xs = torch.randn(32, 3, 4, 4)
    ys = torch.randint(0, 4, (32,))
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(xs, ys), batch_size=8)
    for epoch in adl.remaining_epochs_until(4):
        for x, y in loader:
            optim.zero_grad()
            torch.nn.functional.cross_entropy(adp(x), y).backward()
            optim.step()
        if adaptdl_amd.env.num_restarts() == 0 and epoch == 1:
            adaptdl_amd.collective.teardown()
            adaptdl_amd.checkpoint.save_all_states()
            return 2
    # After restart the momentum buffers must live in the flat storage.
    for bucket in adp.gns.engine.buckets:
        assert bucket.sgd_momentum is not None
        assert float(bucket.sgd_momentum.abs().sum()) > 0
    assert np.isfinite(adp.gns.var_avg())
    adaptdl_amd.collective.teardown()
    return 0


def test_fused_optimizer_checkpoint_restart():
    _run_fused_checkpoint_roundtrip()
