"""channels_last (NHWC) parameter layout through the gradient engine.

On MI355X the flagship conv workloads run with channels_last memory format
(MIOpen's igemm kernels are NHWC-native; NCHW runs insert transpose
kernels).  The GradSyncEngine must therefore hand autograd .grad views
whose strides match the parameter layout, while the flat bucket holds the
same bytes in storage order.  These tests check, on CPU, that gradients /
optimizer steps / GNS statistics are identical between an NCHW model and
its channels_last twin.
"""

import numpy as np
import pytest
import torch

from adaptdl_amd.torch._engine import GradSyncEngine, _segment_view
from adaptdl_amd.ops import _flat


class _Owner(object):
    def _local_stat(self, bucket, accum):
        pass

    def _accum_stat(self, bucket):
        pass

    def _total_stat(self, bucket, scale):
        if scale != 1.0:
            bucket.flat.mul_(scale)

    def _on_accum_done(self):
        pass

    def _on_sync_done(self):
        pass


def _small_convnet():
    torch.manual_seed(7)
    return torch.nn.Sequential(
        torch.nn.Conv2d(3, 8, 3, padding=1, bias=False),
        torch.nn.BatchNorm2d(8),
        torch.nn.ReLU(),
        torch.nn.Conv2d(8, 4, 1, bias=False),
        torch.nn.AdaptiveAvgPool2d(1),
        torch.nn.Flatten(),
        torch.nn.Linear(4, 2))


def test_segment_view_layouts():
    p = torch.randn(6, 10, 5, 5).to(memory_format=torch.channels_last)
    seg = torch.zeros(p.numel())
    v = _segment_view(seg, p)
    assert v.shape == p.shape
    assert v.stride() == p.stride()
    v.copy_(p)
    assert torch.equal(_flat(p), seg)
    # default-contiguous param keeps the plain view
    q = torch.randn(4, 3)
    assert _segment_view(torch.zeros(12), q).stride() == q.stride()


def test_channels_last_grads_match_nchw():
    model_a = _small_convnet()
    model_b = _small_convnet()
    model_b.load_state_dict(model_a.state_dict())
    model_b = model_b.to(memory_format=torch.channels_last)

    eng_a = GradSyncEngine([list(model_a.parameters())], _Owner())
    eng_b = GradSyncEngine([list(model_b.parameters())], _Owner())

    x = torch.randn(4, 3, 8, 8)
    for model, eng, inp in ((model_a, eng_a, x),
                            (model_b, eng_b,
                             x.contiguous(memory_format=torch.channels_last))):
        eng.zero_grad()
        out = model(inp)
        out.pow(2).sum().backward()

    for (pa, _, _), (pb, _, _) in zip(
            (s for b in eng_a.buckets for s in b.segments),
            (s for b in eng_b.buckets for s in b.segments)):
        assert pa.shape == pb.shape
        assert torch.allclose(pa.grad, pb.grad, atol=1e-5), pa.shape
    eng_a.detach()
    eng_b.detach()


from conftest import elastic_multiprocessing


@elastic_multiprocessing
def _run_full_adp_step():
    import adaptdl_amd.collective as collective
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch import data as _data
    from adaptdl_amd.torch import epoch as _epoch

    collective.initialize()
    results = []
    for fmt in ("nchw", "cl"):
        torch.manual_seed(3)
        model = _small_convnet()
        if fmt == "cl":
            model = model.to(memory_format=torch.channels_last)
        optim = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
        adp = adl.AdaptiveDataParallel(model, optim,
                                       name="cltest-" + fmt)
        xs = torch.randn(32, 3, 8, 8)
        ys = torch.randint(0, 2, (32,))
        loader = adl.AdaptiveDataLoader(
            torch.utils.data.TensorDataset(xs, ys), batch_size=16)
        for epoch in adl.remaining_epochs_until(1):
            for x, y in loader:
                if fmt == "cl":
                    x = x.contiguous(memory_format=torch.channels_last)
                optim.zero_grad()
                loss = torch.nn.functional.cross_entropy(adp(x), y)
                loss.backward()
                optim.step()
        results.append({k: v.detach().clone().contiguous()
                        for k, v in model.state_dict().items()})
        adp.gns.engine.detach()
        # reset module-level loader/epoch state between the two runs
        _data.AdaptiveDataLoaderHelper._current = None
        _data.AdaptiveDataLoaderHelper._training = None
        _data.AdaptiveDataLoaderHelper._position.clear()
        if _epoch._EPOCH_STATE is not None:
            _epoch._EPOCH_STATE.finished_epochs = 0
            _epoch._EPOCH_STATE.current_epoch = None

    for k in results[0]:
        assert torch.allclose(results[0][k], results[1][k],
                              atol=1e-5), k
    collective.teardown()


def test_channels_last_full_adp_step():
    """Two SGD steps via the full ADP stack: NCHW vs channels_last equal."""
    _run_full_adp_step()


def test_weight_cast_cache_invalidation():
    """FusedConv2d caches its bf16 weight cast per optimizer cycle;
    the cache must refresh on a cycle-serial bump (fused optimizers)
    and on an in-place weight mutation (stock optimizers)."""
    from adaptdl_amd.torch import _engine
    from adaptdl_amd.torch.layers import FusedConv2d

    conv = FusedConv2d(8, 8, 3, padding=1, bias=False)
    wb1 = conv._cast_weight()
    assert conv._cast_weight() is wb1  # same cycle: cached

    _engine._cycle_serial += 1         # new optimizer cycle
    wb2 = conv._cast_weight()
    assert wb2 is not wb1

    with torch.no_grad():
        conv.weight.add_(1.0)          # stock-optimizer-style update
    wb3 = conv._cast_weight()
    assert wb3 is not wb2
    assert torch.allclose(wb3.float(),
                          conv.weight.detach().to(torch.bfloat16)
                          .float().contiguous(
                              memory_format=torch.channels_last))
