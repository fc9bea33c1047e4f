import numpy as np
import torch

from adaptdl_amd.torch.scaling_rules import (AdaScale, AdamScale,
                                             LinearScale, SqrtScale)


class _MockGNS:
    def __init__(self, sqr, var):
        self.raw_sqr_avg = np.asarray(sqr)
        self.raw_var_avg = np.asarray(var)
        self.accum_scale = 1.0
        self.accum_count = 1
        self.should_zero_grad = True

    def gain(self, scale):
        var = float(np.sum(np.maximum(self.raw_var_avg, 1e-6)))
        sqr = float(np.sum(np.maximum(self.raw_sqr_avg, 0.0)))
        return (var + sqr) / (var / scale + sqr)

    def get_progress(self):
        return 0.0

    def set_progress(self, p):
        pass

    def reset_accumulation(self):
        pass


class _MockADP:
    require_backward_grad_sync = True

    def __init__(self, gns):
        self.gns = gns


def test_adascale_lr_factor_vector():
    gns = _MockGNS([0.5, 1.0], [2.0, 1.0])
    rule = AdaScale()
    rule.adp = _MockADP(gns)
    scale = 4.0
    expected = (np.array([2.0, 1.0]) + np.array([0.5, 1.0])) / \
               (np.array([2.0, 1.0]) / scale + np.array([0.5, 1.0]))
    assert np.allclose(rule.scale_lr(scale), expected)


def test_adamscale_is_sqrt_of_adascale():
    gns = _MockGNS([0.5], [2.0])
    ada, adam = AdaScale(), AdamScale()
    ada.adp = adam.adp = _MockADP(gns)
    assert np.allclose(adam.scale_lr(9.0), np.sqrt(ada.scale_lr(9.0)))


def test_linear_and_sqrt():
    lin, sq = LinearScale(), SqrtScale()
    assert lin.scale_lr(16.0) == 16.0
    assert sq.scale_lr(16.0) == 4.0


def test_adascale_clamps_negative_var():
    gns = _MockGNS([1.0], [-5.0])
    rule = AdaScale()
    rule.adp = _MockADP(gns)
    # var clamped at 1e-6 => factor ~ 1.
    assert np.allclose(rule.scale_lr(8.0), (1e-6 + 1) / (1e-6 / 8 + 1))


def test_step_applies_scaled_lr():
    """The patched step must see lr * factor and restore lr afterwards."""
    from adaptdl_amd.torch.gradient_noise_scale import GradientNoiseScale

    class ADP:
        require_backward_grad_sync = True

        def _after_sync(self):
            pass

    torch.manual_seed(0)
    model = torch.nn.Linear(2, 1, bias=False)
    optim = torch.optim.SGD(model.parameters(), lr=1.0)
    adp = ADP()
    gns = GradientNoiseScale(adp, optim, num_replicas=1)
    adp.gns = gns
    rule = AdaScale()
    rule.initialize(adp, optim, patch_optimizer=True)
    # Fix GNS state so the expected factor is deterministic.
    gns._state["sqr_avg"] = np.array([1.0])
    gns._state["var_avg"] = np.array([3.0])
    w0 = model.weight.detach().clone()
    x = torch.tensor([[1.0, 2.0]])
    model(x).sum().backward()
    factor = (3.0 + 1.0) / (3.0 / 1.0 + 1.0)  # scale=1 => 1.0
    assert np.isclose(factor, 1.0)
    optim.step()
    # w = w0 - lr * factor * grad; grad == x
    assert torch.allclose(model.weight.detach(),
                          w0 - 1.0 * factor * x)
    assert optim.param_groups[0]["lr"] == 1.0  # restored
    progress = gns.get_progress()
    assert np.isclose(progress, gns.gain(1.0))
