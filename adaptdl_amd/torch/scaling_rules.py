"""Learning-rate scaling rules applied at each (scaled) optimizer step.

The optimizer's ``step``/``zero_grad`` are patched so user training loops
stay unchanged: ``step`` applies per-param-group LR factors from the active
rule, runs the original step, restores LRs, and advances scale-invariant
progress by the current gain; ``zero_grad`` defers to the GNS accumulation
bookkeeping.  Rules and semantics match the reference
(``/root/reference/adaptdl/adaptdl/torch/scaling_rules.py``).
"""

import functools
import math
import warnings
from types import MethodType

import numpy as np

from adaptdl_amd.torch.data import current_dataloader

__all__ = ["ScalingRuleBase", "AdaScale", "AdamScale", "LinearScale",
           "SqrtScale", "LEGWScale"]


class ScalingRuleBase(object):
    """Base class for scaling rules.

    Usage (normally done for you by AdaptiveDataParallel)::

        optim = torch.optim.SGD(model.parameters(), lr=0.001)
        model = AdaptiveDataParallel(model, optim, scaling_rule=AdaScale())
        ...
        loss.backward()
        optim.step()   # patched: applies the scaled learning rate
    """

    def __init__(self):
        self.adp = None
        self._optimizer = None
        self._orig_optimizer_step = None

    def scale_lr(self, scale):
        raise NotImplementedError

    def zero_grad(self, *args, **kwargs):
        if self.adp.gns.should_zero_grad:
            self.adp.gns.reset_accumulation()
        else:
            warnings.warn("skipping zero_grad for accumulated gradient")

    def step(self, *args, **kwargs):
        """Run one optimizer step with a scaled learning rate."""
        if not self.adp:
            raise ValueError("AdaptiveDataParallel instance is not set!")
        if not self.adp.require_backward_grad_sync:
            return
        scale = self.adp.gns.accum_scale * self.adp.gns.accum_count
        initial_lr = [pg["lr"] for pg in self._optimizer.param_groups]
        scaled_lr = np.multiply(self.scale_lr(scale), initial_lr)
        for lr, pg in zip(scaled_lr, self._optimizer.param_groups):
            pg["lr"] = lr
        self._orig_optimizer_step(*args, **kwargs)
        for lr, pg in zip(initial_lr, self._optimizer.param_groups):
            pg["lr"] = lr
        self.adp.gns.set_progress(self.adp.gns.get_progress()
                                  + self.adp.gns.gain(scale))

    def _patch_optimizer(self):
        @functools.wraps(self._optimizer.step)
        def step_wrapper(optim, *args, **kwargs):
            return self.step(*args, **kwargs)

        @functools.wraps(self._optimizer.zero_grad)
        def zero_wrapper(optim, *args, **kwargs):
            return self.zero_grad(*args, **kwargs)

        self._optimizer.step = MethodType(step_wrapper, self._optimizer)
        self._optimizer.zero_grad = MethodType(zero_wrapper, self._optimizer)

    def initialize(self, adp, optimizer, patch_optimizer=False):
        self.adp = adp
        self._optimizer = optimizer
        self._orig_optimizer_step = optimizer.step
        if patch_optimizer:
            self._patch_optimizer()


class AdaScale(ScalingRuleBase):
    """AdaScale (arxiv 2007.05105): per-group lr factor
    (var + sqr) / (var / scale + sqr)."""

    def scale_lr(self, scale):
        var = self.adp.gns.raw_var_avg
        sqr = self.adp.gns.raw_sqr_avg
        var = np.maximum(var, 1e-6)
        sqr = np.maximum(sqr, 0.0)
        return (var + sqr) / (var / scale + sqr)


class AdamScale(AdaScale):
    """AdaScale variant for Adam/AdamW/RMSprop: sqrt of the AdaScale gain."""

    def scale_lr(self, scale, power=0.5):
        return np.power(super().scale_lr(scale=scale), power)


class LinearScale(ScalingRuleBase):

    def scale_lr(self, scale):
        return scale


class SqrtScale(ScalingRuleBase):

    def scale_lr(self, scale):
        return math.sqrt(scale)


class LEGWScale(ScalingRuleBase):
    """LEGW (arxiv 1901.08256): sqrt(scale) with linear warmup over
    base_warmup_epochs * scale epochs of progress."""

    def __init__(self, base_warmup_epochs, data_size):
        super().__init__()
        self._base_warmup_epochs = base_warmup_epochs
        self._data_size = data_size

    def scale_lr(self, scale):
        dataloader = current_dataloader()
        total_steps = self._base_warmup_epochs * scale * \
            self._data_size / dataloader.batch_size
        max_lr_multiplier = math.sqrt(scale)
        progress = self.adp.gns.get_progress()
        if progress < total_steps:
            return max_lr_multiplier * (progress / total_steps)
        return max_lr_multiplier
