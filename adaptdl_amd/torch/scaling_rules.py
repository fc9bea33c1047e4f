"""Learning-rate scaling rules applied at each (scaled) optimizer step.

The optimizer's ``step``/``zero_grad`` are taken over so user training
loops stay unchanged: ``step`` runs the original optimizer step inside a
context that multiplies each param group's LR by the active rule's
factor, then advances scale-invariant progress by the current gain;
``zero_grad`` defers to the GNS accumulation bookkeeping (it must NOT
clear gradients mid-accumulation).

Behavioral parity target (rule formulas + patching semantics):
``/root/reference/adaptdl/adaptdl/torch/scaling_rules.py`` — the code
here is an independent implementation; only the published rule math
(AdaScale/LEGW etc.) is shared.
"""

import contextlib
import math
import warnings

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

    #: Attributes wired up by :meth:`initialize`.
    adp = None
    _optimizer = None
    _orig_optimizer_step = None

    def __init__(self):
        self.adp = None
        self._optimizer = None
        self._orig_optimizer_step = None

    # ---- rule interface -------------------------------------------------

    def scale_lr(self, scale):
        """Per-param-group LR multiplier at batch-size scale ``scale``."""
        raise NotImplementedError

    # ---- patched optimizer entry points ---------------------------------

    @contextlib.contextmanager
    def _lr_scaled(self, scale):
        """Temporarily multiply every param group's LR by the rule factor."""
        groups = self._optimizer.param_groups
        saved = {id(pg): pg["lr"] for pg in groups}
        factors = np.broadcast_to(np.asarray(self.scale_lr(scale)),
                                  (len(groups),))
        try:
            for factor, pg in zip(factors, groups):
                pg["lr"] = saved[id(pg)] * float(factor)
            yield
        finally:
            for pg in groups:
                pg["lr"] = saved[id(pg)]

    def step(self, *args, **kwargs):
        """Run one optimizer step with a scaled learning rate."""
        if self.adp is None:
            raise ValueError("AdaptiveDataParallel instance is not set!")
        gns = self.adp.gns
        if not self.adp.require_backward_grad_sync:
            # Mid-accumulation microbatch: the real step happens at the
            # end of the accumulation cycle.
            return None
        scale = gns.accum_scale * gns.accum_count
        with self._lr_scaled(scale):
            self._orig_optimizer_step(*args, **kwargs)
        gns.set_progress(gns.get_progress() + gns.gain(scale))
        return None

    def zero_grad(self, *args, **kwargs):
        gns = self.adp.gns
        if gns.should_zero_grad:
            gns.reset_accumulation()
        else:
            warnings.warn("skipping zero_grad for accumulated gradient")

    # ---- wiring ----------------------------------------------------------

    def initialize(self, adp, optimizer, patch_optimizer=False):
        """Attach to an ADP instance; optionally take over the optimizer's
        ``step``/``zero_grad`` so plain training loops pick up the rule."""
        self.adp = adp
        self._optimizer = optimizer
        self._orig_optimizer_step = optimizer.step
        if patch_optimizer:
            rule = self
            # Plain function assignment: attribute lookup on the instance
            # shadows the class methods, no MethodType dance needed.
            def patched_step(*a, **kw):
                return rule.step(*a, **kw)
            def patched_zero_grad(*a, **kw):
                return rule.zero_grad(*a, **kw)
            patched_step.__name__ = "step"
            patched_zero_grad.__name__ = "zero_grad"
            optimizer.step = patched_step
            optimizer.zero_grad = patched_zero_grad


class AdaScale(ScalingRuleBase):
    """AdaScale (arxiv 2007.05105): per-group lr factor
    (var + sqr) / (var / scale + sqr)."""

    def scale_lr(self, scale):
        gns = self.adp.gns
        var = np.maximum(gns.raw_var_avg, 1e-6)
        sqr = np.maximum(gns.raw_sqr_avg, 0.0)
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
        warmup_steps = (self._base_warmup_epochs * scale
                        * self._data_size / dataloader.batch_size)
        peak = math.sqrt(scale)
        progress = self.adp.gns.get_progress()
        if progress < warmup_steps:
            return peak * progress / warmup_steps
        return peak
