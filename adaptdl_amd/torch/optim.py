"""Fused flat-bucket optimizers for the gradient engine.

Reference parity note: the reference has no custom optimizers (its
examples use plain torch.optim.SGD/Adam, which also work here); these
exist purely as the MI355X-native fast path.

When attached to a GradSyncEngine (AdaptiveDataParallel does this
automatically), parameters and optimizer state are flattened into the
same persistent per-(group, dtype) buffers the gradient buckets use, so
one fused CDNA4 kernel per bucket performs the whole update
(ops/hip/gns_kernels.hip k_fused_sgd / k_fused_adamw) instead of
per-parameter torch ops.  Parameter tensors become layout-preserving
views into the flat buffers (channels_last params keep their strides),
so model/optimizer state_dicts and checkpoints remain standard.

Un-attached (or on CPU without the extension via the ops fallbacks),
behavior is identical to the torch base classes.
"""

import torch

from adaptdl_amd import ops
from adaptdl_amd.torch._engine import _segment_view

__all__ = ["FusedSGD", "FusedAdam", "FusedAdamW"]


def _flatten_engine_params(engine):
    """Move every bucket's parameters into a flat buffer (views back)."""
    for bucket in engine.buckets:
        if bucket.flat.dtype != torch.float32:
            # The fused update kernels are fp32 (master-precision
            # training).  True-bf16-parameter models train through the
            # stock torch optimizers (the engine's statistics kernels
            # DO handle bf16 buckets).
            raise ValueError(
                "Fused optimizers require float32 parameters/gradients; "
                "got a {} bucket.  Use torch.optim.SGD/Adam(W) for "
                "low-precision-parameter models.".format(
                    bucket.flat.dtype))
        if getattr(bucket, "param_flat", None) is not None:
            continue
        flat = torch.empty_like(bucket.flat)
        for p, off, n in bucket.segments:
            view = _segment_view(flat[off:off + n], p)
            view.copy_(p.data)
            p.data = view
        bucket.param_flat = flat


class FusedSGD(torch.optim.SGD):
    """SGD whose step is one fused kernel per gradient bucket."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._engine = None

    def attach_engine(self, engine):
        _flatten_engine_params(engine)
        for bucket in engine.buckets:
            bucket.sgd_momentum = None
        self._engine = engine

    def _bucket_momentum(self, bucket):
        if bucket.sgd_momentum is None:
            bucket.sgd_momentum = torch.zeros_like(bucket.flat)
            # Expose standard per-param state views for checkpointing.
            for p, off, n in bucket.segments:
                self.state[p]["momentum_buffer"] = _segment_view(
                    bucket.sgd_momentum[off:off + n], p)
        return bucket.sgd_momentum

    @torch.no_grad()
    def reset_state(self):
        """Zero the momentum buffers (bench statistics-probe isolation)."""
        if self._engine is None:
            for st in self.state.values():
                if st.get("momentum_buffer") is not None:
                    st["momentum_buffer"].zero_()
            return
        for bucket in self._engine.buckets:
            if bucket.sgd_momentum is not None:
                bucket.sgd_momentum.zero_()

    @torch.no_grad()
    def step(self, closure=None):
        if self._engine is None:
            return super().step(closure)
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for bucket in self._engine.buckets:
            group = self.param_groups[bucket.group_idx]
            momentum = group["momentum"]
            buf = self._bucket_momentum(bucket) if momentum else None
            ops.fused_sgd_step(
                bucket.param_flat, bucket.flat, buf, group["lr"],
                momentum, group["weight_decay"], group["dampening"],
                group["nesterov"])
        return loss

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        if self._engine is None:
            return
        # Re-home loaded momentum buffers into the flat storage.
        for bucket in self._engine.buckets:
            for p, off, n in bucket.segments:
                st = self.state.get(p)
                if not st or "momentum_buffer" not in st or \
                        st["momentum_buffer"] is None:
                    continue
                loaded = st["momentum_buffer"]
                if bucket.sgd_momentum is None:
                    bucket.sgd_momentum = torch.zeros_like(bucket.flat)
                view = _segment_view(
                    bucket.sgd_momentum[off:off + n], p)
                view.copy_(loaded)
                st["momentum_buffer"] = view


class _FusedAdamBase(torch.optim.AdamW):
    _adam_mode = False  # True: classic Adam (L2), False: decoupled AdamW

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._engine = None

    def attach_engine(self, engine):
        _flatten_engine_params(engine)
        for bucket in engine.buckets:
            bucket.adam_state = None
        self._engine = engine

    def _bucket_state(self, bucket):
        if bucket.adam_state is None:
            exp_avg = torch.zeros_like(bucket.flat)
            exp_avg_sq = torch.zeros_like(bucket.flat)
            # Device-resident preconditioner scalars for the one-launch-
            # per-bucket GNS statistics (identity until step >= 5).
            precond = torch.zeros(4, dtype=torch.float32,
                                  device=bucket.flat.device)
            ops.set_precond_scalars(
                precond, self.param_groups[bucket.group_idx]["betas"][1],
                self.param_groups[bucket.group_idx]["eps"], 0)
            bucket.adam_state = {"exp_avg": exp_avg,
                                 "exp_avg_sq": exp_avg_sq, "step": 0,
                                 "precond": precond}
            for p, off, n in bucket.segments:
                self.state[p]["step"] = torch.tensor(0.0)
                self.state[p]["exp_avg"] = _segment_view(
                    exp_avg[off:off + n], p)
                self.state[p]["exp_avg_sq"] = _segment_view(
                    exp_avg_sq[off:off + n], p)
        return bucket.adam_state

    @torch.no_grad()
    def step(self, closure=None):
        if self._engine is None:
            return super().step(closure)
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for bucket in self._engine.buckets:
            group = self.param_groups[bucket.group_idx]
            state = self._bucket_state(bucket)
            state["step"] += 1
            beta1, beta2 = group["betas"]
            ops.fused_adamw_step(
                bucket.param_flat, bucket.flat, state["exp_avg"],
                state["exp_avg_sq"], group["lr"], beta1, beta2,
                group["eps"], group["weight_decay"], state["step"],
                self._adam_mode)
            # Refresh the device-side preconditioner scalars so graphed
            # statistic replays see the new step's bias correction.
            ops.set_precond_scalars(state["precond"], beta2,
                                    group["eps"], state["step"])
            for p, _, _ in bucket.segments:
                self.state[p]["step"].fill_(state["step"])
        return loss

    @torch.no_grad()
    def reset_state(self):
        """Zero Adam moments and step counts (bench probe isolation)."""
        if self._engine is not None:
            for bucket in self._engine.buckets:
                if bucket.adam_state is not None:
                    bucket.adam_state["exp_avg"].zero_()
                    bucket.adam_state["exp_avg_sq"].zero_()
                    bucket.adam_state["step"] = 0
                    group = self.param_groups[bucket.group_idx]
                    ops.set_precond_scalars(bucket.adam_state["precond"],
                                            group["betas"][1],
                                            group["eps"], 0)
        for st in self.state.values():
            for key in ("exp_avg", "exp_avg_sq"):
                if st.get(key) is not None:
                    st[key].zero_()
            step = st.get("step")
            if isinstance(step, torch.Tensor):
                step.zero_()
            elif step is not None:
                st["step"] = 0

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        if self._engine is None:
            return
        for bucket in self._engine.buckets:
            if bucket.adam_state is None:
                bucket.adam_state = {
                    "exp_avg": torch.zeros_like(bucket.flat),
                    "exp_avg_sq": torch.zeros_like(bucket.flat),
                    "step": 0,
                    "precond": torch.zeros(
                        4, dtype=torch.float32,
                        device=bucket.flat.device)}
            for p, off, n in bucket.segments:
                st = self.state.get(p)
                if not st or "exp_avg" not in st:
                    continue
                for key in ("exp_avg", "exp_avg_sq"):
                    view = _segment_view(
                        bucket.adam_state[key][off:off + n], p)
                    view.copy_(st[key])
                    st[key] = view
                step = st.get("step", 0)
                bucket.adam_state["step"] = int(
                    step.item() if isinstance(step, torch.Tensor)
                    else step)
            group = self.param_groups[bucket.group_idx]
            ops.set_precond_scalars(bucket.adam_state["precond"],
                                    group["betas"][1], group["eps"],
                                    bucket.adam_state["step"])


class FusedAdamW(_FusedAdamBase):
    """AdamW (decoupled weight decay), fused per bucket."""
    _adam_mode = False


class FusedAdam(_FusedAdamBase):
    """Adam (L2 weight decay), fused per bucket."""
    _adam_mode = True

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, **kwargs):
        # torch.optim.Adam semantics: weight_decay defaults to 0 (the
        # AdamW base class would default it to 0.01).
        super().__init__(params, lr=lr, betas=betas, eps=eps,
                         weight_decay=weight_decay, **kwargs)
