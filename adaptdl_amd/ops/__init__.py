"""Compute kernels for the gradient-statistics/optimizer hot path.

On an MI355X GPU these dispatch to the in-tree HIP/CDNA4 extension
(``adaptdl_amd/ops/hip``) — a single fused pass over each gradient bucket
computes the scale (grad averaging / AMP unscale) together with the fp64
sum-of-squares statistics that drive the gradient noise scale, and fused
flat-bucket optimizers implement the update step.  On CPU (CI, gloo tests)
the same math runs on plain torch ops.

A CUDA-device call without the compiled extension raises loudly rather than
silently falling back (the HIP path must be the one that runs on GPU).
"""

import logging

import torch

LOG = logging.getLogger(__name__)

_EXT = None
_EXT_ERR = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import adaptdl_amd_hip  # built in-tree by setup.py build_ext
        _EXT = adaptdl_amd_hip
    except ImportError as e:
        _EXT_ERR = e
    return _EXT


def has_extension():
    return _load_extension() is not None


def _require_ext(t):
    if t.is_cuda:
        ext = _load_extension()
        if ext is None:
            raise RuntimeError(
                "adaptdl_amd_hip extension is required for GPU tensors but "
                "could not be imported ({}). Build it with: python setup.py "
                "build_ext --inplace (PYTORCH_ROCM_ARCH=gfx950)"
                .format(_EXT_ERR))
        return ext
    return None


def _flat(t):
    """Flatten to a storage-order VIEW (never a copy).

    ``view(-1)`` only works for default-contiguous tensors; dense tensors
    in a non-default stride order (channels_last params / Adam moments)
    are flattened with as_strided over the same storage so in-place
    updates still hit the original tensor.  Element order follows storage
    order, which is what the flat-bucket segments use too.
    """
    if t.is_contiguous():
        return t.view(-1)
    if t.dim() in (4, 5) and (
            t.is_contiguous(memory_format=torch.channels_last)
            or (t.dim() == 5 and
                t.is_contiguous(memory_format=torch.channels_last_3d))):
        return t.detach().as_strided((t.numel(),), (1,),
                                     t.storage_offset())
    raise RuntimeError(
        "cannot flatten non-dense tensor of shape {} / strides {} without "
        "a copy".format(tuple(t.shape), tuple(t.stride())))


def sqsum(t, out):
    """out += sum(t.double() ** 2)  (out: 0-dim float64 on same device)."""
    ext = _require_ext(t)
    if ext is not None:
        ext.sqsum(_flat(t), out)
    else:
        out.add_(t.double().pow(2).sum())


def scale_and_sqsum(t, scale, out):
    """t *= scale; out += sum(t.double() ** 2), in one pass."""
    ext = _require_ext(t)
    if ext is not None:
        ext.scale_and_sqsum(_flat(t), float(scale), out)
    else:
        if scale != 1.0:
            t.mul_(scale)
        out.add_(t.double().pow(2).sum())


def sqsum_diff_update(cur, prev, out):
    """out += sum((cur - prev)**2); prev.copy_(cur), in one pass.

    Used during gradient accumulation: ``cur`` is the running gradient sum
    and ``prev`` its value after the previous microbatch, so the difference
    is the newest microbatch's gradient contribution.
    """
    ext = _require_ext(cur)
    if ext is not None:
        ext.sqsum_diff_update(_flat(cur), _flat(prev), out)
    else:
        out.add_((cur.double() - prev.double()).pow(2).sum())
        prev.copy_(cur)


def sqsum_avg(cur, prev, out):
    """out += sum(((cur + prev) / 2)**2) (differenced GNS estimator)."""
    ext = _require_ext(cur)
    if ext is not None:
        ext.sqsum_avg(_flat(cur), _flat(prev), out)
    else:
        out.add_(((cur.double() + prev.double()) / 2).pow(2).sum())


def fused_sgd_step(param, grad, momentum_buf, lr, momentum, weight_decay,
                   dampening, nesterov):
    """Flat-bucket SGD update (single fused pass on GPU)."""
    ext = _require_ext(param)
    if ext is not None:
        ext.fused_sgd(_flat(param), _flat(grad),
                      _flat(momentum_buf) if momentum_buf is not None
                      else torch.empty(0, device=param.device,
                                       dtype=param.dtype),
                      float(lr), float(momentum), float(weight_decay),
                      float(dampening), bool(nesterov))
    else:
        d_p = grad
        if weight_decay != 0:
            d_p = d_p.add(param, alpha=weight_decay)
        if momentum != 0:
            momentum_buf.mul_(momentum).add_(d_p, alpha=1 - dampening)
            if nesterov:
                d_p = d_p.add(momentum_buf, alpha=momentum)
            else:
                d_p = momentum_buf
        param.add_(d_p, alpha=-lr)


def fused_adamw_step(param, grad, exp_avg, exp_avg_sq, lr, beta1, beta2,
                     eps, weight_decay, step, adam_mode):
    """Flat-bucket Adam/AdamW update (single fused pass on GPU).

    adam_mode: True => L2-regularization Adam; False => decoupled AdamW.
    """
    ext = _require_ext(param)
    if ext is not None:
        ext.fused_adamw(_flat(param), _flat(grad), _flat(exp_avg),
                        _flat(exp_avg_sq), float(lr), float(beta1),
                        float(beta2), float(eps), float(weight_decay),
                        int(step), bool(adam_mode))
    else:
        if adam_mode and weight_decay != 0:
            grad = grad.add(param, alpha=weight_decay)
        elif not adam_mode and weight_decay != 0:
            param.mul_(1 - lr * weight_decay)
        exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
        exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
        bias1 = 1 - beta1 ** step
        bias2 = 1 - beta2 ** step
        denom = (exp_avg_sq / bias2).sqrt_().add_(eps)
        param.addcdiv_(exp_avg, denom, value=-lr / bias1)


def precond_sqsum(grad, exp_avg_sq, beta2, eps, step, out):
    """out += sum((grad / pinv)**2) with Adam preconditioner
    pinv = sqrt(exp_avg_sq / (1 - beta2**step)) + eps  (fp64 accumulate)."""
    ext = _require_ext(grad)
    if ext is not None:
        ext.precond_sqsum(_flat(grad), _flat(exp_avg_sq), float(beta2),
                          float(eps), int(step), out)
    else:
        corr = 1.0 - beta2 ** step
        pinv = (_flat(exp_avg_sq).double() / corr).sqrt().add_(eps)
        out.add_((_flat(grad).double() / pinv).pow(2).sum())


def set_precond_scalars(pc, beta2, eps, step, min_steps=5):
    """Write the device-resident preconditioner scalars for
    :func:`precond_sqsum_dev`: pc = {inv_corr_sqrt, eps, use_precond, 0}.

    Identity preconditioning (use_precond=0) below ``min_steps`` — the
    same warmup gate the per-segment path applies host-side.  Called
    from the (eager) fused-Adam step, so hipGraph replays of the
    statistics kernels pick up each step's bias correction.
    """
    import math
    if step >= min_steps:
        vals = [1.0 / math.sqrt(1.0 - beta2 ** step), eps, 1.0, 0.0]
    else:
        vals = [1.0, 0.0, 0.0, 0.0]
    pc.copy_(torch.tensor(vals, dtype=torch.float32))


def precond_sqsum_dev(grad, exp_avg_sq, pc, out):
    """out += sum((grad / pinv)**2) with the preconditioner scalars read
    from device tensor ``pc`` (see :func:`set_precond_scalars`) — one
    launch per flat bucket, safe inside hipGraph capture."""
    ext = _require_ext(grad)
    if ext is not None:
        ext.precond_sqsum_dev(_flat(grad), _flat(exp_avg_sq), pc, out)
    else:
        ics, eps, use_p = (float(pc[0]), float(pc[1]), float(pc[2]))
        g = _flat(grad).double()
        if use_p == 0.0:
            out.add_(g.pow(2).sum())
        else:
            pinv = _flat(exp_avg_sq).double().sqrt().mul_(ics).add_(eps)
            out.add_((g / pinv).pow(2).sum())
