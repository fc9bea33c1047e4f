"""Fused MI355X layers for conv workloads.

FusedBatchNormAct2d is a drop-in BatchNorm2d(+ReLU) whose GPU path runs
the hand-written CDNA4 NHWC kernels (adaptdl_amd/ops/hip/bn_kernels.hip)
— one bandwidth-bound reduce + one apply pass per direction, with the
ReLU and its backward mask folded in.  Motivation: on channels_last bf16
ResNets MIOpen's NHWC spatial batchnorm is 56% of all kernel time
(profiles/bench_r02_channels_last_kernel_stats.csv).  On CPU (and for
layouts the kernels do not cover) the same module falls back to the
standard torch ops, so CPU tests exercise identical semantics.

Reference parity note: the reference has no custom layers (models use
nn.BatchNorm2d, e.g. /root/reference/examples/pytorch-cifar/models/
resnet.py); this module exists purely as the MI355X-native fast path.
"""

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from adaptdl_amd import ops


def _num_rblocks(m, c):
    """Mirror of bn_kernels.hip stage-1 grid sizing (partials rows)."""
    rpb = max(256 // (c // 8), 1)
    return max(min((m + rpb - 1) // rpb, 1024), 1)


def _hip_bn_ok(x):
    c = x.shape[1] if x.dim() == 4 else 0
    return (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4 and
            c % 8 == 0 and c // 8 <= 256 and
            x.is_contiguous(memory_format=torch.channels_last) and
            ops.has_extension())


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, z, weight, bias, running_mean, running_var,
                training, momentum, eps, relu):
        ext = ops._load_extension()
        y = torch.empty_like(x)
        c = x.shape[1]
        m = x.numel() // c
        opt = dict(dtype=torch.float32, device=x.device)
        ws = torch.empty(2 * c * _num_rblocks(m, c), **opt) \
            if training else torch.empty(0, **opt)
        sums = torch.empty(2 * c, **opt)
        save_mean = torch.empty(c, **opt)
        save_rstd = torch.empty(c, **opt)
        scale = torch.empty(c, **opt)
        shift = torch.empty(c, **opt)
        none = torch.empty(0, **opt)
        znone = torch.empty(0, dtype=x.dtype, device=x.device)
        ext.bn_fwd(x, z if z is not None else znone, y, weight, bias,
                   running_mean if running_mean is not None else none,
                   running_var if running_var is not None else none,
                   float(momentum), float(eps), bool(training), bool(relu),
                   ws, sums, save_mean, save_rstd, scale, shift)
        ctx.save_for_backward(x, z if z is not None else znone, weight,
                              save_mean, save_rstd, scale, shift)
        ctx.bn_train = bool(training)
        ctx.bn_relu = bool(relu)
        ctx.bn_has_z = z is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ops._load_extension()
        x, z, weight, save_mean, save_rstd, scale, shift = \
            ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = torch.empty_like(x)
        dz = torch.empty_like(x) if ctx.bn_has_z else \
            torch.empty(0, dtype=x.dtype, device=x.device)
        c = x.shape[1]
        m = x.numel() // c
        opt = dict(dtype=torch.float32, device=x.device)
        ws = torch.empty(2 * c * _num_rblocks(m, c), **opt)
        sums = torch.empty(2 * c, **opt)
        dgamma = torch.empty(c, **opt)
        dbeta = torch.empty(c, **opt)
        pqr = torch.empty(3 * c, **opt)
        ext.bn_bwd(x, z, dy, dx, dz, weight, save_mean, save_rstd, scale,
                   shift, ctx.bn_train, ctx.bn_relu, ws, sums, dgamma,
                   dbeta, pqr)
        return (dx, dz if ctx.bn_has_z else None, dgamma, dbeta) + \
            (None,) * 6


class FusedBatchNormAct2d(nn.BatchNorm2d):
    """BatchNorm2d with an optionally fused ReLU.

    GPU (bf16 channels_last): hand-written CDNA4 kernels.
    Anything else: standard F.batch_norm (+ relu), identical math.
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True, relu=False):
        super().__init__(num_features, eps=eps, momentum=momentum,
                         affine=affine,
                         track_running_stats=track_running_stats)
        self.relu = relu

    def forward(self, x, residual=None):
        use_batch_stats = self.training or not self.track_running_stats
        if _hip_bn_ok(x) and self.affine and \
                (residual is None or
                 (residual.dtype == x.dtype and residual.is_contiguous(
                     memory_format=torch.channels_last))):
            if self.training and self.track_running_stats and \
                    self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            momentum = self.momentum if self.momentum is not None else 0.0
            return _FusedBNFunction.apply(
                x, residual, self.weight, self.bias, self.running_mean,
                self.running_var, use_batch_stats, momentum, self.eps,
                self.relu)
        y = super().forward(x)
        if residual is not None:
            y = y + residual
        return F.relu(y) if self.relu else y


class _FusedConvFunction(torch.autograd.Function):
    """conv2d with the weight gradient computed by the MFMA wrw kernel.

    Forward runs the library conv (MIOpen); backward computes
    grad_input via aten.convolution_backward and grad_weight via
    ops/hip/conv_kernels.hip (3x3 stride-1 NHWC bf16, fp32 output).
    """

    @staticmethod
    def forward(ctx, x, weight, wb, stride, padding, dilation, groups):
        ext = ops._load_extension()
        n, c, h, w = x.shape
        k = wb.shape[0]
        # The custom fwd/bwd-data kernel is numerics-verified and beats
        # MIOpen's UNTUNED find choice on C=64 (213 vs 229 us), but
        # loses to its tuned in-context pick (154 us) -- end-to-end it
        # cost ~2.5 ms/step, so it stays opt-in (bench_r06 vs r05).
        if os.getenv("ADAPTDL_EXPERIMENTAL_CONV_MM") == "1" \
                and ext.conv_mm_ok(n, h, w, c, k):
            y = torch.empty(n, k, h, w, dtype=x.dtype, device=x.device) \
                .contiguous(memory_format=torch.channels_last)
            ext.conv_mm(x, wb, y)
        else:
            y = torch.nn.functional.conv2d(x, wb, None, stride, padding,
                                           dilation, groups)
        ctx.save_for_backward(x, wb)
        ctx.conv_args = (stride, padding, dilation, groups)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ops._load_extension()
        x, wb = ctx.saved_tensors
        stride, padding, dilation, groups = ctx.conv_args
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = None
        if ctx.needs_input_grad[0]:
            n, c, h, w = x.shape
            k = wb.shape[0]
            # MIOpen's backward-data igemm (137 us on layer1) still
            # beats conv_mm (213 us) -- keep it unless forced.
            if os.getenv("ADAPTDL_EXPERIMENTAL_CONV_MM") == "1" \
                    and ext.conv_mm_ok(n, h, w, k, c):
                # dx = conv(dy, W flipped with C/K roles swapped)
                wt = wb.flip(2, 3).permute(1, 0, 2, 3) \
                    .contiguous(memory_format=torch.channels_last)
                dx = torch.empty_like(x)
                ext.conv_mm(dy, wt, dx)
            else:
                dx = torch.ops.aten.convolution_backward(
                    dy, x, wb, None, stride, padding, dilation, False,
                    [0, 0], groups, [True, False, False])[0]
        n, c, h, w = x.shape
        k = dy.shape[1]
        nsplit = ext.conv_wrw_nsplit(n, h, w, c, k)
        ws = torch.empty(nsplit * k * 9 * c, dtype=torch.float32,
                         device=x.device)
        dw = torch.empty(k, c, 3, 3, dtype=torch.float32,
                         device=x.device) \
            .contiguous(memory_format=torch.channels_last)
        ext.conv_wrw(x, dy, ws, dw)
        return dx, dw, None, None, None, None, None


class _S2ConvFunction(torch.autograd.Function):
    """3x3 stride-2 conv with backward-data on the polyphase MFMA kernel.

    The stride-2 backward-data otherwise falls to a CK scatter kernel
    measured at ~1035 us/call (~75 GF/s) -- the largest per-call conv
    entry of the ResNet step (profiles/bench_r05_kernel_stats.csv).
    Forward and the weight gradient stay on MIOpen.
    """

    @staticmethod
    def forward(ctx, x, weight, stride, padding, dilation, groups):
        wb = weight.detach().to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        y = torch.nn.functional.conv2d(x, wb, None, stride, padding,
                                       dilation, groups)
        ctx.save_for_backward(x, wb)
        ctx.conv_args = (stride, padding, dilation, groups)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ops._load_extension()
        x, wb = ctx.saved_tensors
        stride, padding, dilation, groups = ctx.conv_args
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            # dx[even/odd phases] = dense mini-convs of dy with the
            # (C,3,3,K)-transposed weight (no flip: taps keep their
            # original (dh, dw) indices in the polyphase decomposition).
            wt = wb.permute(1, 2, 3, 0).contiguous()
            dx = torch.empty_like(x)
            n2, k2, ho2, wo2 = dy.shape
            if os.getenv("ADAPTDL_S2_W8B") == "1" and wo2 == 8 \
                    and ext.conv_s2_bwd_w8b_ok(n2, ho2, wo2, k2,
                                               x.shape[1]):
                ext.conv_s2_bwd_w8b(dy, wt, dx)   # P2=4 redesign A/B
            else:
                ext.conv_s2_bwd(dy, wt, dx)
        if ctx.needs_input_grad[1]:
            ext2 = ops._load_extension()
            n, k, ho, wo = dy.shape
            c = x.shape[1]
            if os.getenv("ADAPTDL_S2_WRW") == "1" \
                    and ext2.conv_s2_wrw_ok(n, ho, wo, c, k):
                nsplit = ext2.conv_s2_wrw_nsplit(n, ho, wo, c, k)
                ws = torch.empty(nsplit * k * 9 * c,
                                 dtype=torch.float32, device=x.device)
                dw = torch.empty(k, c, 3, 3, dtype=torch.float32,
                                 device=x.device) \
                    .contiguous(memory_format=torch.channels_last)
                ext2.conv_s2_wrw(x, dy, ws, dw)
            else:
                dw = torch.ops.aten.convolution_backward(
                    dy, x, wb, None, stride, padding, dilation, False,
                    [0, 0], groups, [False, True, False])[1]
        return dx, dw, None, None, None, None


class _S2Conv1x1Function(torch.autograd.Function):
    """1x1 stride-2 conv (ResNet downsample shortcut) as plain GEMMs.

    The strided 1x1 conv touches only the even-index pixels, so all
    three directions are matmuls over the subsampled image (hipBLASLt on
    MI355X) instead of MIOpen/CK's strided conv kernels:
        y            = x[:, ::2, ::2] @ W^T
        dx[::2,::2]  = dy @ W          (zeros elsewhere)
        dW           = dy^T @ x[:, ::2, ::2]
    Semantics are identical on CPU, so the path is CPU-testable;
    measured-perf gating happens at the caller.
    """

    @staticmethod
    def forward(ctx, x, weight):
        wb = weight.detach().to(x.dtype)
        xs = x[:, :, ::2, ::2]                      # (N, C, Ho, Wo)
        xs_nhwc = xs.permute(0, 2, 3, 1)
        y = (xs_nhwc @ wb.view(wb.shape[0], wb.shape[1]).t()) \
            .permute(0, 3, 1, 2) \
            .contiguous(memory_format=torch.channels_last)
        ctx.save_for_backward(x, wb)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, wb = ctx.saved_tensors
        k, c = wb.shape[0], wb.shape[1]
        w2d = wb.view(k, c)
        dy_nhwc = dy.permute(0, 2, 3, 1)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = torch.zeros_like(x)
            dx[:, :, ::2, ::2] = (dy_nhwc @ w2d.to(dy.dtype)) \
                .permute(0, 3, 1, 2)
        if ctx.needs_input_grad[1]:
            # bf16 GEMM (fp32 accumulation inside), fp32 master grad out
            xs_nhwc = x[:, :, ::2, ::2].permute(0, 2, 3, 1)
            dw = (dy_nhwc.reshape(-1, k).t()
                  @ xs_nhwc.reshape(-1, c).to(dy.dtype)) \
                .float().view(k, c, 1, 1)
        return dx, dw


class FusedConv2d(nn.Conv2d):
    """Conv2d whose 3x3/s1 weight gradient runs the MFMA wrw kernel."""

    def _wrw_ok(self, x):
        if not (x.is_cuda and x.dtype == torch.bfloat16 and
                x.dim() == 4 and ops.has_extension() and
                self.bias is None and self.groups == 1 and
                self.kernel_size == (3, 3) and self.stride == (1, 1) and
                self.padding == (1, 1) and self.dilation == (1, 1) and
                x.is_contiguous(memory_format=torch.channels_last)):
            return False
        ext = ops._load_extension()
        n, c, h, w = x.shape
        return bool(ext.conv_wrw_ok(n, h, w, c, self.out_channels))

    def _s2_ok(self, x):
        if os.getenv("ADAPTDL_S2_BWD") == "0":   # A/B escape hatch
            return False
        if not (x.is_cuda and x.dtype == torch.bfloat16 and
                x.dim() == 4 and ops.has_extension() and
                self.bias is None and self.groups == 1 and
                self.kernel_size == (3, 3) and self.stride == (2, 2) and
                self.padding == (1, 1) and self.dilation == (1, 1) and
                x.is_contiguous(memory_format=torch.channels_last)):
            return False
        ext = ops._load_extension()
        n, c, h, w = x.shape
        if h % 2 or w % 2:
            return False
        # The Wo=8 k-split variant is numerics-verified but measured
        # SLOWER in context (47.2 vs 43.4 ms/step, bench_r10 vs r09):
        # CT=16 makes each block re-read dy 8x across 128 tiny chunk
        # iterations of 5 barriers + an f32 merge each.  Off by default
        # until the per-chunk overhead is redesigned (ROADMAP item 2).
        if w // 2 != 16 and os.getenv("ADAPTDL_S2_W8") != "1" \
                and os.getenv("ADAPTDL_S2_W8B") != "1":
            return False
        return bool(ext.conv_s2_bwd_ok(n, h // 2, w // 2,
                                       self.out_channels, c))

    def _s2_1x1_ok(self, x):
        # GEMM path for the 1x1 stride-2 downsample shortcut.  Gated
        # off by default pending an in-context A/B next round (the
        # Wo=8 s2_bwd lesson: never default-on an unmeasured path).
        if os.getenv("ADAPTDL_S2_1X1") != "1":
            return False
        return (x.is_cuda and x.dtype == torch.bfloat16 and
                x.dim() == 4 and self.bias is None and
                self.groups == 1 and self.kernel_size == (1, 1) and
                self.stride == (2, 2) and self.padding == (0, 0) and
                x.shape[2] % 2 == 0 and x.shape[3] % 2 == 0 and
                x.is_contiguous(memory_format=torch.channels_last))

    def _cast_weight(self):
        """bf16 channels_last weight, cached across the accumulation
        microbatches of one optimizer cycle.

        autocast re-casts module weights on EVERY forward (its cast
        cache lives only inside one autocast region), which measured
        ~4% of flagship kernel time as small elementwise launches.
        Weights only change between optimizer cycles, so the cast is
        keyed on (engine cycle serial, weight._version) — the serial
        covers the fused optimizers (raw-kernel updates do not bump
        _version), _version covers stock optimizers without an engine.
        """
        from adaptdl_amd.torch import _engine
        key = (_engine.cycle_serial(), self.weight._version)
        cached = getattr(self, "_wb_cache", None)
        if cached is not None and cached[1] == key \
                and os.getenv("ADAPTDL_NO_WB_CACHE") != "1":
            return cached[0]
        wb = self.weight.detach().to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        self._wb_cache = (wb, key)
        return wb

    def forward(self, x):
        if torch.is_autocast_enabled() and x.is_cuda and \
                x.dtype != torch.bfloat16 and \
                torch.get_autocast_dtype("cuda") == torch.bfloat16:
            x = x.to(torch.bfloat16)
        if self._wrw_ok(x) and (self.weight.requires_grad or
                                x.requires_grad):
            return _FusedConvFunction.apply(x, self.weight,
                                            self._cast_weight(),
                                            self.stride, self.padding,
                                            self.dilation, self.groups)
        if x.requires_grad and self._s2_ok(x):
            return _S2ConvFunction.apply(x, self.weight, self.stride,
                                         self.padding, self.dilation,
                                         self.groups)
        if self._s2_1x1_ok(x):
            return _S2Conv1x1Function.apply(x, self.weight)
        return super().forward(x)
