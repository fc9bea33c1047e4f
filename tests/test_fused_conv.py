"""FusedConv2d / MFMA 3x3 wrw kernel numerics (GPU).

Compares the CDNA4 implicit-GEMM weight-gradient kernel
(adaptdl_amd/ops/hip/conv_kernels.hip) against a plain PyTorch fp32
reference of the same op, per the numerics-test contract.
"""

import pytest
import torch
import torch.nn.functional as F

from adaptdl_amd.torch.layers import FusedConv2d


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 32, 32, 64),      # layer1-like
    (2, 64, 16, 16, 128),     # channel growth
    (3, 128, 16, 16, 128),    # layer2-like
    # ImageNet-resolution widths (ResNet-50 at 224): padded-width chunks
    (2, 64, 56, 56, 64),      # r50 layer1 conv2
    (2, 128, 28, 28, 128),    # r50 layer2 conv2
    (2, 256, 14, 14, 256),    # r50 layer3 conv2
    (2, 512, 7, 7, 512),      # r50 layer4 conv2 (odd W + tail lines)
])
def test_wrw_kernel_matches_fp32(shape):
    from adaptdl_amd import ops
    ext = ops._load_extension()
    torch.manual_seed(3)
    n, c, h, w, k = shape
    dev = torch.device("cuda")
    x = (torch.randn(n, c, h, w, device=dev) * 0.5).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    dy = (torch.randn(n, k, h, w, device=dev) * 0.5).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)

    assert ext.conv_wrw_ok(n, h, w, c, k)
    nsplit = ext.conv_wrw_nsplit(n, h, w, c, k)
    ws = torch.empty(nsplit * k * 9 * c, dtype=torch.float32, device=dev)
    dw = torch.empty(k, c, 3, 3, dtype=torch.float32, device=dev) \
        .contiguous(memory_format=torch.channels_last)
    ext.conv_wrw(x, dy, ws, dw)

    # fp32 reference on the same bf16 values
    x32 = x.float().requires_grad_(True)
    w32 = torch.zeros(k, c, 3, 3, device=dev, requires_grad=True)
    y = F.conv2d(x32, w32, padding=1)
    y.backward(dy.float())
    ref = w32.grad

    assert torch.allclose(dw, ref, atol=0.1, rtol=5e-2), \
        (dw - ref).abs().max().item()


@pytest.mark.gpu
def test_fused_conv2d_module_full_backward():
    torch.manual_seed(4)
    dev = torch.device("cuda")
    conv = FusedConv2d(64, 64, 3, padding=1, bias=False).to(dev)
    ref = torch.nn.Conv2d(64, 64, 3, padding=1, bias=False).to(dev)
    with torch.no_grad():
        ref.weight.copy_(conv.weight)

    x = (torch.randn(4, 64, 16, 16, device=dev)).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x32 = x.detach().float().requires_grad_(True)

    y = conv(x)
    yr = ref(x32)
    assert torch.allclose(y.float(), yr, atol=0.2, rtol=5e-2)

    dy = torch.randn_like(yr).to(torch.bfloat16)
    y.backward(dy.contiguous(memory_format=torch.channels_last))
    yr.backward(dy.float())
    assert conv.weight.grad is not None
    assert conv.weight.grad.dtype == torch.float32
    assert torch.allclose(conv.weight.grad, ref.weight.grad,
                          atol=0.3, rtol=5e-2)
    assert torch.allclose(x.grad.float(), x32.grad, atol=0.3, rtol=5e-2)


@pytest.mark.gpu
def test_wrw_engages_in_resnet():
    from adaptdl_amd.models import ResNet18
    from adaptdl_amd.torch import layers as L
    calls = []
    orig = L._FusedConvFunction.apply

    def counting(*args):
        calls.append(1)
        return orig(*args)

    L._FusedConvFunction.apply = counting
    try:
        model = ResNet18().to("cuda").to(memory_format=torch.channels_last)
        x = torch.randn(8, 3, 32, 32, device="cuda").to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
        out.float().sum().backward()
    finally:
        L._FusedConvFunction.apply = orig
    # 3x3/s1 convs with C,K >= 64 and W in {16,32}: layers 1-2 stride-1
    # convs take the MFMA wrw path (8x8/4x4 spatial -> MIOpen fallback).
    assert len(calls) >= 7, len(calls)


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 32, 32, 64),
    (2, 64, 16, 16, 128),
    (3, 128, 16, 16, 128),
])
def test_conv_mm_forward_matches_fp32(shape):
    from adaptdl_amd import ops
    ext = ops._load_extension()
    torch.manual_seed(8)
    n, c, h, w, k = shape
    dev = torch.device("cuda")
    x = (torch.randn(n, c, h, w, device=dev) * 0.5).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    assert ext.conv_mm_ok(n, h, w, c, k)
    y = torch.empty(n, k, h, w, dtype=torch.bfloat16, device=dev) \
        .contiguous(memory_format=torch.channels_last)
    ext.conv_mm(x, wt, y)
    ref = F.conv2d(x.float(), wt.float(), padding=1)
    assert torch.allclose(y.float(), ref, atol=0.15, rtol=5e-2), \
        (y.float() - ref).abs().max().item()


@pytest.mark.gpu
def test_conv_mm_full_autograd_roundtrip(monkeypatch):
    """FusedConv2d with the custom fwd + bwd-data + wrw kernels
    (experimental path, enabled explicitly)."""
    monkeypatch.setenv("ADAPTDL_EXPERIMENTAL_CONV_MM", "1")
    torch.manual_seed(9)
    dev = torch.device("cuda")
    conv = FusedConv2d(64, 64, 3, padding=1, bias=False).to(dev)
    ref = torch.nn.Conv2d(64, 64, 3, padding=1, bias=False).to(dev)
    with torch.no_grad():
        ref.weight.copy_(conv.weight)
    x = torch.randn(4, 64, 32, 32, device=dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x32 = x.detach().float().requires_grad_(True)
    y = conv(x)
    yr = ref(x32)
    assert torch.allclose(y.float(), yr, atol=0.2, rtol=5e-2)
    # use the SAME bf16-rounded dy on both sides (weight gradients sum
    # ~4k terms; an asymmetric fp32-vs-bf16 dy alone shifts them ~1.7%)
    dy = torch.randn_like(yr).to(torch.bfloat16)
    y.backward(dy.contiguous(memory_format=torch.channels_last))
    yr.backward(dy.float())
    assert torch.allclose(x.grad.float(), x32.grad, atol=0.3, rtol=5e-2)
    assert torch.allclose(conv.weight.grad, ref.weight.grad,
                          atol=0.3, rtol=5e-2)


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 16, 16, 128),     # layer2.0.conv1-like (dy side)
    (3, 32, 16, 16, 64),      # small channels
    (2, 128, 8, 8, 256),      # layer3.0.conv1-like (Wo=8 k-split path)
    (3, 32, 8, 8, 128),       # Wo=8, K=128
])
def test_s2_bwd_kernel_matches_fp32(shape):
    """Polyphase stride-2 backward-data vs fp32 autograd reference."""
    from adaptdl_amd import ops
    ext = ops._load_extension()
    torch.manual_seed(5)
    n, c, ho, wo, k = shape
    hi, wi = 2 * ho, 2 * wo
    dev = torch.device("cuda")
    dy = (torch.randn(n, k, ho, wo, device=dev) * 0.5) \
        .to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(k, c, 3, 3, device=dev) * 0.2).to(torch.bfloat16)

    assert ext.conv_s2_bwd_ok(n, ho, wo, k, c)
    wt = w.permute(1, 2, 3, 0).contiguous()
    dx = torch.empty(n, c, hi, wi, dtype=torch.bfloat16, device=dev) \
        .contiguous(memory_format=torch.channels_last)
    ext.conv_s2_bwd(dy, wt, dx)

    ref = torch.nn.grad.conv2d_input(
        (n, c, hi, wi), w.float(), dy.float(), stride=2, padding=1)
    assert torch.allclose(dx.float(), ref, atol=0.1, rtol=5e-2), \
        (dx.float() - ref).abs().max().item()


@pytest.mark.gpu
def test_s2_conv_module_backward():
    """FusedConv2d stride-2 path: dx/dw through _S2ConvFunction."""
    torch.manual_seed(6)
    dev = torch.device("cuda")
    conv = FusedConv2d(64, 128, 3, stride=2, padding=1, bias=False).to(dev)
    x = torch.randn(2, 64, 32, 32, device=dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = conv(x)
    assert y.shape == (2, 128, 16, 16)
    y.backward(torch.randn_like(y))
    assert x.grad is not None and conv.weight.grad is not None

    # reference grads on the same bf16-rounded dy
    x32 = x.detach().float().requires_grad_(True)
    w32 = conv.weight.detach().float().requires_grad_(True)
    dy = torch.randn(2, 128, 16, 16, device=dev).to(torch.bfloat16)
    x.grad = None
    conv.weight.grad = None
    conv(x).backward(dy.contiguous(memory_format=torch.channels_last))
    y32 = F.conv2d(x32, w32, stride=2, padding=1)
    y32.backward(dy.float())
    assert torch.allclose(x.grad.float(), x32.grad, atol=0.1, rtol=5e-2)


@pytest.mark.gpu
def test_s2_fwd_kernel_matches_fp32():
    """Experimental polyphase stride-2 FORWARD vs fp32 reference.

    Opt-in: the kernel's index math is simulation-verified and it
    compiles clean, but the round's GPU budget ran out before this test
    could run on hardware -- it stays env-gated so an unvalidated test
    cannot abort the -x suite.  Enable with ADAPTDL_EXPERIMENTAL_S2_FWD=1
    (first task of the next GPU session).
    """
    import os
    if os.getenv("ADAPTDL_EXPERIMENTAL_S2_FWD") != "1":
        pytest.skip("experimental s2 fwd kernel: not yet GPU-validated; "
                    "set ADAPTDL_EXPERIMENTAL_S2_FWD=1")
    from adaptdl_amd import ops
    ext = ops._load_extension()
    torch.manual_seed(7)
    n, c, h, w_, k = 3, 64, 32, 32, 128
    dev = torch.device("cuda")
    x = (torch.randn(n, c, h, w_, device=dev) * 0.5).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(k, c, 3, 3, device=dev) * 0.2).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    assert ext.conv_s2_fwd_ok(n, h, w_, c, k)
    y = torch.empty(n, k, h // 2, w_ // 2, dtype=torch.bfloat16,
                    device=dev).contiguous(memory_format=torch.channels_last)
    ext.conv_s2_fwd(x, w, y)
    ref = F.conv2d(x.float(), w.float(), stride=2, padding=1)
    assert torch.allclose(y.float(), ref, atol=0.1, rtol=5e-2), \
        (y.float() - ref).abs().max().item()


def test_s2_1x1_gemm_function_matches_autograd():
    """1x1 stride-2 conv as GEMMs (_S2Conv1x1Function) vs conv2d
    autograd, on CPU fp32 (identical semantics on GPU bf16)."""
    from adaptdl_amd.torch.layers import _S2Conv1x1Function
    torch.manual_seed(8)
    n, c, h, w_, k = 3, 16, 8, 8, 32
    x = torch.randn(n, c, h, w_, requires_grad=True)
    weight = torch.randn(k, c, 1, 1, requires_grad=True)
    y = _S2Conv1x1Function.apply(x, weight)
    dy = torch.randn_like(y)
    y.backward(dy)

    x2 = x.detach().clone().requires_grad_(True)
    w2 = weight.detach().clone().requires_grad_(True)
    y2 = F.conv2d(x2, w2, stride=2)
    y2.backward(dy)

    assert torch.allclose(y, y2, atol=1e-4)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(weight.grad, w2.grad, atol=1e-3)


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 16, 16, 128),     # layer2.0.conv1 (x 32x32x64 -> dy 16x16x128)
    (3, 128, 8, 8, 256),      # layer3.0.conv1
])
def test_s2_wrw_kernel_matches_fp32(shape):
    """Experimental polyphase stride-2 weight gradient vs fp32 reference.

    Opt-in like the s2 forward: simulation-verified + compiled but not
    yet run on hardware (GPU budget exhausted this round); enable with
    ADAPTDL_EXPERIMENTAL_S2_FWD=1 (shares the experimental gate).
    """
    import os
    if os.getenv("ADAPTDL_EXPERIMENTAL_S2_FWD") != "1":
        pytest.skip("experimental s2 wrw kernel: not yet GPU-validated; "
                    "set ADAPTDL_EXPERIMENTAL_S2_FWD=1")
    from adaptdl_amd import ops
    ext = ops._load_extension()
    torch.manual_seed(9)
    n, c, ho, wo, k = shape
    hi, wi = 2 * ho, 2 * wo
    dev = torch.device("cuda")
    x = (torch.randn(n, c, hi, wi, device=dev) * 0.5).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    dy = (torch.randn(n, k, ho, wo, device=dev) * 0.5) \
        .to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    assert ext.conv_s2_wrw_ok(n, ho, wo, c, k)
    nsplit = ext.conv_s2_wrw_nsplit(n, ho, wo, c, k)
    ws = torch.empty(nsplit * k * 9 * c, dtype=torch.float32, device=dev)
    dw = torch.empty(k, c, 3, 3, dtype=torch.float32, device=dev) \
        .contiguous(memory_format=torch.channels_last)
    ext.conv_s2_wrw(x, dy, ws, dw)

    x32 = x.float().requires_grad_(True)
    w32 = torch.zeros(k, c, 3, 3, device=dev, requires_grad=True)
    y = F.conv2d(x32, w32, stride=2, padding=1)
    y.backward(dy.float())
    assert torch.allclose(dw, w32.grad, atol=0.1, rtol=5e-2), \
        (dw - w32.grad).abs().max().item()


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    (2, 128, 8, 8, 256),
    (3, 32, 8, 8, 128),
])
def test_s2_bwd_w8b_matches_fp32(shape):
    """P2=4 redesign of the Wo=8 stride-2 bwd-data (experimental;
    shares the unvalidated-kernel gate)."""
    import os
    if os.getenv("ADAPTDL_EXPERIMENTAL_S2_FWD") != "1":
        pytest.skip("experimental w8b kernel: not yet GPU-validated; "
                    "set ADAPTDL_EXPERIMENTAL_S2_FWD=1")
    from adaptdl_amd import ops
    ext = ops._load_extension()
    torch.manual_seed(10)
    n, c, ho, wo, k = shape
    hi, wi = 2 * ho, 2 * wo
    dev = torch.device("cuda")
    dy = (torch.randn(n, k, ho, wo, device=dev) * 0.5) \
        .to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(k, c, 3, 3, device=dev) * 0.2).to(torch.bfloat16)
    assert ext.conv_s2_bwd_w8b_ok(n, ho, wo, k, c)
    wt = w.permute(1, 2, 3, 0).contiguous()
    dx = torch.empty(n, c, hi, wi, dtype=torch.bfloat16, device=dev) \
        .contiguous(memory_format=torch.channels_last)
    ext.conv_s2_bwd_w8b(dy, wt, dx)
    ref = torch.nn.grad.conv2d_input(
        (n, c, hi, wi), w.float(), dy.float(), stride=2, padding=1)
    assert torch.allclose(dx.float(), ref, atol=0.1, rtol=5e-2), \
        (dx.float() - ref).abs().max().item()
