"""FusedBatchNormAct2d numerics.

CPU tests check the fallback path equals BatchNorm2d(+ReLU) exactly.
GPU tests compare the CDNA4 NHWC kernels (bn_kernels.hip) against a plain
PyTorch fp32 reference of the same op, per the numerics-test contract.
"""

import pytest
import torch
import torch.nn.functional as F

from adaptdl_amd.torch.layers import FusedBatchNormAct2d


def _ref_bn(x32, bn, relu, training):
    y = F.batch_norm(x32, bn.running_mean, bn.running_var,
                     bn.weight, bn.bias, training, 0.1, bn.eps)
    return F.relu(y) if relu else y


@pytest.mark.parametrize("relu", [False, True])
def test_cpu_fallback_matches_batchnorm(relu):
    torch.manual_seed(0)
    bn = FusedBatchNormAct2d(16, relu=relu)
    ref = torch.nn.BatchNorm2d(16)
    ref.load_state_dict(
        {k: v for k, v in bn.state_dict().items()})
    x = torch.randn(4, 16, 5, 5, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y = bn(x)
    yr = ref(x2)
    if relu:
        yr = F.relu(yr)
    assert torch.allclose(y, yr, atol=1e-6)
    y.sum().backward()
    yr.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-6)


@pytest.mark.gpu
@pytest.mark.parametrize("relu", [False, True])
@pytest.mark.parametrize("shape", [(8, 64, 16, 16), (4, 8, 7, 7),
                                   (16, 512, 4, 4), (2, 24, 3, 5)])
def test_gpu_fused_bn_forward_backward(relu, shape):
    torch.manual_seed(1)
    n, c, h, w = shape
    dev = torch.device("cuda")

    bn = FusedBatchNormAct2d(c, relu=relu).to(dev)
    x = (torch.randn(n, c, h, w, device=dev) * 2 + 0.5).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)

    # fp32 reference on the SAME bf16 values.
    x32 = x.detach().float().requires_grad_(True)
    ref = torch.nn.BatchNorm2d(c).to(dev)
    ref.load_state_dict(bn.state_dict())

    y = bn(x)
    assert y.dtype == torch.bfloat16
    yr = ref(x32)
    if relu:
        yr = F.relu(yr)
    assert torch.allclose(y.float(), yr, atol=5e-2, rtol=5e-2)

    dy = torch.randn_like(yr)
    y.backward(dy.to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last))
    yr.backward(dy)
    assert torch.allclose(x.grad.float(), x32.grad, atol=7e-2, rtol=7e-2)
    assert torch.allclose(bn.weight.grad, ref.weight.grad,
                          atol=2e-1, rtol=2e-2)
    assert torch.allclose(bn.bias.grad, ref.bias.grad, atol=2e-1, rtol=2e-2)
    assert torch.allclose(bn.running_mean, ref.running_mean,
                          atol=2e-2, rtol=1e-2)
    assert torch.allclose(bn.running_var, ref.running_var,
                          atol=2e-2, rtol=1e-2)


@pytest.mark.gpu
def test_gpu_fused_bn_eval_mode():
    torch.manual_seed(2)
    c = 64
    dev = torch.device("cuda")
    bn = FusedBatchNormAct2d(c, relu=True).to(dev)
    with torch.no_grad():
        bn.running_mean.uniform_(-1, 1)
        bn.running_var.uniform_(0.5, 2)
    bn.eval()
    ref = torch.nn.BatchNorm2d(c).to(dev)
    ref.load_state_dict(bn.state_dict())
    ref.eval()
    x = torch.randn(8, c, 8, 8, device=dev).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = bn(x)
    yr = F.relu(ref(x.float()))
    assert torch.allclose(y.float(), yr, atol=5e-2, rtol=5e-2)


@pytest.mark.gpu
def test_gpu_fused_bn_used_in_resnet():
    """The HIP path must actually engage inside the flagship model."""
    from adaptdl_amd.models import ResNet18
    from adaptdl_amd.torch import layers as L
    calls = []
    orig = L._FusedBNFunction.apply

    def counting(*args):
        calls.append(1)
        return orig(*args)

    L._FusedBNFunction.apply = counting
    try:
        model = ResNet18().to("cuda").to(memory_format=torch.channels_last)
        x = torch.randn(8, 3, 32, 32, device="cuda").to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
        out.float().sum().backward()
    finally:
        L._FusedBNFunction.apply = orig
    assert len(calls) == 20  # every BN layer of ResNet-18 took the HIP path


@pytest.mark.parametrize("relu", [False, True])
def test_cpu_fallback_residual(relu):
    torch.manual_seed(5)
    bn = FusedBatchNormAct2d(8, relu=relu)
    ref = torch.nn.BatchNorm2d(8)
    ref.load_state_dict(bn.state_dict())
    x = torch.randn(4, 8, 6, 6, requires_grad=True)
    z = torch.randn(4, 8, 6, 6, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    z2 = z.detach().clone().requires_grad_(True)
    y = bn(x, residual=z)
    yr = ref(x2) + z2
    if relu:
        yr = F.relu(yr)
    assert torch.allclose(y, yr, atol=1e-6)
    y.sum().backward()
    yr.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(z.grad, z2.grad, atol=1e-6)


@pytest.mark.gpu
@pytest.mark.parametrize("relu", [False, True])
def test_gpu_fused_bn_residual(relu):
    torch.manual_seed(6)
    c = 64
    dev = torch.device("cuda")
    bn = FusedBatchNormAct2d(c, relu=relu).to(dev)
    ref = torch.nn.BatchNorm2d(c).to(dev)
    ref.load_state_dict(bn.state_dict())

    x = (torch.randn(8, c, 8, 8, device=dev)).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    z = (torch.randn(8, c, 8, 8, device=dev)).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x32 = x.detach().float().requires_grad_(True)
    z32 = z.detach().float().requires_grad_(True)

    y = bn(x, residual=z)
    yr = ref(x32) + z32
    if relu:
        yr = F.relu(yr)
    assert torch.allclose(y.float(), yr, atol=5e-2, rtol=5e-2)

    dy = torch.randn_like(yr)
    y.backward(dy.to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last))
    yr.backward(dy)
    assert torch.allclose(x.grad.float(), x32.grad, atol=7e-2, rtol=7e-2)
    assert torch.allclose(z.grad.float(), z32.grad, atol=7e-2, rtol=7e-2)
    assert torch.allclose(bn.weight.grad, ref.weight.grad,
                          atol=2e-1, rtol=2e-2)
    assert torch.allclose(bn.bias.grad, ref.bias.grad, atol=2e-1, rtol=2e-2)
