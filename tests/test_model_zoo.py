"""CIFAR model zoo: forward shapes and a training step per architecture
(reference: examples/pytorch-cifar/models/* — 15 standard CIFAR nets)."""

import pytest
import torch

from adaptdl_amd.models.cifar import CIFAR_MODELS


@pytest.mark.parametrize("name", sorted(CIFAR_MODELS))
def test_forward_shape(name):
    torch.manual_seed(0)
    model = CIFAR_MODELS[name]()
    y = model(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)
    assert torch.isfinite(y).all()


@pytest.mark.parametrize("name", ["vgg11", "preact_resnet18",
                                  "mobilenetv2", "densenet121"])
def test_train_step_reduces_loss(name):
    torch.manual_seed(0)
    model = CIFAR_MODELS[name]()
    optim = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    x = torch.randn(16, 3, 32, 32)
    t = torch.randint(0, 10, (16,))
    losses = []
    for _ in range(8):
        optim.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), t)
        loss.backward()
        optim.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]


@pytest.mark.gpu
@pytest.mark.parametrize("name", ["preact_resnet18", "mobilenetv2",
                                  "densenet121", "vgg16"])
def test_zoo_gpu_bf16_step(name):
    """Zoo models through the fused conv/BN stack: bf16 channels_last
    forward+backward stays finite on MI355X."""
    torch.manual_seed(0)
    model = CIFAR_MODELS[name]().cuda() \
        .to(memory_format=torch.channels_last)
    x = torch.randn(32, 3, 32, 32, device="cuda") \
        .contiguous(memory_format=torch.channels_last)
    t = torch.randint(0, 10, (32,), device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = torch.nn.functional.cross_entropy(model(x), t)
    loss.backward()
    assert torch.isfinite(loss)
    for p in model.parameters():
        assert p.grad is None or torch.isfinite(p.grad).all()
