"""Model zoo checks: parameter counts match the canonical architectures and
forward shapes are right (CPU, tiny inputs)."""

import torch

from baguanet.models import resnet50, vgg16


def n_params(m):
    return sum(p.numel() for p in m.parameters())


def test_vgg16_param_count():
    # canonical torchvision vgg16 @1000 classes
    assert n_params(vgg16()) == 138_357_544


def test_resnet50_param_count():
    # canonical torchvision resnet50 @1000 classes
    assert n_params(resnet50()) == 25_557_032


def test_vgg16_forward_backward():
    m = vgg16(num_classes=10)
    x = torch.randn(2, 3, 64, 64)  # adaptive pool → any (≥32) input size
    out = m(x)
    assert out.shape == (2, 10)
    out.sum().backward()
    assert all(p.grad is not None for p in m.parameters())


def test_resnet50_forward():
    m = resnet50(num_classes=10)
    m.eval()
    with torch.no_grad():
        out = m(torch.randn(2, 3, 64, 64))
    assert out.shape == (2, 10)
