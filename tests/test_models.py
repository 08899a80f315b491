"""Model-zoo tests (CPU)."""

import torch

from dist_tuto_pth_amd.models.resnet import resnet50


def test_resnet50_shapes_and_params():
    m = resnet50()
    n = sum(p.numel() for p in m.parameters())
    # standard ResNet-50 parameter count
    assert n == 25_557_032, n
    x = torch.randn(2, 3, 64, 64)   # small spatial for CPU speed
    out = m(x)
    assert out.shape == (2, 1000)


def test_resnet50_backward():
    m = resnet50(num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    loss = m(x).sum()
    loss.backward()
    assert m.conv1.weight.grad is not None
    assert m.fc.bias.grad is not None
