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


def test_mlp_forward_backward_matches_torch():
    """MLP on the generic HIP-op layer vs a plain-torch replica —
    the op layer generalizes beyond the reference ConvNet's shapes."""
    import torch
    import torch.nn.functional as F
    from dist_tuto_pth_amd.models import MLP

    torch.manual_seed(3)
    m = MLP((784, 96, 33, 10), dropout=0.0).eval()
    x = torch.randn(8, 1, 28, 28)
    tgt = torch.randint(0, 10, (8,))

    out = m(x)
    # torch replica with identical parameters
    h = x.reshape(8, 784)
    for i, lin in enumerate(m.layers):
        h = F.linear(h, lin.weight, lin.bias)
        if i < len(m.layers) - 1:
            h = F.relu(h)
    ref = F.log_softmax(h, dim=1)
    assert torch.allclose(out, ref, atol=1e-5)

    loss = F.nll_loss(out, tgt)
    loss.backward()
    h2 = x.reshape(8, 784)
    ps = [(lin.weight.clone().detach().requires_grad_(True),
           lin.bias.clone().detach().requires_grad_(True))
          for lin in m.layers]
    for i, (w, b) in enumerate(ps):
        h2 = F.linear(h2, w, b)
        if i < len(ps) - 1:
            h2 = F.relu(h2)
    F.nll_loss(F.log_softmax(h2, dim=1), tgt).backward()
    for lin, (w, b) in zip(m.layers, ps):
        assert torch.allclose(lin.weight.grad, w.grad, atol=1e-4)
        assert torch.allclose(lin.bias.grad, b.grad, atol=1e-4)


def test_mlp_trains_on_fixed_batch():
    import torch
    import torch.nn.functional as F
    from dist_tuto_pth_amd.models import MLP
    from dist_tuto_pth_amd.optim import FusedSGD

    torch.manual_seed(4)
    m = MLP((784, 64, 10), dropout=0.0)
    opt = FusedSGD(m.parameters(), lr=0.1, momentum=0.5)
    x = torch.randn(64, 784)
    tgt = torch.randint(0, 10, (64,))
    losses = []
    for _ in range(30):
        opt.zero_grad()
        loss = F.nll_loss(m(x), tgt)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] - 0.3, (losses[0], losses[-1])


import pytest  # noqa: E402


@pytest.mark.gpu
def test_mlp_gpu_matches_cpu():
    import torch
    import torch.nn.functional as F
    from dist_tuto_pth_amd.models import MLP
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(5)
    m = MLP((784, 200, 10), dropout=0.0).eval()
    mg = MLP((784, 200, 10), dropout=0.0).eval().cuda()
    mg.load_state_dict({k: v.cuda() for k, v in m.state_dict().items()})
    x = torch.randn(16, 784)
    tgt = torch.randint(0, 10, (16,))
    loss_c = F.nll_loss(m(x), tgt)
    loss_c.backward()
    loss_g = F.nll_loss(mg(x.cuda()), tgt.cuda())
    loss_g.backward()
    torch.cuda.synchronize()
    assert torch.allclose(loss_g.cpu(), loss_c, atol=1e-5)
    for pc, pg in zip(m.parameters(), mg.parameters()):
        assert torch.allclose(pg.grad.cpu(), pc.grad, atol=1e-4)
