"""FusedSGD must match torch.optim.SGD(lr, momentum) exactly
(train_dist.py:110 semantics)."""

import torch

from dist_tuto_pth_amd.optim import FusedSGD


def test_fused_sgd_matches_torch_sgd():
    torch.manual_seed(0)
    shapes = [(10, 1, 5, 5), (10,), (20, 10, 5, 5), (20,), (50, 320),
              (50,), (10, 50), (10,)]   # Net's 8 tensors
    pa = [torch.randn(s).requires_grad_(True) for s in shapes]
    pb = [p.detach().clone().requires_grad_(True) for p in pa]

    oa = FusedSGD(pa, lr=0.01, momentum=0.5)
    ob = torch.optim.SGD(pb, lr=0.01, momentum=0.5)

    for step in range(5):
        g = [torch.randn(s) for s in shapes]
        for p, gr in zip(pa, g):
            p.grad = gr.clone()
        for p, gr in zip(pb, g):
            p.grad = gr.clone()
        oa.step()
        ob.step()
        for a, b in zip(pa, pb):
            assert torch.allclose(a, b, atol=1e-7), step


def test_fused_sgd_no_momentum_and_zero_grad():
    p = torch.ones(4).requires_grad_(True)
    o = FusedSGD([p], lr=0.1, momentum=0.0)
    p.grad = torch.ones(4)
    o.step()
    assert torch.allclose(p.detach(), torch.full((4,), 0.9))
    o.zero_grad()
    assert torch.all(p.grad == 0)
