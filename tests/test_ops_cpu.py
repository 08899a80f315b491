"""CPU-path op tests: the ops module's CPU implementations against plain
torch autograd (these CPU implementations are in turn the golden
reference for the GPU kernels in test_ops_gpu.py)."""

import torch
import torch.nn.functional as F

from dist_tuto_pth_amd import ops


def _grads(fn, *tensors):
    xs = [t.detach().clone().requires_grad_(True) for t in tensors]
    out = fn(*xs)
    out.sum().backward()
    return out.detach(), [x.grad for x in xs]


def test_conv2d_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(4, 1, 28, 28)
    w = torch.randn(10, 1, 5, 5)
    b = torch.randn(10)
    out1, g1 = _grads(lambda x_, w_, b_: ops.conv2d(x_, w_, b_), x, w, b)
    out2, g2 = _grads(lambda x_, w_, b_: F.conv2d(x_, w_, b_), x, w, b)
    assert torch.allclose(out1, out2, atol=1e-5)
    for a, c in zip(g1, g2):
        assert torch.allclose(a, c, atol=1e-4)


def test_maxpool_relu_matches_torch():
    torch.manual_seed(1)
    x = torch.randn(4, 10, 24, 24)
    out1, g1 = _grads(ops.maxpool2d_relu, x)
    out2, g2 = _grads(lambda x_: F.relu(F.max_pool2d(x_, 2)), x)
    assert torch.allclose(out1, out2)
    assert torch.allclose(g1[0], g2[0])


def test_relu_matches_torch():
    x = torch.randn(100)
    out1, g1 = _grads(ops.relu, x)
    out2, g2 = _grads(F.relu, x)
    assert torch.equal(out1, out2)
    assert torch.equal(g1[0], g2[0])


def test_linear_matches_torch():
    torch.manual_seed(2)
    x = torch.randn(8, 320)
    w = torch.randn(50, 320)
    b = torch.randn(50)
    out1, g1 = _grads(lambda x_, w_, b_: ops.linear(x_, w_, b_), x, w, b)
    out2, g2 = _grads(lambda x_, w_, b_: F.linear(x_, w_, b_), x, w, b)
    assert torch.allclose(out1, out2, atol=1e-4)
    for a, c in zip(g1, g2):
        assert torch.allclose(a, c, atol=1e-3)


def test_linear_fused_relu_matches_torch():
    torch.manual_seed(3)
    x = torch.randn(8, 32)
    w = torch.randn(16, 32)
    b = torch.randn(16)
    out1, g1 = _grads(
        lambda x_, w_, b_: ops.linear(x_, w_, b_, fuse_relu=True), x, w, b)
    out2, g2 = _grads(
        lambda x_, w_, b_: F.relu(F.linear(x_, w_, b_)), x, w, b)
    assert torch.allclose(out1, out2, atol=1e-5)
    for a, c in zip(g1, g2):
        assert torch.allclose(a, c, atol=1e-4)


def test_log_softmax_matches_torch():
    torch.manual_seed(4)
    x = torch.randn(16, 10)
    out1, g1 = _grads(ops.log_softmax, x)
    out2, g2 = _grads(lambda x_: F.log_softmax(x_, dim=1), x)
    assert torch.allclose(out1, out2, atol=1e-6)
    assert torch.allclose(g1[0], g2[0], atol=1e-6)


def test_nll_and_fused_match_torch():
    torch.manual_seed(5)
    x = torch.randn(16, 10)
    tgt = torch.randint(0, 10, (16,))

    xa = x.clone().requires_grad_(True)
    la = ops.nll_loss(ops.log_softmax(xa), tgt)
    la.backward()

    xb = x.clone().requires_grad_(True)
    lb = F.nll_loss(F.log_softmax(xb, dim=1), tgt)
    lb.backward()

    xc = x.clone().requires_grad_(True)
    lc = ops.log_softmax_nll(xc, tgt)
    lc.backward()

    assert torch.allclose(la, lb, atol=1e-6)
    assert torch.allclose(lc, lb, atol=1e-6)
    assert torch.allclose(xa.grad, xb.grad, atol=1e-6)
    assert torch.allclose(xc.grad, xb.grad, atol=1e-6)


def test_dropout_semantics():
    torch.manual_seed(6)
    x = torch.ones(1000)
    out = ops.dropout(x, p=0.5, training=True)
    kept = (out != 0)
    assert 300 < kept.sum() < 700          # ~half kept
    assert torch.allclose(out[kept], torch.full((int(kept.sum()),), 2.0))
    # eval mode: identity
    assert torch.equal(ops.dropout(x, p=0.5, training=False), x)


def test_dropout2d_channelwise():
    torch.manual_seed(7)
    x = torch.ones(8, 20, 4, 4)
    out = ops.dropout2d(x, p=0.5, training=True)
    # each (b, c) channel is uniformly zero or uniformly scaled
    flat = out.reshape(8 * 20, -1)
    for ch in flat:
        assert torch.all(ch == 0) or torch.allclose(
            ch, torch.full_like(ch, 2.0))


def test_net_forward_shapes_and_loss():
    from dist_tuto_pth_amd.models import Net
    torch.manual_seed(8)
    net = Net()
    x = torch.randn(4, 1, 28, 28)
    out = net(x)
    assert out.shape == (4, 10)
    # log_softmax rows sum to 1 in prob space
    assert torch.allclose(out.exp().sum(1), torch.ones(4), atol=1e-5)
    tgt = torch.randint(0, 10, (4,))
    loss = ops.nll_loss(out, tgt)
    loss.backward()
    assert net.conv1.weight.grad is not None
    assert net.fc2.bias.grad is not None


def test_net_param_count_matches_reference():
    # 21,840 params (SURVEY.md §2.1 'Net')
    from dist_tuto_pth_amd.models import Net
    n = sum(p.numel() for p in Net().parameters())
    assert n == 21840


def test_net_step_available_false_on_cpu():
    """The single-launch cooperative step kernel reports unavailable on
    a CPU-only machine instead of raising (the GPU paths fail loudly,
    the capability probe does not)."""
    from dist_tuto_pth_amd.ops.fused import net_step_available
    import torch
    if not torch.cuda.is_available():
        assert net_step_available() is False
