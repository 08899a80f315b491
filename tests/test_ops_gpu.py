"""GPU numerics tests: each CDNA4 HIP kernel (fwd+bwd) against the plain
torch fp32 reference of the same op (SURVEY.md §4 'unit tests per HIP
kernel')."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from dist_tuto_pth_amd import ops
    from dist_tuto_pth_amd.utils.native import load_native
else:
    pytest.skip("no GPU", allow_module_level=True)

import torch.nn.functional as F

DEV = "cuda:0"


@pytest.fixture(scope="module", autouse=True)
def _native_loaded():
    # the HIP extension must actually load — no silent fallback
    load_native("_kernels")


def _cmp_fwd_bwd(fn_gpu, fn_ref, tensors, atol=1e-4, rtol=1e-4):
    gpu_in = [t.detach().to(DEV).requires_grad_(t.requires_grad)
              for t in tensors]
    cpu_in = [t.detach().clone().requires_grad_(t.requires_grad)
              for t in tensors]
    out_g = fn_gpu(*gpu_in)
    out_c = fn_ref(*cpu_in)
    assert torch.allclose(out_g.cpu(), out_c, atol=atol, rtol=rtol), \
        (out_g.cpu() - out_c).abs().max()
    gseed = torch.randn_like(out_c)
    out_g.backward(gseed.to(DEV))
    out_c.backward(gseed)
    for a, b in zip(gpu_in, cpu_in):
        if a.requires_grad:
            assert torch.allclose(a.grad.cpu(), b.grad, atol=atol,
                                  rtol=rtol), \
                (a.grad.cpu() - b.grad).abs().max()


def test_conv2d_k1_shape():
    torch.manual_seed(0)
    x = torch.randn(16, 1, 28, 28).requires_grad_(True)
    w = torch.randn(10, 1, 5, 5).requires_grad_(True)
    b = torch.randn(10).requires_grad_(True)
    _cmp_fwd_bwd(lambda x_, w_, b_: ops.conv2d(x_, w_, b_),
                 lambda x_, w_, b_: F.conv2d(x_, w_, b_), [x, w, b])


def test_conv2d_k2_shape():
    torch.manual_seed(1)
    x = torch.randn(16, 10, 12, 12).requires_grad_(True)
    w = torch.randn(20, 10, 5, 5).requires_grad_(True)
    b = torch.randn(20).requires_grad_(True)
    _cmp_fwd_bwd(lambda x_, w_, b_: ops.conv2d(x_, w_, b_),
                 lambda x_, w_, b_: F.conv2d(x_, w_, b_), [x, w, b],
                 atol=5e-4)


def test_conv2d_large_batch():
    torch.manual_seed(2)
    x = torch.randn(512, 1, 28, 28).requires_grad_(True)
    w = torch.randn(10, 1, 5, 5).requires_grad_(True)
    b = torch.randn(10).requires_grad_(True)
    _cmp_fwd_bwd(lambda x_, w_, b_: ops.conv2d(x_, w_, b_),
                 lambda x_, w_, b_: F.conv2d(x_, w_, b_), [x, w, b],
                 atol=2e-3)


def test_maxpool_relu():
    torch.manual_seed(3)
    x = torch.randn(32, 10, 24, 24).requires_grad_(True)
    _cmp_fwd_bwd(ops.maxpool2d_relu,
                 lambda x_: F.relu(F.max_pool2d(x_, 2)), [x])


def test_relu():
    torch.manual_seed(4)
    x = torch.randn(1000003).requires_grad_(True)  # odd size: tail path
    _cmp_fwd_bwd(ops.relu, F.relu, [x])


def test_linear():
    torch.manual_seed(5)
    x = torch.randn(128, 320).requires_grad_(True)
    w = torch.randn(50, 320).requires_grad_(True)
    b = torch.randn(50).requires_grad_(True)
    _cmp_fwd_bwd(lambda x_, w_, b_: ops.linear(x_, w_, b_),
                 lambda x_, w_, b_: F.linear(x_, w_, b_), [x, w, b],
                 atol=1e-3)


def test_linear_fused_relu():
    torch.manual_seed(6)
    x = torch.randn(128, 320).requires_grad_(True)
    w = torch.randn(50, 320).requires_grad_(True)
    b = torch.randn(50).requires_grad_(True)
    _cmp_fwd_bwd(
        lambda x_, w_, b_: ops.linear(x_, w_, b_, fuse_relu=True),
        lambda x_, w_, b_: F.relu(F.linear(x_, w_, b_)), [x, w, b],
        atol=1e-3)


def test_log_softmax():
    torch.manual_seed(7)
    x = torch.randn(256, 10).requires_grad_(True)
    _cmp_fwd_bwd(ops.log_softmax, lambda x_: F.log_softmax(x_, dim=1), [x])


def test_nll_loss():
    torch.manual_seed(8)
    x = torch.randn(256, 10)
    tgt = torch.randint(0, 10, (256,))
    xg = x.to(DEV).requires_grad_(True)
    xc = x.clone().requires_grad_(True)
    lg = ops.nll_loss(ops.log_softmax(xg), tgt.to(DEV))
    lc = F.nll_loss(F.log_softmax(xc, dim=1), tgt)
    assert torch.allclose(lg.cpu(), lc, atol=1e-5)
    lg.backward()
    lc.backward()
    assert torch.allclose(xg.grad.cpu(), xc.grad, atol=1e-5)


def test_fused_log_softmax_nll():
    torch.manual_seed(9)
    x = torch.randn(512, 10)
    tgt = torch.randint(0, 10, (512,))
    xg = x.to(DEV).requires_grad_(True)
    xc = x.clone().requires_grad_(True)
    lg = ops.log_softmax_nll(xg, tgt.to(DEV))
    lc = F.nll_loss(F.log_softmax(xc, dim=1), tgt)
    assert torch.allclose(lg.cpu(), lc, atol=1e-5)
    lg.backward()
    lc.backward()
    assert torch.allclose(xg.grad.cpu(), xc.grad, atol=1e-5)


def test_dropout_gpu_stats():
    x = torch.ones(100000, device=DEV, requires_grad=True)
    out = ops.dropout(x, p=0.5, training=True)
    kept = (out != 0)
    frac = kept.float().mean().item()
    assert 0.45 < frac < 0.55
    assert torch.allclose(out[kept],
                          torch.full((int(kept.sum()),), 2.0, device=DEV))
    out.sum().backward()
    # grad = 2 where kept else 0
    assert torch.allclose(x.grad[kept],
                          torch.full((int(kept.sum()),), 2.0, device=DEV))
    assert torch.all(x.grad[~kept] == 0)


def test_dropout2d_gpu_channelwise():
    x = torch.ones(16, 20, 4, 4, device=DEV)
    out = ops.dropout2d(x, p=0.5, training=True)
    flat = out.reshape(16 * 20, -1)
    zero = (flat == 0).all(dim=1)
    scaled = (flat == 2.0).all(dim=1)
    assert torch.all(zero | scaled)
    assert 0.25 < zero.float().mean().item() < 0.75


def test_sgd_step_gpu_matches_torch():
    torch.manual_seed(10)
    from dist_tuto_pth_amd.optim import FusedSGD
    shapes = [(10, 1, 5, 5), (10,), (20, 10, 5, 5), (20,), (50, 320),
              (50,), (10, 50), (10,)]
    pa = [torch.randn(s, device=DEV).requires_grad_(True) for s in shapes]
    pb = [p.detach().cpu().clone().requires_grad_(True) for p in pa]
    oa = FusedSGD(pa, lr=0.01, momentum=0.5)
    ob = torch.optim.SGD(pb, lr=0.01, momentum=0.5)
    for step in range(3):
        gs = [torch.randn(s) for s in shapes]
        for p, gr in zip(pa, gs):
            p.grad = gr.to(DEV)
        for p, gr in zip(pb, gs):
            p.grad = gr.clone()
        oa.step()
        ob.step()
        for a, b in zip(pa, pb):
            assert torch.allclose(a.detach().cpu(), b.detach(), atol=1e-6)


def test_add_inplace_f32_and_bf16():
    k = load_native("_kernels")
    s = torch.cuda.current_stream().cuda_stream
    for n in (1 << 20, (1 << 20) + 3):
        a = torch.randn(n, device=DEV)
        b = torch.randn(n, device=DEV)
        ref = a + b
        k.add_inplace(a.data_ptr(), b.data_ptr(), n, 7, s)
        torch.cuda.synchronize()
        assert torch.allclose(a, ref)
    for n in (1 << 20, (1 << 20) + 5):
        a = torch.randn(n, device=DEV, dtype=torch.bfloat16)
        b = torch.randn(n, device=DEV, dtype=torch.bfloat16)
        ref = a + b
        k.add_inplace(a.data_ptr(), b.data_ptr(), n, 9, s)
        torch.cuda.synchronize()
        assert torch.allclose(a.float(), ref.float(), atol=1e-2)


def test_net_forward_backward_gpu_vs_cpu():
    from dist_tuto_pth_amd.models import Net
    torch.manual_seed(11)
    net_c = Net().eval()           # eval: no dropout randomness
    net_g = Net().eval().to(DEV)
    net_g.load_state_dict({k: v.to(DEV)
                           for k, v in net_c.state_dict().items()})
    x = torch.randn(64, 1, 28, 28)
    tgt = torch.randint(0, 10, (64,))
    lc = F.nll_loss(net_c(x), tgt)
    lg = F.nll_loss(net_g(x.to(DEV)), tgt.to(DEV))
    assert torch.allclose(lg.cpu(), lc, atol=1e-4)
    lc.backward()
    lg.backward()
    for pc, pg in zip(net_c.parameters(), net_g.parameters()):
        assert torch.allclose(pg.grad.cpu(), pc.grad, atol=1e-3), \
            (pg.grad.cpu() - pc.grad).abs().max()


def test_conv2d_large_weights_unstaged():
    """Weights bigger than the 64 KB LDS budget take the global-read
    path (staged=0) instead of failing the launch."""
    torch.manual_seed(20)
    x = torch.randn(4, 32, 16, 16, device=DEV, requires_grad=True)
    w = torch.randn(64, 32, 5, 5, device=DEV, requires_grad=True)  # 204 KB
    b = torch.randn(64, device=DEV, requires_grad=True)
    out = ops.conv2d(x, w, b)
    xc = x.detach().cpu().requires_grad_(True)
    wc = w.detach().cpu().requires_grad_(True)
    bc = b.detach().cpu().requires_grad_(True)
    ref = F.conv2d(xc, wc, bc)
    assert torch.allclose(out.cpu(), ref, atol=2e-3, rtol=1e-3)
    out.sum().backward()
    ref.sum().backward()
    assert torch.allclose(x.grad.cpu(), xc.grad, atol=2e-3, rtol=1e-3)
    assert torch.allclose(w.grad.cpu(), wc.grad, atol=2e-2, rtol=1e-3)


def test_linear_large_weights_unstaged():
    torch.manual_seed(21)
    x = torch.randn(16, 784, device=DEV, requires_grad=True)
    w = torch.randn(200, 784, device=DEV, requires_grad=True)  # 627 KB
    b = torch.randn(200, device=DEV, requires_grad=True)
    out = ops.linear(x, w, b)
    xc = x.detach().cpu().requires_grad_(True)
    wc = w.detach().cpu().requires_grad_(True)
    bc = b.detach().cpu().requires_grad_(True)
    ref = F.linear(xc, wc, bc)
    assert torch.allclose(out.cpu(), ref, atol=2e-3, rtol=1e-3)
    out.sum().backward()
    ref.sum().backward()
    assert torch.allclose(x.grad.cpu(), xc.grad, atol=2e-3, rtol=1e-3)
    assert torch.allclose(w.grad.cpu(), wc.grad, atol=2e-2, rtol=1e-3)
