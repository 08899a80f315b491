"""Fused whole-Net training step (GPU fast path).

Four dispatches per training step instead of ~30 per-op launches:
one kernel for the entire forward (conv1..log_softmax+NLL,
train_dist.py:64-71 + :120, conv+pool register-fused, per-block loss
partials), one for the data backward (sibling-workgroup split,
4-wide register-blocked transposed conv), one segmented partial
weight-gradient kernel ([conv2|conv1x24|fc1|fc2] tiles x batch chunks),
and one combine kernel that also finalizes the loss, advances the
dropout seed, and (single-GPU) applies the SGD+momentum update.
rocprof evidence and the optimization ladder live in profiles/.
Numerically identical to the modular path in eval mode; dropout masks
use the same device-seed stream but a different indexing, so
train-mode losses match only in distribution.

Seed convention (all fused paths): a training step CONSUMES the current
device seed and the combine kernel at the end of backward ADVANCES it —
bump-after everywhere, so the autograd path (net_fused_loss) and the
fused-step paths can interleave without reusing dropout masks.  A
train-mode forward without a matching backward does not advance the
seed.
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch

from ..utils.native import load_native
from . import _seed_ptr, _stream

_ws_cache: Dict[Tuple, Dict[str, torch.Tensor]] = {}


def _ws(B: int, device) -> Dict[str, torch.Tensor]:
    key = (B, device.index)
    w = _ws_cache.get(key)
    if w is None:
        f = lambda *shape: torch.empty(*shape, device=device)  # noqa: E731
        u8 = lambda *shape: torch.empty(*shape, device=device,  # noqa: E731
                                        dtype=torch.uint8)
        w = {
            "p1": f(B, 1440), "idx1": u8(B, 1440), "m2": u8(B, 20),
            "p2": f(B, 320), "idx2": u8(B, 320), "h1": f(B, 50),
            "m3": u8(B, 50), "d3": f(B, 50), "logp": f(B, 10),
            "glog": f(B, 10), "gh1": f(B, 50), "ga2": f(B, 1280),
            # partial rows are GW_ROW = 21840 + 23*260 wide (conv1's
            # 24 weight-grad sub-blocks write disjoint slices;
            # kernels.hip)
            "ga1": f(B, 5760), "part": f(32, 27820), "loss": f(()),
            # loss_part must cover the LARGEST forward grid any caller
            # can launch (grid_for(B,1) <= 4096) — sized once here so
            # every entry point shares a safe buffer regardless of
            # which ran first (advisor r1 finding #1)
            "loss_part": f(4096), "one": torch.ones((), device=device),
        }
        _ws_cache[key] = w
    return w


def attach_flat_grads(net) -> torch.Tensor:
    """Point every parameter's ``.grad`` at a slice of ONE flat buffer
    so the DP gradient average is a single all-reduce with no
    pack/unpack copies (the xGMI-friendly layout: one large message
    instead of 8 tiny ones — SURVEY.md §5 'bucketed').  Returns the
    flat buffer."""
    params = [p for p in net.parameters()]
    total = sum(p.numel() for p in params)
    flat = torch.zeros(total, device=params[0].device)
    off = 0
    for p in params:
        p.grad = flat[off:off + p.numel()].view_as(p)
        off += p.numel()
    return flat


class _NetFusedLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2, wf1, bf1, wf2, bf2, tgt, training):
        k = load_native("_kernels")
        B = x.shape[0]
        ws = _ws(B, x.device)
        k.net_fused_fwd(
            x.data_ptr(), w1.data_ptr(), b1.data_ptr(), w2.data_ptr(),
            b2.data_ptr(), wf1.data_ptr(), bf1.data_ptr(), wf2.data_ptr(),
            bf2.data_ptr(), tgt.data_ptr(),
            ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
            ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
            ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
            ws["loss"].data_ptr(), 0, _seed_ptr(x.device), B, training,
            _stream())
        ctx.save_for_backward(x, w2, wf1, wf2, tgt)
        ctx.meta = (B, training)
        ctx.shapes = [w1.shape, b1.shape, w2.shape, b2.shape, wf1.shape,
                      bf1.shape, wf2.shape, bf2.shape]
        return ws["loss"].clone()

    @staticmethod
    def backward(ctx, gl):
        k = load_native("_kernels")
        x, w2, wf1, wf2, tgt = ctx.saved_tensors
        B, training = ctx.meta
        ws = _ws(B, x.device)
        dev = x.device
        grads = [torch.empty(s, device=dev) for s in ctx.shapes]
        gl = gl.contiguous()
        k.net_fused_bwd(
            x.data_ptr(), w2.data_ptr(), wf1.data_ptr(), wf2.data_ptr(),
            tgt.data_ptr(), gl.data_ptr(),
            ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
            ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
            ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
            ws["glog"].data_ptr(), ws["gh1"].data_ptr(),
            ws["ga2"].data_ptr(), ws["ga1"].data_ptr(),
            ws["part"].data_ptr(),
            *[g.data_ptr() for g in grads], B, training, 0, 0,
            _seed_ptr(x.device), _stream())
        return (None, *grads, None, None)


def net_fused_step(net, x: torch.Tensor, tgt: torch.Tensor) -> torch.Tensor:
    """Forward + backward in one call, OVERWRITING each parameter's
    ``.grad`` (no autograd graph, no accumulate-add kernels — the
    minimal-launch training step used by bench.py).  Returns the loss
    as a device scalar (no host sync)."""
    k = load_native("_kernels")
    B = x.shape[0]
    ws = _ws(B, x.device)
    params = [net.conv1.weight, net.conv1.bias, net.conv2.weight,
              net.conv2.bias, net.fc1.weight, net.fc1.bias,
              net.fc2.weight, net.fc2.bias]
    for p in params:
        if p.grad is None:
            p.grad = torch.empty_like(p)
    s = _stream()
    # loss_part mode: no prologue dispatch — the forward writes
    # per-block loss partials (fwd grid <= 4096 at any B); the combine
    # kernel finalizes the loss scalar and bumps the dropout seed
    lp = ws["loss_part"].data_ptr()
    k.net_fused_fwd(
        x.data_ptr(), *[p.data_ptr() for p in params], tgt.data_ptr(),
        ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
        ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
        ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
        ws["loss"].data_ptr(), lp, _seed_ptr(x.device), B,
        net.training, s)
    k.net_fused_bwd(
        x.data_ptr(), params[2].data_ptr(), params[4].data_ptr(),
        params[6].data_ptr(), tgt.data_ptr(), ws["one"].data_ptr(),
        ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
        ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
        ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
        ws["glog"].data_ptr(), ws["gh1"].data_ptr(), ws["ga2"].data_ptr(),
        ws["ga1"].data_ptr(), ws["part"].data_ptr(),
        *[p.grad.data_ptr() for p in params], B, net.training,
        lp, ws["loss"].data_ptr(), _seed_ptr(x.device), s)
    return ws["loss"]


def net_fused_step_opt(net, x: torch.Tensor, tgt: torch.Tensor,
                       opt) -> torch.Tensor:
    """``net_fused_step`` with the SGD+momentum update fused into the
    combine kernel (one dispatch fewer; single-GPU training only — the
    DP path needs the all-reduce between combine and step).  ``opt``
    must be a FusedSGD; its momentum buffers are updated in-kernel."""
    k = load_native("_kernels")
    B = x.shape[0]
    ws = _ws(B, x.device)
    params = [net.conv1.weight, net.conv1.bias, net.conv2.weight,
              net.conv2.bias, net.fc1.weight, net.fc1.bias,
              net.fc2.weight, net.fc2.bias]
    for p in params:
        if p.grad is None:
            p.grad = torch.empty_like(p)
    s = _stream()
    lp = ws["loss_part"].data_ptr()
    k.net_fused_fwd(
        x.data_ptr(), *[p.data_ptr() for p in params], tgt.data_ptr(),
        ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
        ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
        ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
        ws["loss"].data_ptr(), lp, _seed_ptr(x.device), B,
        net.training, s)
    bufs = [b.data_ptr() for b in opt._bufs] if opt._bufs else []
    k.net_fused_bwd_sgd(
        x.data_ptr(), params[2].data_ptr(), params[4].data_ptr(),
        params[6].data_ptr(), tgt.data_ptr(), ws["one"].data_ptr(),
        ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
        ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
        ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
        ws["glog"].data_ptr(), ws["gh1"].data_ptr(), ws["ga2"].data_ptr(),
        ws["ga1"].data_ptr(), ws["part"].data_ptr(),
        [p.grad.data_ptr() for p in params],
        [p.data_ptr() for p in params], bufs, opt.lr, opt.momentum, B,
        net.training, lp, ws["loss"].data_ptr(), _seed_ptr(x.device), s)
    return ws["loss"]


def net_fused_step_fb(net, x: torch.Tensor, tgt: torch.Tensor,
                      opt=None) -> torch.Tensor:
    """Three-dispatch training step: forward AND data-backward share
    ONE kernel (each workgroup runs its sample's fwd then immediately
    its bwd — valid because the loss gradient is the constant 1 in the
    training loop), then gw-partial, then combine.  With ``opt`` (a
    FusedSGD) the combine also applies the update; without it, grads
    land in ``.grad`` for the DP all-reduce.  Trade-off vs
    net_fused_step: one dispatch + w2 re-staging saved, but the
    backward loses its sibling-workgroup split (grid = B), so at small
    B the chip is underfilled — measured, see profiles/."""
    k = load_native("_kernels")
    B = x.shape[0]
    ws = _ws(B, x.device)
    params = [net.conv1.weight, net.conv1.bias, net.conv2.weight,
              net.conv2.bias, net.fc1.weight, net.fc1.bias,
              net.fc2.weight, net.fc2.bias]
    for p in params:
        if p.grad is None:
            p.grad = torch.empty_like(p)
    if opt is not None:
        prm = [p.data_ptr() for p in params]
        bufs = [b.data_ptr() for b in opt._bufs] if opt._bufs else []
        lr, mu = opt.lr, opt.momentum
    else:
        prm, bufs, lr, mu = [], [], 0.0, 0.0
    k.net_fused_fwdbwd(
        x.data_ptr(), *[p.data_ptr() for p in params], tgt.data_ptr(),
        ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
        ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
        ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
        ws["glog"].data_ptr(), ws["gh1"].data_ptr(), ws["ga2"].data_ptr(),
        ws["ga1"].data_ptr(), ws["part"].data_ptr(),
        [p.grad.data_ptr() for p in params], prm, bufs, lr, mu, B,
        net.training, ws["loss_part"].data_ptr(), ws["loss"].data_ptr(),
        _seed_ptr(x.device), _stream())
    return ws["loss"]


def net_step_available() -> bool:
    """True when the device supports the single-launch cooperative
    training-step kernel (hipLaunchCooperativeKernel)."""
    try:
        return bool(load_native("_kernels").net_step_available())
    except Exception:
        return False


def net_fused_train_step(net, x: torch.Tensor, tgt: torch.Tensor,
                         opt=None, do_sgd: bool = True) -> torch.Tensor:
    """The WHOLE training step — forward, backward, weight-gradient
    reduction and (optionally) the SGD+momentum update — in ONE
    cooperative kernel launch (csrc/kernels.hip net_step_kernel).

    The 6-dispatch fused path pays a ~4.5 us device dispatch floor per
    kernel at the reference's batch sizes; this replaces them with one
    dispatch and grid barriers.  Gradients are still written to each
    parameter's ``.grad`` (so the DP path can all-reduce them when
    ``do_sgd=False``).  ``opt`` must be a FusedSGD when ``do_sgd`` —
    its momentum buffers are updated in-kernel.  Returns the loss as a
    device scalar (no host sync).
    """
    k = load_native("_kernels")
    B = x.shape[0]
    ws = _ws(B, x.device)
    params = [net.conv1.weight, net.conv1.bias, net.conv2.weight,
              net.conv2.bias, net.fc1.weight, net.fc1.bias,
              net.fc2.weight, net.fc2.bias]
    for p in params:
        if p.grad is None:
            p.grad = torch.empty_like(p)
    if do_sgd:
        lr, mu = opt.lr, opt.momentum
        bufs = [b.data_ptr() for b in opt._bufs] if opt._bufs else []
    else:
        lr, mu, bufs = 0.0, 0.0, []
    k.net_step(
        x.data_ptr(), tgt.data_ptr(), ws["one"].data_ptr(),
        ws["p1"].data_ptr(), ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
        ws["p2"].data_ptr(), ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
        ws["m3"].data_ptr(), ws["d3"].data_ptr(), ws["logp"].data_ptr(),
        ws["glog"].data_ptr(), ws["gh1"].data_ptr(), ws["ga2"].data_ptr(),
        ws["ga1"].data_ptr(), ws["part"].data_ptr(),
        ws["loss_part"].data_ptr(), ws["loss"].data_ptr(),
        _seed_ptr(x.device),
        [p.data_ptr() for p in params],
        [p.grad.data_ptr() for p in params],
        bufs, lr, mu, do_sgd, B, net.training, _stream())
    return ws["loss"]


def net_fused_loss(net, x: torch.Tensor, tgt: torch.Tensor) -> torch.Tensor:
    """Mean NLL loss of ``net`` (a models.Net) on (x, tgt), computed by
    the fused kernels.  Gradients flow to the 8 parameters."""
    assert x.is_cuda, "fused path is the GPU fast path"
    return _NetFusedLoss.apply(
        x.contiguous(), net.conv1.weight, net.conv1.bias,
        net.conv2.weight, net.conv2.bias, net.fc1.weight, net.fc1.bias,
        net.fc2.weight, net.fc2.bias, tgt.contiguous(), net.training)
