"""Training-path ops.

Every op the reference's ``Net.forward`` + SGD step trigger implicitly
through torch (train_dist.py:58-71,110,118-124) is a hand-written CDNA4
HIP kernel here (``csrc/kernels.hip``), with forward *and* backward:

  K1/K2  conv2d (direct, LDS-tiled)        -> conv2d
  K3+K4  maxpool2d(k2) fused with ReLU     -> maxpool2d_relu
  K5/K6  dropout2d (channel) / dropout     -> dropout2d / dropout
  K7/K8  linear (+fused ReLU epilogue)     -> linear
  K9+K10 log_softmax fused with NLL loss   -> log_softmax, nll_loss,
                                              log_softmax_nll
  K11-13 fused multi-tensor SGD+momentum   -> optim.FusedSGD

Dispatch: GPU tensors REQUIRE the native extension (loud failure, no
eager fallback — the HIP path must be the one that runs on a GPU box);
CPU tensors run a plain-torch fp32 reference implementation of the same
op, which is also the golden model for the GPU numerics tests
(tests/test_ops_gpu.py).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ..utils.native import load_native


def _k():
    return load_native("_kernels")


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


# ---------------------------------------------------------------------------
# K1/K2 — direct convolution (no padding, stride 1: the only form Net uses)
# ---------------------------------------------------------------------------
class _Conv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        if x.is_cuda:
            k = _k()
            B, C, H, W = x.shape
            K, _, R, S = w.shape
            OH, OW = H - R + 1, W - S + 1
            out = torch.empty(B, K, OH, OW, device=x.device, dtype=x.dtype)
            k.conv2d_fwd(x.data_ptr(), w.data_ptr(),
                         b.data_ptr() if b is not None else 0,
                         out.data_ptr(), B, C, H, W, K, R, S, _stream())
            return out
        return F.conv2d(x, w, b)

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        gy = gy.contiguous()
        if x.is_cuda:
            k = _k()
            B, C, H, W = x.shape
            K, _, R, S = w.shape
            gx = torch.empty_like(x)
            gw = torch.empty_like(w)   # fully written by the kernel
            gb = torch.empty(K, device=x.device, dtype=x.dtype) \
                if ctx.needs_input_grad[2] else None
            k.conv2d_bwd(x.data_ptr(), w.data_ptr(), gy.data_ptr(),
                         gx.data_ptr(), gw.data_ptr(),
                         gb.data_ptr() if gb is not None else 0,
                         B, C, H, W, K, R, S, _stream())
            return gx, gw, gb
        gx = torch.nn.grad.conv2d_input(x.shape, w, gy)
        gw = torch.nn.grad.conv2d_weight(x, w.shape, gy)
        gb = gy.sum(dim=(0, 2, 3)) if ctx.needs_input_grad[2] else None
        return gx, gw, gb


def conv2d(x, w, b=None):
    return _Conv2d.apply(x.contiguous(), w, b)


# ---------------------------------------------------------------------------
# K3+K4 — fused 2x2 maxpool + ReLU (train_dist.py:65-66 applies
# relu(max_pool2d(.., 2))); backward routes grad to the argmax where the
# pooled max was positive.
# ---------------------------------------------------------------------------
class _MaxPool2dRelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            k = _k()
            B, C, H, W = x.shape
            OH, OW = H // 2, W // 2
            out = torch.empty(B, C, OH, OW, device=x.device, dtype=x.dtype)
            idx = torch.empty(B, C, OH, OW, device=x.device,
                              dtype=torch.int32)
            k.maxpool2d_relu_fwd(x.data_ptr(), out.data_ptr(),
                                 idx.data_ptr(), B, C, H, W, _stream())
            ctx.save_for_backward(idx)
            ctx.in_shape = x.shape
            return out
        pooled, idx = F.max_pool2d(x, 2, return_indices=True)
        out = F.relu(pooled)
        ctx.save_for_backward(idx, pooled)
        ctx.in_shape = x.shape
        return out

    @staticmethod
    def backward(ctx, gy):
        gy = gy.contiguous()
        if gy.is_cuda:
            (idx,) = ctx.saved_tensors
            k = _k()
            B, C, H, W = ctx.in_shape
            gx = torch.empty(ctx.in_shape, device=gy.device,
                             dtype=gy.dtype)  # kernel writes all slots
            k.maxpool2d_relu_bwd(gy.data_ptr(), idx.data_ptr(),
                                 gx.data_ptr(), B, C, H, W, _stream())
            return gx
        idx, pooled = ctx.saved_tensors
        gy = gy * (pooled > 0)
        return F.max_unpool2d(gy, idx, 2, output_size=ctx.in_shape[-2:])


def maxpool2d_relu(x):
    return _MaxPool2dRelu.apply(x.contiguous())


def relu(x):
    return _Relu.apply(x.contiguous())


class _Relu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            k = _k()
            out = torch.empty_like(x)
            k.relu_fwd(x.data_ptr(), out.data_ptr(), x.numel(), _stream())
            ctx.save_for_backward(out)
            return out
        out = F.relu(x)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, gy):
        (out,) = ctx.saved_tensors
        gy = gy.contiguous()
        if gy.is_cuda:
            k = _k()
            gx = torch.empty_like(gy)
            k.relu_bwd(gy.data_ptr(), out.data_ptr(), gx.data_ptr(),
                       gy.numel(), _stream())
            return gx
        return gy * (out > 0)


# ---------------------------------------------------------------------------
# K5/K6 — dropout2d (per-channel mask, train_dist.py:60,66) and
# elementwise dropout (train_dist.py:69), device-side philox-style RNG.
# ---------------------------------------------------------------------------
# per-device RNG seed counter living in DEVICE memory: the dropout
# kernels bump-and-read it on the stream, so masks advance correctly
# across hipGraph replays (host-side seeds would be frozen at capture).
_seed_bufs = {}


def _seed_ptr(device) -> int:
    key = device.index
    if key not in _seed_bufs:
        _seed_bufs[key] = torch.tensor([12345], dtype=torch.int64,
                                       device=device)
    return _seed_bufs[key].data_ptr()


class _Dropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, training, channelwise):
        if not training or p == 0.0:
            ctx.mask = None
            return x
        scale = 1.0 / (1.0 - p)
        if x.is_cuda:
            k = _k()
            out = torch.empty_like(x)
            if channelwise:
                B, C = x.shape[0], x.shape[1]
                hw = x.numel() // (B * C)
                mask = torch.empty(B * C, device=x.device,
                                   dtype=torch.uint8)
                k.dropout2d_fwd(x.data_ptr(), out.data_ptr(),
                                mask.data_ptr(), B * C, hw, p,
                                _seed_ptr(x.device), _stream())
            else:
                mask = torch.empty(x.numel(), device=x.device,
                                   dtype=torch.uint8)
                k.dropout_fwd(x.data_ptr(), out.data_ptr(),
                              mask.data_ptr(), x.numel(), p,
                              _seed_ptr(x.device), _stream())
            ctx.save_for_backward(mask)
            ctx.meta = (scale, channelwise, x.shape)
            ctx.mask = mask
            return out
        # CPU reference
        if channelwise:
            B, C = x.shape[0], x.shape[1]
            mask = (torch.rand(B, C, device=x.device) >= p).to(x.dtype)
            out = x * mask.view(B, C, *([1] * (x.dim() - 2))) * scale
        else:
            mask = (torch.rand_like(x) >= p).to(x.dtype)
            out = x * mask * scale
        ctx.save_for_backward(mask)
        ctx.meta = (scale, channelwise, x.shape)
        ctx.mask = mask
        return out

    @staticmethod
    def backward(ctx, gy):
        if ctx.mask is None:
            return gy, None, None, None
        (mask,) = ctx.saved_tensors
        scale, channelwise, shape = ctx.meta
        gy = gy.contiguous()
        if gy.is_cuda:
            k = _k()
            gx = torch.empty_like(gy)
            if channelwise:
                B, C = shape[0], shape[1]
                hw = gy.numel() // (B * C)
                k.dropout2d_bwd(gy.data_ptr(), mask.data_ptr(),
                                gx.data_ptr(), B * C, hw, scale, _stream())
            else:
                k.dropout_bwd(gy.data_ptr(), mask.data_ptr(), gx.data_ptr(),
                              gy.numel(), scale, _stream())
            return gx, None, None, None
        if channelwise:
            B, C = shape[0], shape[1]
            gx = gy * mask.view(B, C, *([1] * (gy.dim() - 2))) * scale
        else:
            gx = gy * mask * scale
        return gx, None, None, None


def dropout(x, p=0.5, training=True):
    return _Dropout.apply(x.contiguous(), p, training, False)


def dropout2d(x, p=0.5, training=True):
    return _Dropout.apply(x.contiguous(), p, training, True)


# ---------------------------------------------------------------------------
# K7/K8 — linear with optional fused-ReLU epilogue (fc1 path is
# relu(fc1(x)), train_dist.py:68)
# ---------------------------------------------------------------------------
class _Linear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, fuse_relu):
        if x.is_cuda:
            k = _k()
            B, K = x.shape
            N = w.shape[0]
            out = torch.empty(B, N, device=x.device, dtype=x.dtype)
            k.linear_fwd(x.data_ptr(), w.data_ptr(),
                         b.data_ptr() if b is not None else 0,
                         out.data_ptr(), B, K, N, fuse_relu, _stream())
        else:
            out = F.linear(x, w, b)
            if fuse_relu:
                out = F.relu(out)
        ctx.save_for_backward(x, w, out)
        ctx.fuse_relu = fuse_relu
        ctx.has_bias = b is not None
        return out

    @staticmethod
    def backward(ctx, gy):
        x, w, out = ctx.saved_tensors
        gy = gy.contiguous()
        if x.is_cuda:
            k = _k()
            B, K = x.shape
            N = w.shape[0]
            gx = torch.empty_like(x)
            gw = torch.empty_like(w)
            gb = torch.empty(N, device=x.device, dtype=x.dtype) \
                if ctx.has_bias else None
            k.linear_bwd(x.data_ptr(), w.data_ptr(), gy.data_ptr(),
                         out.data_ptr() if ctx.fuse_relu else 0,
                         gx.data_ptr(), gw.data_ptr(),
                         gb.data_ptr() if gb is not None else 0,
                         B, K, N, _stream())
            return gx, gw, gb, None
        if ctx.fuse_relu:
            gy = gy * (out > 0)
        gx = gy @ w
        gw = gy.t() @ x
        gb = gy.sum(0) if ctx.has_bias else None
        return gx, gw, gb, None


def linear(x, w, b=None, fuse_relu=False):
    return _Linear.apply(x.contiguous(), w, b, fuse_relu)


# ---------------------------------------------------------------------------
# K9+K10 — log_softmax over dim 1 (train_dist.py:71) and NLL loss
# (train_dist.py:120), plus the fused form used by the trainer.
# ---------------------------------------------------------------------------
class _LogSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            k = _k()
            B, N = x.shape
            out = torch.empty_like(x)
            k.log_softmax_fwd(x.data_ptr(), out.data_ptr(), B, N, _stream())
        else:
            out = F.log_softmax(x, dim=1)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, gy):
        (out,) = ctx.saved_tensors
        gy = gy.contiguous()
        if out.is_cuda:
            k = _k()
            B, N = out.shape
            gx = torch.empty_like(out)
            k.log_softmax_bwd(gy.data_ptr(), out.data_ptr(), gx.data_ptr(),
                              B, N, _stream())
            return gx
        return gy - out.exp() * gy.sum(dim=1, keepdim=True)


def log_softmax(x):
    return _LogSoftmax.apply(x.contiguous())


class _NllLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logp, target):
        ctx.save_for_backward(logp, target)
        if logp.is_cuda:
            k = _k()
            B, N = logp.shape
            loss = torch.empty((), device=logp.device, dtype=logp.dtype)
            k.nll_loss_fwd(logp.data_ptr(), target.data_ptr(),
                           loss.data_ptr(), B, N, _stream())
            return loss
        return F.nll_loss(logp, target)

    @staticmethod
    def backward(ctx, gloss):
        logp, target = ctx.saved_tensors
        if logp.is_cuda:
            k = _k()
            B, N = logp.shape
            gx = torch.zeros_like(logp)
            gl = gloss.contiguous()
            k.nll_loss_bwd(target.data_ptr(), gx.data_ptr(),
                           gl.data_ptr(), B, N, _stream())
            return gx, None
        B, N = logp.shape
        gx = torch.zeros_like(logp)
        gx[torch.arange(B), target] = -gloss / B
        return gx, None


def nll_loss(logp, target):
    return _NllLoss.apply(logp.contiguous(), target.contiguous())


class _LogSoftmaxNll(torch.autograd.Function):
    """Fused K9+K10: one kernel computes log_softmax and the mean NLL;
    backward is the closed form softmax(x) - onehot, scaled."""

    @staticmethod
    def forward(ctx, logits, target):
        if logits.is_cuda:
            k = _k()
            B, N = logits.shape
            logp = torch.empty_like(logits)
            loss = torch.empty((), device=logits.device, dtype=logits.dtype)
            k.log_softmax_nll_fwd(logits.data_ptr(), target.data_ptr(),
                                  logp.data_ptr(), loss.data_ptr(),
                                  B, N, _stream())
        else:
            logp = F.log_softmax(logits, dim=1)
            loss = F.nll_loss(logp, target)
        ctx.save_for_backward(logp, target)
        return loss

    @staticmethod
    def backward(ctx, gloss):
        logp, target = ctx.saved_tensors
        B, N = logp.shape
        if logp.is_cuda:
            k = _k()
            gx = torch.empty_like(logp)
            gl = gloss.contiguous()
            k.log_softmax_nll_bwd(logp.data_ptr(), target.data_ptr(),
                                  gx.data_ptr(), gl.data_ptr(), B, N,
                                  _stream())
            return gx, None
        gx = logp.exp()
        gx[torch.arange(B), target] -= 1.0
        return gx * (gloss / B), None


def log_softmax_nll(logits, target):
    return _LogSoftmaxNll.apply(logits.contiguous(), target.contiguous())
