"""Fused SGD + momentum optimizer (K11-K13 of SURVEY.md §2.4b).

The reference uses ``optim.SGD(lr=0.01, momentum=0.5)``
(train_dist.py:110) with ``optimizer.zero_grad()`` each step
(train_dist.py:118).  Here the whole update — momentum buffer update,
parameter step, and optional gradient zeroing (K12 folded in) — is ONE
multi-tensor HIP kernel launch on GPU; CPU falls back to the plain
formula (the golden reference for the GPU numerics test).

Update rule (torch SGD semantics, as the reference's optimizer):
    buf = mu * buf + grad ;  p -= lr * buf
"""

from __future__ import annotations

from typing import Iterable

import torch

from .utils.native import load_native


class FusedSGD:
    def __init__(self, params: Iterable[torch.Tensor], lr: float = 0.01,
                 momentum: float = 0.0, zero_grad_in_step: bool = False):
        self.params = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no parameters to optimize")
        self.lr = lr
        self.momentum = momentum
        self.zero_grad_in_step = zero_grad_in_step
        self._bufs = [torch.zeros_like(p) for p in self.params] \
            if momentum != 0.0 else None
        self._is_cuda = self.params[0].is_cuda

    @torch.no_grad()
    def step(self):
        if self._is_cuda:
            k = load_native("_kernels")
            ptrs_p, ptrs_g, ptrs_b, numels = [], [], [], []
            for i, p in enumerate(self.params):
                if p.grad is None:
                    continue
                ptrs_p.append(p.data_ptr())
                ptrs_g.append(p.grad.data_ptr())
                ptrs_b.append(self._bufs[i].data_ptr()
                              if self._bufs is not None else 0)
                numels.append(p.numel())
            k.sgd_step(ptrs_p, ptrs_g, ptrs_b, numels, self.lr,
                       self.momentum, self.zero_grad_in_step,
                       torch.cuda.current_stream().cuda_stream)
            return
        for i, p in enumerate(self.params):
            if p.grad is None:
                continue
            g = p.grad
            if self._bufs is not None:
                buf = self._bufs[i]
                buf.mul_(self.momentum).add_(g)
                g = buf
            p.add_(g, alpha=-self.lr)
            if self.zero_grad_in_step:
                p.grad.zero_()

    @torch.no_grad()
    def zero_grad(self):
        for p in self.params:
            if p.grad is not None:
                p.grad.zero_()
