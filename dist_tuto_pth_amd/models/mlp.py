"""MLP on the same CDNA4 HIP op layer as `Net` — demonstrates that the
op layer (`dist_tuto_pth_amd.ops`) generalizes beyond the reference's
exact ConvNet shapes: arbitrary-width `linear` (+fused ReLU), dropout,
and the fused log_softmax+NLL loss all take any (B, K, N).

Not part of the reference surface (the tutorial ships exactly one
model, train_dist.py:53-71); provided as the second model family for
the training loop and the DP path.
"""

from __future__ import annotations

from typing import Sequence

import torch
import torch.nn as nn

from .. import ops


class MLP(nn.Module):
    """`widths` = [in, hidden..., out]; ReLU (fused into the linear
    kernel) + dropout between layers, log_softmax output."""

    def __init__(self, widths: Sequence[int] = (784, 256, 128, 10),
                 dropout: float = 0.2):
        super().__init__()
        assert len(widths) >= 2
        self.widths = list(widths)
        self.p = dropout
        self.layers = nn.ModuleList(
            nn.Linear(a, b) for a, b in zip(widths[:-1], widths[1:]))

    def forward_logits(self, x: torch.Tensor) -> torch.Tensor:
        x = x.reshape(x.shape[0], self.widths[0])
        for i, lin in enumerate(self.layers):
            last = i == len(self.layers) - 1
            x = ops.linear(x, lin.weight, lin.bias, fuse_relu=not last)
            if not last:
                x = ops.dropout(x, p=self.p, training=self.training)
        return x

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.log_softmax(self.forward_logits(x))
