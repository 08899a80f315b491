"""The reference ConvNet (train_dist.py:53-71), built on this package's
CDNA4 HIP ops.

Architecture (21,840 params, SURVEY.md §2.1 'Net'):
  conv1: Conv2d(1,10,k=5)   B x 1 x 28 x 28 -> B x 10 x 24 x 24
  pool+relu (fused)                         -> B x 10 x 12 x 12
  conv2: Conv2d(10,20,k=5)                  -> B x 20 x 8 x 8
  dropout2d -> pool+relu (fused)            -> B x 20 x 4 x 4 -> flat 320
  fc1: Linear(320,50) + fused ReLU -> dropout
  fc2: Linear(50,10) -> log_softmax(dim=1)

The reference's ``log_softmax`` without a ``dim`` argument
(train_dist.py:71) meant dim=1 on 2-D input; made explicit here
(SURVEY.md §2.5.4).  Parameters live in stock ``nn.Conv2d``/``nn.Linear``
holders (same init semantics as the reference); compute goes through
``dist_tuto_pth_amd.ops`` — HIP kernels on GPU, the plain-torch fp32
reference on CPU.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 10, kernel_size=5)
        self.conv2 = nn.Conv2d(10, 20, kernel_size=5)
        self.conv2_drop = nn.Dropout2d()
        self.fc1 = nn.Linear(320, 50)
        self.fc2 = nn.Linear(50, 10)

    def forward(self, x):
        x = ops.conv2d(x, self.conv1.weight, self.conv1.bias)
        x = ops.maxpool2d_relu(x)
        x = ops.conv2d(x, self.conv2.weight, self.conv2.bias)
        x = ops.dropout2d(x, p=self.conv2_drop.p, training=self.training)
        x = ops.maxpool2d_relu(x)
        x = x.reshape(-1, 320)
        x = ops.linear(x, self.fc1.weight, self.fc1.bias, fuse_relu=True)
        x = ops.dropout(x, p=0.5, training=self.training)
        x = ops.linear(x, self.fc2.weight, self.fc2.bias)
        return ops.log_softmax(x)

    def forward_logits(self, x):
        """Forward stopping before log_softmax — pairs with the fused
        ``ops.log_softmax_nll`` loss (K9+K10 fusion)."""
        x = ops.conv2d(x, self.conv1.weight, self.conv1.bias)
        x = ops.maxpool2d_relu(x)
        x = ops.conv2d(x, self.conv2.weight, self.conv2.bias)
        x = ops.dropout2d(x, p=self.conv2_drop.p, training=self.training)
        x = ops.maxpool2d_relu(x)
        x = x.reshape(-1, 320)
        x = ops.linear(x, self.fc1.weight, self.fc1.bias, fuse_relu=True)
        x = ops.dropout(x, p=0.5, training=self.training)
        return ops.linear(x, self.fc2.weight, self.fc2.bias)
