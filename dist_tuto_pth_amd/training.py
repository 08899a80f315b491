"""The distributed synchronous-SGD training loop (L5; the ``run`` of
train_dist.py:103-127).

Semantics preserved: seed 1234, SGD lr=0.01 momentum=0.5, per step
zero_grad -> forward -> NLL -> backward -> average_gradients ->
optimizer.step, per-rank mean epoch loss reported — with the reference's
bugs fixed (working average_gradients, no
graph-holding loss accumulation, one rank <-> one GPU).

Two gradient-sync modes:
  * ``mode="average_gradients"`` — the reference's per-parameter blocking
    all-reduce after backward (train_dist.py:94-100).
  * ``mode="ddp"`` — the production-shaped path: bucketed all-reduce
    overlapped with backward (tuto.md:216,320).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from . import dist, ops
from .models import Net
from .optim import FusedSGD
from .parallel import DistributedDataParallel, average_gradients, \
    partition_dataset


def run(rank: int, size: int, epochs: int = 10, device: str = "cpu",
        mode: str = "average_gradients", batch_size: int = 128,
        dataset=None, log=None, steps_per_epoch: Optional[int] = None,
        fused_loss: Optional[bool] = None) -> List[float]:
    """Train the ConvNet with synchronous distributed SGD; returns the
    per-epoch mean losses (the reference prints them,
    train_dist.py:125-127 — identical across ranks is the correctness
    signal)."""
    torch.manual_seed(1234)
    train_set, bsz = partition_dataset(dataset, batch_size=batch_size)
    model = Net().to(device)
    # replicas start identical: same seed => same init on every rank
    # (train_dist.py:105); broadcast pins it even if seeds diverge.
    if size > 1:
        for p in model.parameters():
            dist.broadcast(p.data, src=0)
    ddp = None
    if mode == "ddp":
        ddp = DistributedDataParallel(model)
    optimizer = FusedSGD(model.parameters(), lr=0.01, momentum=0.5)
    if fused_loss is None:
        fused_loss = device != "cpu"

    num_batches = len(train_set)
    losses = []
    for epoch in range(epochs):
        epoch_loss = 0.0
        nsteps = 0
        for data, target in train_set:
            data, target = data.to(device), target.to(device)
            optimizer.zero_grad()
            if ddp is not None:
                loss = ops.nll_loss(ddp(data), target)
                loss.backward()
                ddp.finish_gradients()
            else:
                if fused_loss:
                    loss = ops.log_softmax_nll(model.forward_logits(data),
                                               target)
                else:
                    loss = ops.nll_loss(model(data), target)
                loss.backward()
                average_gradients(model)
            optimizer.step()
            epoch_loss += loss.item()   # scalar, not a graph-holding
            nsteps += 1                 # tensor
            if steps_per_epoch is not None and nsteps >= steps_per_epoch:
                break
        mean = epoch_loss / max(nsteps, 1)
        losses.append(mean)
        if log is not None:
            log(f"Rank {dist.get_rank()}, epoch {epoch}: {mean}")
    return losses, model
