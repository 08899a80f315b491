#!/usr/bin/env python3
"""A/B the in-launch gw fold (DTP_GW_FOLD) on the two step shapes.

Shape "sgd-fused" is the world-1 flagship step (net_fused_step_opt:
fwd / bwd / partial / combine+SGD one dispatch).  Shape "ddp" is the
per-rank compute of the world>1 DP step (net_fused_step: combine
WITHOUT SGD, then the optimizer as its own dispatch after the — here
absent — all-reduce), i.e. what every rank pays on the driver's
multi-GPU scaling bench.  The fold variant folds the combine into the
partial kernel's last-arriving blocks (kernels.hip), so it deletes one
dispatch from "ddp" and the combine+SGD dispatch from "sgd-fused".

Run both shapes with DTP_GW_FOLD=0/1 before importing the package (the
flag is read once per process), e.g.
    DTP_GW_FOLD=1 python benchmarks/fold_ab.py --shape ddp
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from dist_tuto_pth_amd.models import Net  # noqa: E402
from dist_tuto_pth_amd.optim import FusedSGD  # noqa: E402
from dist_tuto_pth_amd.ops.fused import (  # noqa: E402
    attach_flat_grads, net_fused_step, net_fused_step_opt)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--shape", choices=["sgd-fused", "ddp"], default="ddp")
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--warmup", type=int, default=50)
    args = ap.parse_args()

    torch.manual_seed(1234)
    dev = "cuda:0"
    model = Net().to(dev)
    opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.5,
                   zero_grad_in_step=True)
    attach_flat_grads(model)
    g = torch.Generator(device="cpu").manual_seed(7)
    x = torch.randn(args.batch, 1, 28, 28, generator=g).to(dev)
    tgt = torch.randint(0, 10, (args.batch,), generator=g).to(dev)

    def step():
        if args.shape == "sgd-fused":
            net_fused_step_opt(model, x, tgt, opt)
        else:
            net_fused_step(model, x, tgt)
            opt.step()

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    print(f"shape={args.shape} fold={os.environ.get('DTP_GW_FOLD', 'def')}"
          f" B={args.batch} us_per_step={dt * 1e6:.2f}"
          f" samples_per_sec={args.batch / dt:.0f}")


if __name__ == "__main__":
    main()
