#!/usr/bin/env python3
"""Distributed synchronous-SGD training — the rebuilt `train_dist.py`
(reference train_dist.py:103-147, bugs of SURVEY.md §2.5 fixed).

Partitions an MNIST-shaped synthetic dataset across ranks
(train_dist.py:74-91 semantics; no network in this environment), trains
the ConvNet with per-parameter gradient averaging
(train_dist.py:94-100) or the bucketed overlapped DDP, and prints the
per-rank mean epoch loss (train_dist.py:125-127) — losses track each
other across ranks and the replicas stay bit-identical.

Run:  python examples/train_dist.py [--world 2] [--epochs 2]
      python examples/train_dist.py --backend rccl   # one rank/GPU
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dist_tuto_pth_amd import dist, training
from dist_tuto_pth_amd.dist.launcher import launch
from dist_tuto_pth_amd.parallel import SyntheticMNIST

def run(rank, size):
    # config travels to spawned children through the environment
    device = "cpu"
    if dist.get_backend() == "rccl":
        import torch
        device = f"cuda:{torch.cuda.current_device()}"
    ds = SyntheticMNIST(n=int(os.environ.get("TRAIN_SAMPLES", "4096")))
    training.run(rank, size,
                 epochs=int(os.environ.get("TRAIN_EPOCHS", "2")),
                 device=device,
                 mode=os.environ.get("TRAIN_MODE", "average_gradients"),
                 dataset=ds, log=print)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--world", type=int, default=2)
    p.add_argument("--backend", default="gloo",
                   choices=["tcp", "gloo", "rccl"])
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--mode", default="average_gradients",
                   choices=["average_gradients", "ddp"])
    p.add_argument("--samples", type=int, default=4096)
    args = p.parse_args()
    os.environ["TRAIN_EPOCHS"] = str(args.epochs)
    os.environ["TRAIN_MODE"] = args.mode
    os.environ["TRAIN_SAMPLES"] = str(args.samples)
    launch(run, args.world, backend=args.backend, timeout=1800)


if __name__ == "__main__":
    main()
