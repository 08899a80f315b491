#!/usr/bin/env python3
"""Gather-to-root demo — the rebuilt `ptp.py` (reference ptp.py:21-28).

Every rank contributes a ones(1) tensor; rank 0 gathers and prints the
sum, which must equal the world size (the reference's self-check,
SURVEY.md §4.1).  Modern API: ``dist.gather(tensor, gather_list, dst)``
instead of the 0.x positional-group form (ptp.py:26), plus the legacy
root-split pair ``gather_recv``/``gather_send`` (ptp.py:17-19) shown in
``run_legacy_pair``.

Run:  python examples/ptp.py [--world 2] [--backend tcp|gloo|rccl]
"""

import argparse
import sys
import os

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.dist.launcher import launch


def run(rank, size):
    device = "cpu"
    if dist.get_backend() == "rccl":
        device = f"cuda:{torch.cuda.current_device()}"
    tensor = torch.ones(1, device=device)
    tensor_list = [torch.zeros(1, device=device) for _ in range(size)] \
        if rank == 0 else None
    dist.gather(tensor, gather_list=tensor_list, dst=0)
    if rank == 0:
        total = sum(t.item() for t in tensor_list)
        print(f"Rank {rank} gathered sum: {total} (expected {float(size)})")
        assert total == float(size)


def run_legacy_pair(rank, size):
    tensor = torch.ones(1)
    if rank == 0:
        tl = [torch.zeros(1) for _ in range(size)]
        dist.gather_recv(tl, tensor)
        print(f"Rank 0 (legacy pair) sum: {sum(t.item() for t in tl)}")
    else:
        dist.gather_send(tensor, root=0)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--world", type=int, default=2)
    p.add_argument("--backend", default="tcp",
                   choices=["tcp", "gloo", "rccl"])
    args = p.parse_args()
    launch(run, args.world, backend=args.backend)
    if args.backend != "rccl":
        launch(run_legacy_pair, args.world, backend=args.backend)


if __name__ == "__main__":
    main()
