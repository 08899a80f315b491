#!/usr/bin/env python3
"""Point-to-point demos (tuto.md:77-122): blocking send/recv of a
single fp32 scalar between two ranks, then the non-blocking
isend/irecv + wait() form.  This is BASELINE.md config 1 (CPU plumbing,
no GPU required).

Run:  python examples/send_recv.py [--backend tcp|gloo|rccl]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.dist.launcher import launch


def run_blocking(rank, size):
    device = "cpu" if dist.get_backend() in ("gloo", "tcp") \
        else f"cuda:{torch.cuda.current_device()}"
    tensor = torch.zeros(1, device=device)
    if rank == 0:
        tensor += 1
        dist.send(tensor, dst=1)          # tuto.md:87
    else:
        dist.recv(tensor, src=0)          # tuto.md:90
    print(f"Rank {rank} has data {tensor[0].item()}")


def run_nonblocking(rank, size):
    tensor = torch.zeros(1)
    if rank == 0:
        tensor += 1
        req = dist.isend(tensor, dst=1)   # tuto.md:108
        print("Rank 0 started sending")
    else:
        req = dist.irecv(tensor, src=0)   # tuto.md:112
        print("Rank 1 started receiving")
    req.wait()
    print(f"Rank {rank} has data {tensor[0].item()}")


def run_latency(rank, size, iters=1000):
    """Round-trip latency of the 1-float ping-pong (config 1 metric)."""
    t = torch.zeros(1)
    # warmup
    for _ in range(10):
        if rank == 0:
            dist.send(t, 1)
            dist.recv(t, 1)
        else:
            dist.recv(t, 0)
            dist.send(t, 0)
    t0 = time.perf_counter()
    for _ in range(iters):
        if rank == 0:
            dist.send(t, 1)
            dist.recv(t, 1)
        else:
            dist.recv(t, 0)
            dist.send(t, 0)
    el = time.perf_counter() - t0
    if rank == 0:
        print(f"1-float round trip: {el / iters * 1e6:.2f} us "
              f"({iters} iters)")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--backend", default="tcp",
                   choices=["tcp", "gloo", "rccl"])
    args = p.parse_args()
    launch(run_blocking, 2, backend=args.backend)
    if args.backend != "rccl":
        launch(run_nonblocking, 2, backend=args.backend)
        launch(run_latency, 2, backend=args.backend)


if __name__ == "__main__":
    main()
