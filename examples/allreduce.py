#!/usr/bin/env python3
"""Hand-rolled ring all-reduce demo — the rebuilt `allreduce.py` /
`gloo.py` (reference allreduce.py:8-47, with the buffer-seeding and
accumulation bugs fixed per SURVEY.md §2.5.1).

Three algorithms over the same p2p surface:
  * ``ring``      — the reference's double-buffered full-tensor ring
  * ``chunked``   — reduce-scatter + all-gather (the tuto.md:354
                    exercise), bandwidth-optimal on a ring
  * ``fullmesh``  — the MI355X-native direct exchange using all 7 xGMI
                    links at once (GPU/rccl only; algorithms/xgmi.py)

Run:  python examples/allreduce.py [--world 4] [--algo chunked]
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.algorithms import chunked_ring_all_reduce, \
    ring_all_reduce
from dist_tuto_pth_amd.dist.launcher import launch

def run(rank, size):
    algo = os.environ.get("ALLREDUCE_ALGO", "chunked")
    numel = int(os.environ.get("ALLREDUCE_NUMEL", "1000"))
    iters = int(os.environ.get("ALLREDUCE_ITERS", "4"))
    device = "cpu"
    if dist.get_backend() == "rccl":
        device = f"cuda:{torch.cuda.current_device()}"
    g = torch.Generator().manual_seed(1234 + rank)
    t = torch.rand(numel, generator=g).to(device)

    # the reference demo re-reduces 4 times (gloo.py:42-46: clone,
    # all_reduce, set_); every iteration is checked against the
    # built-in collective on the same input
    for it in range(iters):
        ref = t.clone()
        dist.all_reduce(ref, op=dist.ReduceOp.SUM)

        if algo == "ring":
            out = torch.zeros_like(t)
            ring_all_reduce(t, out)
            t = out
        elif algo == "chunked":
            chunked_ring_all_reduce(t)
        elif algo == "fullmesh":
            from dist_tuto_pth_amd.algorithms.xgmi import xgmi_all_reduce
            xgmi_all_reduce(t, algo="fullmesh")
        else:
            raise SystemExit(f"unknown algo {algo}")

        ok = torch.allclose(t, ref, rtol=1e-5, atol=1e-4 * (size ** it))
        print(f"Rank {rank} iter {it}: {algo} all-reduce "
              f"{'matches' if ok else 'MISMATCHES'} the built-in "
              f"(sum[0]={t[0].item():.4f})")
        assert ok


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--world", type=int, default=4)
    p.add_argument("--backend", default="gloo",
                   choices=["tcp", "gloo", "rccl"])
    p.add_argument("--algo", default="chunked",
                   choices=["ring", "chunked", "fullmesh"])
    p.add_argument("--numel", type=int, default=1000)
    p.add_argument("--iters", type=int, default=4,
                   help="repeat count (the gloo.py demo runs 4)")
    args = p.parse_args()
    os.environ["ALLREDUCE_ALGO"] = args.algo
    os.environ["ALLREDUCE_NUMEL"] = str(args.numel)
    os.environ["ALLREDUCE_ITERS"] = str(args.iters)
    launch(run, args.world, backend=args.backend)


if __name__ == "__main__":
    main()
