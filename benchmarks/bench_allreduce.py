#!/usr/bin/env python3
"""All-reduce bus-bandwidth benchmark.

  config 2: 256 MB fp32 across N MI355X (the allreduce.py path)
  config 5: bf16 1 GB (Llama-3-8B-layer-sized) at 1/2/4/8 GPUs

Compares the RCCL built-in all-reduce against this package's hand-rolled
algorithms (fullmesh / ring — algorithms/xgmi.py).  Bus bandwidth uses
the standard nccl-tests convention: busBW = 2*(N-1)/N * bytes / time.

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 benchmarks/bench_allreduce.py \
      --size-mb 256 --dtype fp32 --algos rccl,fullmesh,ring
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dist_tuto_pth_amd import dist  # noqa: E402
from dist_tuto_pth_amd.algorithms.xgmi import xgmi_all_reduce  # noqa: E402
from dist_tuto_pth_amd.algorithms.ipc import (  # noqa: E402
    IpcTransport, fullmesh_all_reduce_ipc)
from dist_tuto_pth_amd.utils.native import load_native  # noqa: E402

_TP = [None]


def _ipc_transport(world, rank, dev_idx, numel, esz):
    if _TP[0] is None:
        store = dist.group.WORLD._impl._store
        chunk_cap = ((numel + world - 1) // world + 64) * esz
        _TP[0] = IpcTransport(store, rank, world, chunk_cap,
                              device=dev_idx, tag="benchp2p")
    return _TP[0]


def run_p2p(t, iters, warmup, world, rank, dev_idx):
    """--algos p2p: the self-owned transport — IPC mesh + one-sided
    hipMemcpyAsync pushes over xGMI, no RCCL on the wire."""
    k = load_native("_kernels")
    tp = _ipc_transport(world, rank, dev_idx, t.numel(), t.element_size())

    def op():
        fullmesh_all_reduce_ipc(t, tp, k, rank, world)

    for _ in range(warmup):
        op()
    dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        op()
    torch.cuda.synchronize()
    dist.barrier()
    el = (time.perf_counter() - t0) / iters
    e = torch.tensor([el], device=t.device)
    dist.all_reduce(e, op=dist.ReduceOp.MAX)
    el = float(e.item())
    bytes_ = t.numel() * t.element_size()
    bus_bw = 2 * (world - 1) / world * bytes_ / el / 1e9
    alg_bw = bytes_ / el / 1e9
    return el, bus_bw, alg_bw


def run_algo(algo, t, iters, warmup, world, depth=4):
    def op():
        if algo == "rccl":
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
        else:
            xgmi_all_reduce(t, algo=algo, depth=depth)

    def sync():
        if t.is_cuda:
            torch.cuda.synchronize()

    for _ in range(warmup):
        op()
    if world > 1:
        dist.barrier()
    sync()
    t0 = time.perf_counter()
    for _ in range(iters):
        op()
    sync()
    if world > 1:
        dist.barrier()
    el = (time.perf_counter() - t0) / iters
    # max over ranks
    if world > 1:
        e = torch.tensor([el], device=t.device)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        el = float(e.item())
    bytes_ = t.numel() * t.element_size()
    bus_bw = 2 * (world - 1) / world * bytes_ / el / 1e9 if world > 1 else 0.0
    alg_bw = bytes_ / el / 1e9
    return el, bus_bw, alg_bw


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--size-mb", type=float, default=256.0)
    p.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32")
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--algos", default="rccl,fullmesh,ring")
    p.add_argument("--pipeline-depth", type=int, default=4,
                   help="fullmesh sub-chunks per owned chunk: exchange "
                        "of sub-chunk d+1 overlaps the reduce of d "
                        "(1 = the unpipelined three-stage form)")
    p.add_argument("--no-check", action="store_true",
                   help="skip verifying the hand-rolled algorithms "
                        "against RCCL before timing")
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", 1))
    if world == 1:
        print(json.dumps({"metric": "all-reduce bus BW (GB/s)",
                          "n_gpus": 1, "note": "all-reduce needs >= 2 "
                          "ranks; launch via torch.distributed.run"}))
        return
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    # DTP_BENCH_CPU=1: identical launch/arg/rendezvous path on
    # CPU/gloo so the torchrun contract is testable without a GPU
    # (tests/test_bench_contract.py); only the built-in all-reduce
    # runs there.
    cpu_mode = os.environ.get("DTP_BENCH_CPU") == "1"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if cpu_mode:
        dist.init_process_group("gloo", world_size=world, rank=rank)
        dev_idx = 0
        args.algos = "rccl"
        args.no_check = True
    else:
        dev_idx = local_rank % torch.cuda.device_count()
        dist.init_process_group("rccl", world_size=world, rank=rank,
                                device_id=dev_idx)
        torch.cuda.set_device(dev_idx)
    dtype = torch.float32 if args.dtype == "fp32" else torch.bfloat16
    numel = int(args.size_mb * 1e6) // (4 if dtype == torch.float32 else 2)
    g = torch.Generator().manual_seed(42 + rank)
    base = torch.randn(numel, generator=g).to(dtype)
    if not cpu_mode:
        base = base.cuda()

    if not args.no_check and world > 1:
        ref = base.clone()
        dist.all_reduce(ref, op=dist.ReduceOp.SUM)
        algos = args.algos.split(",")
        for algo in [a for a in ("fullmesh", "ring", "p2p") if a in algos]:
            t = base.clone()
            if algo == "p2p":
                fullmesh_all_reduce_ipc(
                    t, _ipc_transport(world, rank, dev_idx, t.numel(),
                                      t.element_size()),
                    load_native("_kernels"), rank, world)
            else:
                xgmi_all_reduce(t, algo=algo, depth=args.pipeline_depth)
            torch.cuda.synchronize()
            ok = torch.allclose(t.float(), ref.float(),
                                atol=1e-2 if dtype == torch.bfloat16
                                else 1e-4, rtol=1e-2)
            if rank == 0:
                print(f"# check {algo}: {'OK' if ok else 'MISMATCH'} "
                      f"max|d|={float((t.float()-ref.float()).abs().max())}",
                      flush=True)

    results = {}
    for algo in args.algos.split(","):
        if world == 1 and algo != "rccl":
            continue  # p2p algorithms need peers
        t = base.clone()
        if algo == "p2p":
            el, bus, alg = run_p2p(t, args.iters, args.warmup, world,
                                   rank, dev_idx)
        else:
            el, bus, alg = run_algo(algo, t, args.iters, args.warmup, world,
                                    depth=args.pipeline_depth)
        results[algo] = {"ms": el * 1e3, "bus_GBps": bus, "alg_GBps": alg}
        if algo == "fullmesh":
            results[algo]["pipeline_depth"] = args.pipeline_depth

    if rank == 0:
        print(json.dumps({
            "metric": "all-reduce bus BW (GB/s)",
            "n_gpus": world,
            "size_mb": args.size_mb,
            "dtype": args.dtype,
            "iters": args.iters,
            "warmup": args.warmup,
            "results": results,
        }))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
