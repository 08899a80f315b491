#!/usr/bin/env python3
"""IPC-transport all-reduce timing with N ranks sharing ONE device.

This is NOT an xGMI bandwidth number: both ranks' pushes traverse the
same GPU's HBM (a loopback).  What it measures end to end is the real
transport — hipIpc handle exchange through the C++ TCP store,
one-sided hipMemcpyAsync pushes into the peer mesh, store-ADD/blocking-
GET barriers, reduce_columns folds — i.e. every part of
`--algos p2p` except the link under the copies.  On a multi-GPU node
the identical code path crosses xGMI.

Run on a 1-GPU box:
  python benchmarks/bench_ipc_loopback.py --size-mb 256 --iters 10
"""

import argparse
import json
import multiprocessing as mp
import os
import socket
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def rank_fn(rank, world, port, numel, iters, warmup, q):
    try:
        import torch
        from dist_tuto_pth_amd.algorithms.ipc import (
            IpcTransport, fullmesh_all_reduce_ipc)
        from dist_tuto_pth_amd.utils.native import load_native
        torch.cuda.set_device(0)
        rx = load_native("_rcclx")
        store = rx.TcpStore("127.0.0.1", port, rank, world, rank == 0,
                            120_000)
        k = load_native("_kernels")
        g = torch.Generator().manual_seed(7 + rank)
        t = torch.randn(numel, generator=g).cuda()
        ref = sum(torch.randn(numel, generator=torch.Generator()
                              .manual_seed(7 + r)) for r in range(world))
        chunk_cap = ((numel + world - 1) // world + 64) * 4
        tp = IpcTransport(store, rank, world, chunk_cap, device=0,
                          tag="lb")
        base = t.clone()
        for _ in range(warmup):
            t.copy_(base)
            fullmesh_all_reduce_ipc(t, tp, k, rank, world)
        torch.cuda.synchronize()
        t.copy_(base)
        t0 = time.perf_counter()
        for _ in range(iters):
            fullmesh_all_reduce_ipc(t, tp, k, rank, world)
        torch.cuda.synchronize()
        el = (time.perf_counter() - t0) / iters
        # correctness on the first reduction only (values grow after)
        t.copy_(base)
        fullmesh_all_reduce_ipc(t, tp, k, rank, world)
        torch.cuda.synchronize()
        ok = bool(torch.allclose(t.cpu(), ref, rtol=1e-4, atol=1e-4))
        tp.close()
        q.put(("ok", rank, el, ok))
    except Exception as e:  # noqa: BLE001
        q.put(("err", rank, repr(e), False))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--size-mb", type=float, default=256.0)
    p.add_argument("--ranks", type=int, default=2)
    p.add_argument("--iters", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    args = p.parse_args()
    numel = int(args.size_mb * 1e6 / 4)
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=rank_fn,
                      args=(r, args.ranks, port, numel, args.iters,
                            args.warmup, q))
          for r in range(args.ranks)]
    for pr in ps:
        pr.start()
    results = [q.get(timeout=300) for _ in range(args.ranks)]
    for pr in ps:
        pr.join(30)
    errs = [r for r in results if r[0] != "ok"]
    if errs:
        print(json.dumps({"error": str(errs)}))
        sys.exit(1)
    el = max(r[2] for r in results)
    ok = all(r[3] for r in results)
    bytes_ = numel * 4
    world = args.ranks
    print(json.dumps({
        "metric": "ipc fullmesh all-reduce, LOOPBACK (ranks share 1 GPU)",
        "note": "transport correctness+overhead measurement, not xGMI BW",
        "n_ranks": world, "n_gpus": 1, "size_mb": args.size_mb,
        "iters": args.iters, "ms": el * 1e3, "correct": ok,
        "bus_GBps_loopback": 2 * (world - 1) / world * bytes_ / el / 1e9,
    }))


if __name__ == "__main__":
    main()
