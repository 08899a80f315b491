#!/usr/bin/env python3
"""Flagship benchmark: ConvNet distributed synchronous-SGD training
(BASELINE.json config 3 — "ConvNet on synthetic 28x28 MNIST-shaped data,
DP with average_gradients(), NxMI355X"), metric = whole-job samples/sec.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

One rank per GPU over the native RCCL backend.  Weak scaling: per-GPU
batch fixed at --batch (default 128, the reference's global batch,
train_dist.py:85), so global batch = 128*N.  Synthetic data, random-init
weights (no network in this environment), fp32 (the reference's dtype).
Timing: W untimed warmup steps, barrier+sync, exactly K timed steps,
barrier+sync, MAX elapsed over ranks; rank 0 prints one JSON line.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from dist_tuto_pth_amd import dist, ops  # noqa: E402
from dist_tuto_pth_amd.models import Net  # noqa: E402
from dist_tuto_pth_amd.optim import FusedSGD  # noqa: E402
from dist_tuto_pth_amd.parallel import average_gradients  # noqa: E402
from dist_tuto_pth_amd.parallel.ddp import DistributedDataParallel  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=128,
                   help="per-GPU batch (weak scaling)")
    p.add_argument("--mode", choices=["average_gradients", "ddp"],
                   default="average_gradients")
    p.add_argument("--graph", action="store_true",
                   help="capture the training step in a hipGraph")
    p.add_argument("--no-fused", action="store_true",
                   help="use the modular per-op kernel pipeline instead "
                        "of the fused whole-Net kernels")
    p.add_argument("--fwdbwd", dest="fwdbwd", action="store_true",
                   default=None,
                   help="combined fwd+bwd kernel (3-dispatch step); "
                        "default: auto — on for 192 <= batch <= 512, "
                        "where it measures +1..+4% (profiles/sweeps/"
                        "fwdbwd_band.log), off elsewhere (loses 8% "
                        "at B=128, 4% at B=768)")
    p.add_argument("--no-fwdbwd", dest="fwdbwd", action="store_false")
    p.add_argument("--megakernel", action="store_true",
                   help="run the whole step as ONE cooperative kernel "
                        "launch (measured slower than the 6-dispatch "
                        "fused path at B=128: the weight-gradient phase "
                        "loses workgroup parallelism — profiles/)")
    return p.parse_args()


def main():
    args = parse_args()
    if args.fwdbwd is None:
        args.fwdbwd = 192 <= args.batch <= 512
    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))

    # DTP_BENCH_CPU=1: run the IDENTICAL launch path (torchrun env
    # rendezvous, broadcast, timed loop, JSON line) on CPU/gloo so the
    # driver contract is testable without a GPU.  Never auto-selected.
    cpu_mode = os.environ.get("DTP_BENCH_CPU") == "1"
    if cpu_mode:
        if world > 1:
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group("gloo", world_size=world, rank=rank)
        device = "cpu"
        args.no_fused = True
    else:
        n_dev = torch.cuda.device_count()
        if world > n_dev:
            raise RuntimeError(
                f"world_size {world} exceeds visible GPUs ({n_dev}); "
                f"one rank maps to one MI355X (RCCL does not support "
                f"multiple ranks per device)")
        dev_idx = local_rank % n_dev
        if world > 1:
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group("rccl", world_size=world, rank=rank,
                                    device_id=dev_idx)
        torch.cuda.set_device(dev_idx)
        device = f"cuda:{dev_idx}"

    torch.manual_seed(1234)
    model = Net().to(device)
    if world > 1:
        for p in model.parameters():
            dist.broadcast(p.data, src=0)
    opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.5,
                   zero_grad_in_step=True)
    ddp = DistributedDataParallel(model) if (args.mode == "ddp" and
                                             world > 1) else None

    g = torch.Generator(device="cpu").manual_seed(1234 + rank)
    x = torch.randn(args.batch, 1, 28, 28, generator=g).to(device)
    tgt = torch.randint(0, 10, (args.batch,), generator=g).to(device)

    use_fused = not args.no_fused and args.mode == "average_gradients"
    flat_grads = None
    use_mega = False
    if use_fused:
        from dist_tuto_pth_amd.ops.fused import (attach_flat_grads,
                                                 net_fused_step,
                                                 net_fused_step_fb,
                                                 net_fused_step_opt,
                                                 net_fused_train_step,
                                                 net_step_available)
        flat_grads = attach_flat_grads(model)
        # single-launch cooperative step kernel (whole step = 1 dispatch);
        # opt-in: measured slower than the multi-kernel fused path
        use_mega = (args.megakernel and not args.graph and
                    args.batch <= 512 and net_step_available())

    def step():
        if use_fused:
            if use_mega:
                if world == 1:
                    return net_fused_train_step(model, x, tgt, opt)
                loss = net_fused_train_step(model, x, tgt, do_sgd=False)
                dist.all_reduce(flat_grads, op=dist.ReduceOp.AVG)
                opt.step()
                return loss
            if args.fwdbwd:
                if world == 1:
                    return net_fused_step_fb(model, x, tgt, opt)
                loss = net_fused_step_fb(model, x, tgt)
                dist.all_reduce(flat_grads, op=dist.ReduceOp.AVG)
                opt.step()
                return loss
            if world == 1:
                # SGD update fused into the combine kernel (one
                # dispatch fewer; no all-reduce to wait for)
                return net_fused_step_opt(model, x, tgt, opt)
            loss = net_fused_step(model, x, tgt)
            # one flat all-reduce with built-in averaging: the
            # semantics of average_gradients (train_dist.py:94-100)
            # in a single xGMI message
            dist.all_reduce(flat_grads, op=dist.ReduceOp.AVG)
            opt.step()
            return loss
        if ddp is not None:
            loss = ops.nll_loss(ddp(x), tgt)
            loss.backward()
            ddp.finish_gradients()
        else:
            loss = ops.log_softmax_nll(model.forward_logits(x), tgt)
            loss.backward()
            if world > 1:
                average_gradients(model)
        opt.step()
        return loss

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if not cpu_mode:
            torch.cuda.synchronize()

    graph = None
    if args.graph:
        # warm once to settle allocator, then capture the whole step
        for _ in range(3):
            step()
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            step()

        def step():  # noqa: F811
            graph.replay()

    for _ in range(args.warmup):
        step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if world > 1:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    samples_per_sec = args.batch * world * args.steps / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": "ConvNet samples/sec",
            "value": samples_per_sec,
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "Net-21840",
                "global_batch": args.batch * world,
                "input": "1x28x28",
                "parallelism": f"dp{world}",
                "grad_sync": args.mode,
                "fused": use_fused,
                "fwdbwd": bool(args.fwdbwd and use_fused),
                "megakernel": use_mega,
                "graph": bool(args.graph),
            },
        }))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
