#!/usr/bin/env python3
"""ResNet-50 DP benchmark: synthetic 3x224x224,
bucketed gradient all-reduce overlapped with backward
(parallel/ddp.py) over the native RCCL backend, one rank per MI355X.

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 benchmarks/bench_resnet.py \
      --steps 30 --warmup 10 --batch 64
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dist_tuto_pth_amd import dist  # noqa: E402
from dist_tuto_pth_amd.models.resnet import resnet50  # noqa: E402
from dist_tuto_pth_amd.parallel.ddp import DistributedDataParallel  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=64,
                   help="per-GPU batch (weak scaling)")
    p.add_argument("--bucket-mb", type=float, default=25.0)
    p.add_argument("--no-overlap", action="store_true",
                   help="reference-style average_gradients after "
                        "backward instead of bucketed overlap")
    p.add_argument("--channels-last", action="store_true",
                   help="NHWC memory format (MIOpen fast path)")
    p.add_argument("--no-find", action="store_true",
                   help="disable MIOpen exhaustive kernel search")
    p.add_argument("--force-comm", action="store_true",
                   help="world-1 tracing mode: init the backend and "
                        "launch the bucket all-reduces anyway so a "
                        "rocprof timeline shows the comm-stream overlap")
    p.add_argument("--dtype", default="fp32", choices=["fp32", "bf16"],
                   help="bf16 = autocast compute (supplementary number; "
                        "the headline config is fp32)")
    args = p.parse_args()

    # MIOpen exhaustive find: pick the fastest conv kernel per shape
    # during warmup (the warmup steps absorb the search cost)
    torch.backends.cudnn.benchmark = not args.no_find

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    dev_idx = local_rank % torch.cuda.device_count()
    if world > 1 or args.force_comm:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group("rccl", world_size=world, rank=rank,
                                device_id=dev_idx)
    torch.cuda.set_device(dev_idx)
    device = f"cuda:{dev_idx}"

    torch.manual_seed(1234)
    model = resnet50().to(device)
    if args.channels_last:
        model = model.to(memory_format=torch.channels_last)
    if world > 1:
        for prm in model.parameters():
            dist.broadcast(prm.data, src=0)
    ddp = None
    if (world > 1 or args.force_comm) and not args.no_overlap:
        ddp = DistributedDataParallel(model, bucket_cap_mb=args.bucket_mb,
                                      force_comm=args.force_comm)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)

    g = torch.Generator().manual_seed(1234 + rank)
    x = torch.randn(args.batch, 3, 224, 224, generator=g).to(device)
    if args.channels_last:
        x = x.to(memory_format=torch.channels_last)
    tgt = torch.randint(0, 1000, (args.batch,), generator=g).to(device)
    crit = torch.nn.CrossEntropyLoss()

    import contextlib
    amp = (torch.autocast("cuda", dtype=torch.bfloat16)
           if args.dtype == "bf16" else contextlib.nullcontext())

    def step():
        opt.zero_grad(set_to_none=True)
        if ddp is not None:
            with amp:
                loss = crit(ddp(x), tgt)
            loss.backward()
            ddp.finish_gradients()
        else:
            with amp:
                loss = crit(model(x), tgt)
            loss.backward()
            if world > 1:
                from dist_tuto_pth_amd.parallel import average_gradients
                average_gradients(model)
        opt.step()
        return loss

    def barrier_sync():
        if world > 1:
            dist.barrier()
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync()
    el = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([el], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        el = float(t.item())

    if rank == 0:
        print(json.dumps({
            "metric": "ResNet-50 samples/sec",
            "value": args.batch * world * args.steps / el,
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": el / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {"model": "ResNet-50", "global_batch":
                       args.batch * world, "input": "3x224x224",
                       "parallelism": f"dp{world}",
                       "channels_last": args.channels_last,
                       "miopen_find": not args.no_find,
                       "grad_sync": "average_gradients"
                       if args.no_overlap else
                       f"ddp_overlap_{args.bucket_mb}MB"},
        }))
    if world > 1 or args.force_comm:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
