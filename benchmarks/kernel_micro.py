#!/usr/bin/env python3
"""Per-kernel microbenchmark of the fused ConvNet training step
(hipEvent timing, 1 GPU).  Pinpoints where the step's microseconds go:
fwd, data-bwd (at each sibling-split), the four weight-gradient tile
segments separately, combine, sgd, prologue.

Run on a GPU box:  python benchmarks/kernel_micro.py [--batch 128]
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dist_tuto_pth_amd.models import Net  # noqa: E402
from dist_tuto_pth_amd.optim import FusedSGD  # noqa: E402
from dist_tuto_pth_amd.ops import _seed_ptr, _stream  # noqa: E402
from dist_tuto_pth_amd.ops.fused import _ws, attach_flat_grads  # noqa: E402
from dist_tuto_pth_amd.utils.native import load_native  # noqa: E402

# tile segment table — keep in sync with csrc/kernels.hip
T_CONV2, T_FC1, T_CONV1, T_FC2 = 20, 63, 24, 2


def time_fn(fn, reps=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1000.0  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=128)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    B = args.batch

    k = load_native("_kernels")
    net = Net().to(dev)
    net.train()
    ws = _ws(B, dev)
    ws.setdefault("one", torch.ones((), device=dev))
    flat = attach_flat_grads(net)
    opt = FusedSGD(net.parameters(), lr=0.01, momentum=0.5,
                   zero_grad_in_step=True)
    x = torch.randn(B, 1, 28, 28, device=dev)
    tgt = torch.randint(0, 10, (B,), device=dev)
    params = [net.conv1.weight, net.conv1.bias, net.conv2.weight,
              net.conv2.bias, net.fc1.weight, net.fc1.bias,
              net.fc2.weight, net.fc2.bias]
    pp = [p.data_ptr() for p in params]
    s = _stream()

    def fwd():
        k.net_fused_fwd(x.data_ptr(), *pp, tgt.data_ptr(),
                        ws["p1"].data_ptr(), ws["idx1"].data_ptr(),
                        ws["m2"].data_ptr(), ws["p2"].data_ptr(),
                        ws["idx2"].data_ptr(), ws["h1"].data_ptr(),
                        ws["m3"].data_ptr(), ws["d3"].data_ptr(),
                        ws["logp"].data_ptr(), ws["loss"].data_ptr(),
                        0, _seed_ptr(dev), B, True, s)

    def bwd():
        k.net_fused_bwd(x.data_ptr(), pp[2], pp[4], pp[6], tgt.data_ptr(),
                        ws["one"].data_ptr(), ws["p1"].data_ptr(),
                        ws["idx1"].data_ptr(), ws["m2"].data_ptr(),
                        ws["p2"].data_ptr(), ws["idx2"].data_ptr(),
                        ws["h1"].data_ptr(), ws["m3"].data_ptr(),
                        ws["d3"].data_ptr(), ws["logp"].data_ptr(),
                        ws["glog"].data_ptr(), ws["gh1"].data_ptr(),
                        ws["ga2"].data_ptr(), ws["ga1"].data_ptr(),
                        ws["part"].data_ptr(),
                        *[p.grad.data_ptr() for p in params], B, True,
                        0, 0, 0, s)

    fwd()
    bwd()
    torch.cuda.synchronize()

    # mirror csrc gw_nch()/gw_c1_subs() so raw segment launches match
    # what the fused step actually runs
    nch = int(os.environ.get("DTP_GW_NCH", 0)) or         (16 if B <= 192 else (24 if B <= 768 else 32))
    nch = min(nch, 32, B)
    bchunk = (B + nch - 1) // nch
    nch1 = min(int(os.environ.get("DTP_GW_NCH1", 0)) or nch, 32, B)
    bchunk1 = (B + nch1 - 1) // nch1
    nch2 = min(int(os.environ.get("DTP_GW_NCH2", 0)) or
               (24 if B <= 192 else nch), 32, B)
    bchunk2 = (B + nch2 - 1) // nch2
    c1_ext = (8 if bchunk <= 16 else 24) - 1

    def seg(base, ntiles):
        def f():
            k.net_gw_partial_raw(
                x.data_ptr(), ws["p1"].data_ptr(), ws["p2"].data_ptr(),
                ws["d3"].data_ptr(), ws["ga1"].data_ptr(),
                ws["ga2"].data_ptr(), ws["gh1"].data_ptr(),
                ws["glog"].data_ptr(), ws["part"].data_ptr(), B, bchunk,
                bchunk1, bchunk2, base, ntiles, nch, nch1, nch2, s)
        return f

    results = {}
    results["fwd"] = time_fn(fwd)
    # full bwd+gw+combine bundle is what bwd() launches; time the pieces
    for sp in (1, 2, 4, 8):
        os.environ["DTP_BWD_SPLIT"] = str(sp)
        results[f"bwd+gw (split={sp})"] = time_fn(bwd)
    os.environ.pop("DTP_BWD_SPLIT", None)
    results["gw conv2 tiles"] = time_fn(seg(0, T_CONV2))
    results["gw conv1 tiles"] = time_fn(seg(T_CONV2, T_CONV1))
    results["gw fc1 tiles"] = time_fn(seg(T_CONV2 + T_CONV1, T_FC1))
    results["gw fc2 tiles"] = time_fn(seg(T_CONV2 + T_CONV1 + T_FC1, T_FC2))
    results["gw all tiles"] = time_fn(seg(0, T_CONV2 + T_FC1 + T_CONV1 + T_FC2))
    results["gw combine"] = time_fn(lambda: k.net_gw_combine_raw(
        ws["part"].data_ptr(), [p.grad.data_ptr() for p in params], nch,
        nch1, nch2, c1_ext, s))
    ws.setdefault("loss_part", torch.empty(4096, device=dev))
    bufs = [b.data_ptr() for b in opt._bufs]
    results["gw combine+sgd+loss"] = time_fn(
        lambda: k.net_gw_combine_sgd_raw(
            ws["part"].data_ptr(), [p.grad.data_ptr() for p in params],
            pp, bufs, nch, nch1, nch2, c1_ext, 0.01, 0.5,
            ws["loss_part"].data_ptr(), ws["loss"].data_ptr(), 128, s))
    results["sgd"] = time_fn(opt.step)

    for name, us in results.items():
        print(f"{name:24s} {us:8.2f} us")


if __name__ == "__main__":
    main()
