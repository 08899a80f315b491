#!/usr/bin/env python3
"""Build every native extension in-tree for gfx950 (MI355X).

Outputs land in ``dist_tuto_pth_amd/_native/*.so`` so they travel with
the repo snapshot to GPU machines.  hipcc cross-compiles without a GPU.
"""

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dist_tuto_pth_amd", "csrc")
OUT = os.path.join(ROOT, "dist_tuto_pth_amd", "_native")

ARCH = os.environ.get("DTP_AMD_ARCH", "gfx950")


def _includes():
    import pybind11
    import sysconfig
    return [
        pybind11.get_include(),
        sysconfig.get_paths()["include"],
        "/opt/rocm/include",
    ]


def _run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def _needs_build(out, srcs):
    if not os.path.exists(out):
        return True
    ot = os.path.getmtime(out)
    return any(os.path.getmtime(s) > ot for s in srcs)


def build(force: bool = False):
    os.makedirs(OUT, exist_ok=True)
    inc = sum([["-I", p] for p in _includes()], [])
    common = ["hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17",
              "-fPIC", "-shared", "-fvisibility=hidden"] + inc

    targets = [
        # (output, sources, extra flags)
        ("_rcclx.so", [os.path.join(CSRC, "rcclx.cpp")],
         ["-L/opt/rocm/lib", "-lrccl"]),
        ("_kernels.so", [os.path.join(CSRC, "kernels.hip")], []),
    ]
    for out_name, srcs, extra in targets:
        out = os.path.join(OUT, out_name)
        if force or _needs_build(out, srcs):
            _run(common + srcs + extra + ["-o", out])
        else:
            print(f"{out_name}: up to date")


if __name__ == "__main__":
    build(force="--force" in sys.argv)
