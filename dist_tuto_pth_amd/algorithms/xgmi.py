"""Hand-tuned MI355X all-reduce (K14 of SURVEY.md §2.4 — the BASELINE
centerpiece).

The reference's ring all-reduce (allreduce.py:8-34) moves the FULL
tensor every step over one neighbor link.  On an 8-GPU MI355X node the
xGMI fabric is a FULL MESH: each GPU has 7 point-to-point links of
~153 GB/s (SURVEY.md §5 topology note), so a single ring is bound by
one link while six idle.  The bandwidth-optimal shape here is therefore
not a deeper ring but a *direct full-mesh* reduce-scatter + all-gather:

  stage 1: every rank sends chunk p to rank p and receives its own
           chunk from all peers — 7 sends + 7 recvs in ONE RCCL group,
           all 7 links busy both directions at once;
  stage 2: one fused column-reduction HIP kernel
           (``reduce_columns``, csrc/kernels.hip) folds the 7 received
           chunks into the owned chunk (bf16 accumulates in fp32 and
           rounds once);
  stage 3: every rank broadcasts its reduced chunk to all peers —
           again one grouped exchange.

Per-link traffic: 2*S/N bytes vs the ring's 2*S*(N-1)/N on its single
link — a factor (N-1) less per-link time at equal total bytes.

A chunked ring (``algo="ring"``) is kept for comparison and as the
direct rewrite of the reference algorithm (reduce-scatter + all-gather,
2(N-1) steps, the tuto.md:354 exercise) with RCCL p2p transport and the
``add_inplace`` HIP kernel as the local reduction.
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch

from .. import dist
from ..utils.native import load_native

_DTYPE = {torch.float32: 7, torch.bfloat16: 9}
_scratch: Dict[Tuple, torch.Tensor] = {}


def _buf(key, numel, dtype, device):
    t = _scratch.get(key)
    if t is None or t.numel() < numel or t.dtype != dtype:
        t = torch.empty(numel, dtype=dtype, device=device)
        _scratch[key] = t
    return t


def _comm_of(g):
    gr = dist._resolve(g)
    impl = gr._impl
    if not hasattr(impl, "comm"):
        raise RuntimeError("xgmi_all_reduce requires the rccl backend")
    return impl.comm, gr


def xgmi_all_reduce(tensor: torch.Tensor, g=None, algo: str = "fullmesh",
                    average: bool = False) -> torch.Tensor:
    """In-place sum (or average) all-reduce over xGMI.  ``tensor`` must
    be a contiguous CUDA tensor of fp32 or bf16."""
    assert tensor.is_cuda and tensor.is_contiguous()
    size = dist.get_world_size(g)
    rank = dist.get_rank(g)
    if size == 1:
        if average:
            pass
        return tensor
    if algo == "fullmesh":
        return _fullmesh(tensor, g, rank, size, average)
    if algo == "ring":
        return _ring(tensor, g, rank, size, average)
    raise ValueError(f"unknown algo {algo!r}")


def _pad_chunks(tensor, size):
    """Return (work_flat, chunk, padded) with chunk aligned to 16 B."""
    flat = tensor.view(-1)
    n = flat.numel()
    align = 16 // flat.element_size()
    chunk = ((n + size - 1) // size + align - 1) // align * align
    if chunk * size == n:
        return flat, chunk, False
    work = _buf(("pad", tensor.device.index, tensor.dtype), chunk * size,
                tensor.dtype, tensor.device)[:chunk * size]
    work[n:].zero_()
    work[:n].copy_(flat)
    return work, chunk, True


def _fullmesh(tensor, g, rank, size, average):
    comm, gr = _comm_of(g)
    k = load_native("_kernels")
    dt = _DTYPE[tensor.dtype]
    stream = torch.cuda.current_stream().cuda_stream
    work, chunk, padded = _pad_chunks(tensor, size)
    esz = work.element_size()
    base = work.data_ptr()
    scratch = _buf(("fm", tensor.device.index, tensor.dtype),
                   (size - 1) * chunk, tensor.dtype, tensor.device)
    sbase = scratch.data_ptr()

    # stage 1 — direct reduce-scatter exchange: all 7 links at once
    comm.group_start()
    for d in range(1, size):
        peer = (rank + d) % size
        comm.send(base + peer * chunk * esz, chunk, dt, peer, stream)
        comm.recv(sbase + (d - 1) * chunk * esz, chunk, dt, peer, stream)
    comm.group_end()

    # stage 2 — fold the received chunks into the owned chunk
    k.reduce_columns(base + rank * chunk * esz, sbase, size - 1, chunk,
                     chunk, (1.0 / size) if average else 1.0, dt, stream)

    # stage 3 — direct all-gather of reduced chunks
    comm.group_start()
    for d in range(1, size):
        peer = (rank + d) % size
        comm.send(base + rank * chunk * esz, chunk, dt, peer, stream)
        comm.recv(base + peer * chunk * esz, chunk, dt, peer, stream)
    comm.group_end()

    if padded:
        tensor.view(-1).copy_(work[:tensor.numel()])
    return tensor


def _ring(tensor, g, rank, size, average):
    """Chunked ring (the corrected reference algorithm, allreduce.py:8-34
    + the tuto.md:354 chunking exercise) on RCCL p2p + HIP add."""
    comm, gr = _comm_of(g)
    k = load_native("_kernels")
    dt = _DTYPE[tensor.dtype]
    stream = torch.cuda.current_stream().cuda_stream
    work, chunk, padded = _pad_chunks(tensor, size)
    esz = work.element_size()
    base = work.data_ptr()
    tmp = _buf(("ring", tensor.device.index, tensor.dtype), chunk,
               tensor.dtype, tensor.device)
    tbase = tmp.data_ptr()
    left = (rank - 1 + size) % size
    right = (rank + 1) % size

    # reduce-scatter around the ring
    for i in range(size - 1):
        s_idx = (rank - i) % size
        r_idx = (rank - i - 1) % size
        comm.group_start()
        comm.send(base + s_idx * chunk * esz, chunk, dt, right, stream)
        comm.recv(tbase, chunk, dt, left, stream)
        comm.group_end()
        k.add_inplace(base + r_idx * chunk * esz, tbase, chunk, dt, stream)

    owned = (rank + 1) % size
    if average:
        if tensor.dtype == torch.float32:
            k.scale_f32(base + owned * chunk * esz, 1.0 / size, chunk,
                        stream)
        else:
            k.reduce_columns(base + owned * chunk * esz, 0, 0, 0, chunk,
                             1.0 / size, dt, stream)

    # all-gather around the ring
    for i in range(size - 1):
        s_idx = (owned - i) % size
        r_idx = (owned - i - 1) % size
        comm.group_start()
        comm.send(base + s_idx * chunk * esz, chunk, dt, right, stream)
        comm.recv(base + r_idx * chunk * esz, chunk, dt, left, stream)
        comm.group_end()

    if padded:
        tensor.view(-1).copy_(work[:tensor.numel()])
    return tensor
