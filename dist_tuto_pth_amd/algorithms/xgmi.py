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

**Pipelining** (``depth > 1``): the three serial stages above leave the
reduce kernel's time as pure added latency.  With ``depth=D`` each
owned chunk is split into D sub-chunks and the stages overlap in the
double-buffered shape the reference's own ring demonstrates
(allreduce.py:21-32 buffer swap):

  comm stream   : X0 X1 X2 ... G0 G1 G2 ...     (X = exchange, G = gather)
  compute stream:    R0 R1 R2 ...               (R = reduce_columns)

The exchange of sub-chunk d+1 runs while sub-chunk d is being reduced,
and the gather of sub-chunk d starts the moment its reduction
completes — only the first exchange and the last gather are exposed.
Events order the streams; all RCCL ops stay on ONE comm stream in an
identical global order on every rank, so the grouped matching cannot
deadlock.

A chunked ring (``algo="ring"``) is kept for comparison and as the
direct rewrite of the reference algorithm (reduce-scatter + all-gather,
2(N-1) steps, the tuto.md:354 exercise) with RCCL p2p transport and the
``add_inplace`` HIP kernel as the local reduction.

The core algorithms are pure functions of an injected ``comm`` (RCCL
group p2p surface) and ``kern`` (reduction kernels on raw pointers), so
tests/test_xgmi_logic.py drives the EXACT production index math at
worlds 2-8 on CPU through a fake comm + ctypes kernels — the only part
the fakes replace is the wire and the device.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

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
                    average: bool = False, depth: int = 4) -> torch.Tensor:
    """In-place sum (or average) all-reduce over xGMI.  ``tensor`` must
    be a contiguous CUDA tensor of fp32 or bf16.  ``depth`` is the
    pipeline depth of the fullmesh algorithm (sub-chunks per owned
    chunk; 1 = the unpipelined three-stage form)."""
    assert tensor.is_cuda and tensor.is_contiguous()
    size = dist.get_world_size(g)
    rank = dist.get_rank(g)
    if size == 1:
        return tensor
    comm, _ = _comm_of(g)
    kern = load_native("_kernels")
    if algo == "fullmesh":
        ctx = _HipStreams(tensor.device)
        fullmesh_all_reduce(tensor, comm, kern, rank, size, average,
                            depth=depth, streams=ctx, buf=_buf)
        return tensor
    if algo == "ring":
        ring_all_reduce(tensor, comm, kern, rank, size, average,
                        stream=torch.cuda.current_stream().cuda_stream,
                        buf=_buf)
        return tensor
    raise ValueError(f"unknown algo {algo!r}")


class _HipStreams:
    """Two-stream context for the pipelined fullmesh: RCCL traffic on a
    dedicated comm stream, ``reduce_columns`` on the caller's compute
    stream, ordered by events.  ``None`` in the core functions means
    sequential execution (the fake-comm CPU tests)."""

    _cache: Dict[int, torch.cuda.Stream] = {}

    def __init__(self, device):
        idx = device.index
        if idx not in self._cache:
            self._cache[idx] = torch.cuda.Stream(device=device)
        self.comm_stream = self._cache[idx]
        self.compute_stream = torch.cuda.current_stream(device)

    def comm_handle(self) -> int:
        return self.comm_stream.cuda_stream

    def compute_handle(self) -> int:
        return self.compute_stream.cuda_stream

    def comm_wait_compute(self):
        ev = torch.cuda.Event()
        ev.record(self.compute_stream)
        self.comm_stream.wait_event(ev)

    def compute_wait_comm(self):
        ev = torch.cuda.Event()
        ev.record(self.comm_stream)
        self.compute_stream.wait_event(ev)


def _pad_chunks(tensor, size, buf):
    """Return (work_flat, chunk, padded) with chunk aligned to 16 B."""
    flat = tensor.view(-1)
    n = flat.numel()
    align = 16 // flat.element_size()
    chunk = ((n + size - 1) // size + align - 1) // align * align
    if chunk * size == n:
        return flat, chunk, False
    work = buf(("pad", tensor.device.index, tensor.dtype), chunk * size,
               tensor.dtype, tensor.device)[:chunk * size]
    work[n:].zero_()
    work[:n].copy_(flat)
    return work, chunk, True


def _sub_splits(chunk: int, depth: int, align: int):
    """Split ``chunk`` elements into <=depth aligned (offset, count)
    slices.  All but the last slice have equal aligned size."""
    if depth <= 1 or chunk <= align:
        return [(0, chunk)]
    sub = ((chunk + depth - 1) // depth + align - 1) // align * align
    out = []
    off = 0
    while off < chunk:
        out.append((off, min(sub, chunk - off)))
        off += sub
    return out


def fullmesh_all_reduce(tensor, comm, kern, rank, size, average=False,
                        depth: int = 4, streams: Optional[_HipStreams] = None,
                        buf=_buf):
    """Direct full-mesh reduce-scatter + all-gather, pipelined at
    ``depth`` sub-chunks.  ``comm``/``kern`` are injected (production:
    the RCCL comm + the HIP kernel module; tests: fakes over CPU
    memory).  ``streams=None`` runs the same index math sequentially on
    stream 0 (CPU fake path)."""
    dt = _DTYPE[tensor.dtype]
    work, chunk, padded = _pad_chunks(tensor, size, buf)
    esz = work.element_size()
    base = work.data_ptr()
    scratch = buf(("fm", tensor.device.index, tensor.dtype),
                  (size - 1) * chunk, tensor.dtype, tensor.device)
    sbase = scratch.data_ptr()
    align = 16 // esz
    subs = _sub_splits(chunk, depth, align)
    scale = (1.0 / size) if average else 1.0

    comm_s = streams.comm_handle() if streams else 0
    compute_s = streams.compute_handle() if streams else 0

    if streams:
        # RCCL traffic must not start before prior compute on the
        # caller's stream produced `work`
        streams.comm_wait_compute()

    # ---- stage 1: post ALL exchange groups back-to-back on the comm
    # stream (X_0 .. X_{D-1}); an event after each marks when that
    # sub-chunk's peer rows have landed in scratch.  Only X_0 is
    # exposed — X_{d>0} overlaps the reductions below.
    ev_x = []
    for off, cnt in subs:
        comm.group_start()
        for d in range(1, size):
            peer = (rank + d) % size
            comm.send(base + (peer * chunk + off) * esz, cnt, dt, peer,
                      comm_s)
            comm.recv(sbase + ((d - 1) * chunk + off) * esz, cnt, dt,
                      peer, comm_s)
        comm.group_end()
        if streams:
            e = torch.cuda.Event()
            e.record(streams.comm_stream)
            ev_x.append(e)

    # ---- stage 2+3 pipelined: reduce sub-chunk d on the compute
    # stream the moment X_d lands, and post its gather group G_d on the
    # comm stream the moment R_d lands — so G_d's wire time overlaps
    # R_{d+1} (the reference's double-buffer idea, allreduce.py:21-32,
    # at sub-chunk granularity).  Comm-stream order is identical on
    # every rank (X_0..X_{D-1}, G_0..G_{D-1}), so the grouped matching
    # cannot deadlock.
    for i, (off, cnt) in enumerate(subs):
        if streams:
            streams.compute_stream.wait_event(ev_x[i])
        # fold the size-1 received rows (leading dim = chunk) into the
        # owned chunk's sub-slice; scale applies the average exactly
        # once, at the owner
        kern.reduce_columns(base + (rank * chunk + off) * esz,
                            sbase + off * esz, size - 1, chunk, cnt,
                            scale, dt, compute_s)
        if streams:
            er = torch.cuda.Event()
            er.record(streams.compute_stream)
            streams.comm_stream.wait_event(er)
        comm.group_start()
        for d in range(1, size):
            peer = (rank + d) % size
            comm.send(base + (rank * chunk + off) * esz, cnt, dt, peer,
                      comm_s)
            comm.recv(base + (peer * chunk + off) * esz, cnt, dt, peer,
                      comm_s)
        comm.group_end()

    if streams:
        streams.compute_wait_comm()

    if padded:
        tensor.view(-1).copy_(work[:tensor.numel()])
    return tensor


def ring_all_reduce(tensor, comm, kern, rank, size, average=False,
                    stream: int = 0, buf=_buf):
    """Chunked ring (the corrected reference algorithm, allreduce.py:8-34
    + the tuto.md:354 chunking exercise) on RCCL p2p + HIP add.  Every
    send/recv pair is grouped — the deadlock-free form of the ring
    step."""
    dt = _DTYPE[tensor.dtype]
    work, chunk, padded = _pad_chunks(tensor, size, buf)
    esz = work.element_size()
    base = work.data_ptr()
    tmp = buf(("ring", tensor.device.index, tensor.dtype), chunk,
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
        kern.add_inplace(base + r_idx * chunk * esz, tbase, chunk, dt,
                         stream)

    owned = (rank + 1) % size
    if average:
        if tensor.dtype == torch.float32:
            kern.scale_f32(base + owned * chunk * esz, 1.0 / size, chunk,
                           stream)
        else:
            kern.reduce_columns(base + owned * chunk * esz, 0, 0, 0, chunk,
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


# backwards-compatible private names (benchmarks, round-1 callers)
def _fullmesh(tensor, g, rank, size, average):
    comm, _ = _comm_of(g)
    kern = load_native("_kernels")
    fullmesh_all_reduce(tensor, comm, kern, rank, size, average, depth=1,
                        streams=_HipStreams(tensor.device))
    return tensor


def _ring(tensor, g, rank, size, average):
    comm, _ = _comm_of(g)
    kern = load_native("_kernels")
    ring_all_reduce(tensor, comm, kern, rank, size, average,
                    stream=torch.cuda.current_stream().cuda_stream)
    return tensor
