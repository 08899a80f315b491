"""Self-owned peer-copy transport for the hand-tuned all-reduce
(BASELINE north star "pipelines
hipMemcpyPeerAsync across the 7 xGMI links").

The RCCL-backed fullmesh (algorithms/xgmi.py) hand-writes the local
reduction but rides ``ncclSend/ncclRecv`` for the wire.  This module
owns the wire too:

  * each rank hipMallocs a mesh buffer of ``size-1`` rows and exports
    it with ``hipIpcGetMemHandle`` (csrc/rcclx.cpp ``DeviceBuffer``);
  * the 64-byte handles travel through the same C++ TCP store used for
    rendezvous (the tuto.md:409-418 master/worker pattern);
  * a "send" is then a stream-ordered ``hipMemcpyAsync`` into the
    IPC-mapped peer pointer — a one-sided xGMI push, no RCCL anywhere;
  * stage boundaries are store-counter barriers after a local stream
    sync (writes are visible once the pushing stream drains).

Sender s writes into receiver r's DENSE slot ``(s - r - 1) mod size``
— the same scratch layout as the RCCL fullmesh, so one
``reduce_columns`` call folds all peer rows at fixed stride.

Three barriers per all-reduce (exchange done / reduce done so rows may
be reused / gather done).  At 256 MB messages a ~100 us store barrier
is noise; the transport is for bandwidth-bound sizes, RCCL remains the
default for small/latency-bound messages.

The algorithm core takes an injected transport; tests/test_ipc_logic.py
drives the exact slot/offset math at worlds 2-8 through a fake
in-memory transport, and tests/test_ipc_gpu.py proves the real handle
plumbing parent<->child on one device.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from ..utils.native import load_native
from .xgmi import _DTYPE, _buf, _pad_chunks


class IpcTransport:
    """One-sided mesh transport over hipIpc handles + the TCP store."""

    def __init__(self, store, rank: int, size: int, row_bytes: int,
                 device: int = 0, tag: str = "ipc0"):
        rx = load_native("_rcclx")
        self._rx = rx
        self.store = store
        self.rank = rank
        self.size = size
        # 256-byte row alignment keeps reduce_columns' vector paths on
        # aligned strides for every dtype
        self.row_bytes = (row_bytes + 255) // 256 * 256
        self.tag = tag
        self._phase = 0
        nrows = max(size - 1, 1)
        self.buf = rx.DeviceBuffer(nrows * self.row_bytes, device)
        store.set(f"{tag}:h:{rank}", self.buf.ipc_handle())
        self.peer_base: Dict[int, int] = {}
        for p in range(size):
            if p == rank:
                continue
            h = store.get(f"{tag}:h:{p}")
            self.peer_base[p] = rx.ipc_open(h)

    # row index where RECEIVER r keeps data from SENDER s
    @staticmethod
    def slot(sender: int, receiver: int, size: int) -> int:
        return (sender - receiver - 1) % size

    def row_ptr(self, slot_idx: int) -> int:
        return self.buf.ptr() + slot_idx * self.row_bytes

    def push(self, peer: int, dst_off: int, src_ptr: int, nbytes: int,
             stream: int):
        """One-sided write of ``nbytes`` from local ``src_ptr`` into
        ``peer``'s mesh row reserved for this rank, at ``dst_off``."""
        assert dst_off + nbytes <= self.row_bytes
        slot_idx = self.slot(self.rank, peer, self.size)
        dst = self.peer_base[peer] + slot_idx * self.row_bytes + dst_off
        self._rx.memcpy_async(dst, src_ptr, nbytes, stream)

    def copy_local(self, dst_ptr: int, src_ptr: int, nbytes: int,
                   stream: int):
        self._rx.memcpy_async(dst_ptr, src_ptr, nbytes, stream)

    def barrier(self, stream: int):
        """Drain the local stream (pushes become visible), then meet
        every rank at the store: the last arriver at the ADD counter
        SETs a done flag, everyone else blocks in GET — the store's
        server-side condition variable does the waiting (one RTT, no
        polling; the first version polled with 0.5 ms sleeps, which
        would have taxed a 256 MB all-reduce by ~30 %)."""
        self._rx.stream_sync(stream)
        key = f"{self.tag}:bar:{self._phase}"
        self._phase += 1
        if self.store.add(key, 1) >= self.size:
            self.store.set(key + ":done", b"1")
        else:
            self.store.get(key + ":done")

    def close(self):
        for p, ptr in self.peer_base.items():
            try:
                self._rx.ipc_close(ptr)
            except Exception:
                pass
        self.peer_base = {}


def fullmesh_all_reduce_ipc(tensor, tp, kern, rank: int, size: int,
                            average: bool = False, buf=_buf,
                            stream: Optional[int] = None):
    """Direct full-mesh reduce-scatter + all-gather with the transport's
    one-sided pushes as the wire (no RCCL).  ``tp`` is an IpcTransport
    (or the fake in tests); ``kern`` the reduction kernels."""
    dt = _DTYPE[tensor.dtype]
    if stream is None:
        stream = (torch.cuda.current_stream().cuda_stream
                  if tensor.is_cuda else 0)
    work, chunk, padded = _pad_chunks(tensor, size, buf)
    esz = work.element_size()
    base = work.data_ptr()
    nb = chunk * esz
    assert nb <= tp.row_bytes, \
        f"chunk {nb}B exceeds transport row capacity {tp.row_bytes}B"
    scale = (1.0 / size) if average else 1.0

    if size == 1:
        return tensor

    # stage 1 — push chunk p into rank p's mesh (all 7 links at once,
    # one-sided)
    for d in range(1, size):
        peer = (rank + d) % size
        tp.push(peer, 0, base + peer * nb, nb, stream)
    tp.barrier(stream)

    # stage 2 — fold the size-1 dense peer rows into the owned chunk
    kern.reduce_columns(base + rank * nb, tp.row_ptr(0), size - 1,
                        tp.row_bytes // esz, chunk, scale, dt, stream)
    # rows may only be reused once EVERY rank's reduce has consumed its
    # mesh
    tp.barrier(stream)

    # stage 3 — push the reduced owned chunk to every peer's mesh
    for d in range(1, size):
        peer = (rank + d) % size
        tp.push(peer, 0, base + rank * nb, nb, stream)
    tp.barrier(stream)

    # land the gathered chunks from the mesh rows into the result
    for d in range(1, size):
        peer = (rank + d) % size
        slot_idx = tp.slot(peer, rank, size)
        tp.copy_local(base + peer * nb, tp.row_ptr(slot_idx), nb, stream)

    if padded:
        tensor.view(-1).copy_(work[:tensor.numel()])
    return tensor
