"""RCCL-over-xGMI backend — the MI355X path of ``dist``.

The native layer (``csrc/rcclx.cpp``, built in-tree as
``_native/_rcclx.so``) owns what the reference tutorial inherits from
PyTorch 0.x THD (tuto.md:404-419): a C++ TCP store for rendezvous
(master/worker handshake, tuto.md:409-418), ``ncclUniqueId`` exchange,
``ncclCommInitRank`` (one rank per MI355X, device = local rank), the six
collectives (tuto.md:197-202), point-to-point send/recv (tuto.md:87-112)
and sub-communicators via ``ncclCommSplit`` (tuto.md:182-184).

This wrapper adapts torch tensors (raw data_ptr + current HIP stream) to
that native module.  It fails loudly if the native extension is missing
on a GPU machine — there is no silent eager fallback.
"""

from __future__ import annotations

import os

import torch

from ..utils.native import load_native


def _lazy_rcclx():
    return load_native("_rcclx")


_DTYPE = {
    torch.float32: 7,
    torch.float64: 8,
    torch.float16: 6,
    torch.bfloat16: 9,
    torch.int32: 2,
    torch.int64: 4,
    torch.uint8: 1,
    torch.int8: 0,
    torch.bool: 1,
}

_OP = {"sum": 0, "product": 1, "max": 2, "min": 3, "avg": 4}


def _check(t: torch.Tensor):
    if not t.is_cuda:
        raise RuntimeError(
            "the rccl backend operates on GPU tensors only (one rank per "
            "MI355X); use backend='gloo' for CPU plumbing")
    if not t.is_contiguous():
        raise RuntimeError("rccl backend requires contiguous tensors")


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


class _InactiveBackend:
    """Placeholder held by ranks that are not members of a sub-group
    (every op raises; mirrors torch.distributed's non-member groups)."""

    def destroy(self):
        pass

    def __getattr__(self, name):
        def _raise(*a, **k):
            raise RuntimeError(
                "this rank is not a member of the process group")
        return _raise


class _RcclBackend:
    """Backend implementation object held by a ProcessGroup.

    Peer ranks arriving here are *group-local*; the underlying RCCL
    communicator is built with the same local numbering (comm split by
    membership, key = position in the ranks list), so they map 1:1.
    """

    def __init__(self, addr, port, world_size, rank, device_id=None,
                 _comm=None, _store=None, _ws=None, _rk=None):
        rx = _lazy_rcclx()
        self._rx = rx
        if _comm is not None:           # sub-group path
            self._comm = _comm
            self._store = _store
            self._world = _ws
            self._rank = _rk
            return
        if not torch.cuda.is_available():
            raise RuntimeError(
                "rccl backend requires a GPU (torch.cuda.is_available() is "
                "False); use backend='gloo' on CPU-only machines")
        dev = (device_id if device_id is not None else rank) \
            % torch.cuda.device_count()
        torch.cuda.set_device(dev)
        # MASTER_PORT belongs to the launcher's own rendezvous when run
        # under torch.distributed.run; the native store binds port+1.
        self._store = rx.TcpStore(addr, port + 1, rank, world_size,
                                  rank == 0, 300_000)
        if rank == 0:
            uid = rx.get_unique_id()
            self._store.set("rccl:uid:world", uid)
        else:
            uid = self._store.get("rccl:uid:world")
        self._comm = rx.Comm(world_size, rank, uid, dev)
        self._world = world_size
        self._rank = rank
        self._ptag = "w"   # p2p-notification namespace (world group)
        self._barrier_buf = torch.zeros(1, device="cuda")

    _split_seq = 0

    # ------------------------------------------------------------------
    def split(self, ranks):
        member = self._rank in ranks
        color = 0 if member else -1     # -1 => NCCL_SPLIT_NOCOLOR
        key = ranks.index(self._rank) if member else 0
        sub = self._comm.split(color, key)
        _RcclBackend._split_seq += 1
        if not member:
            return _InactiveBackend()
        b = _RcclBackend(None, None, None, None, _comm=sub,
                         _store=self._store, _ws=len(ranks),
                         _rk=ranks.index(self._rank))
        b._ptag = f"g{_RcclBackend._split_seq}"
        b._barrier_buf = torch.zeros(1, device="cuda")
        return b

    def destroy(self):
        if getattr(self, "_comm", None) is not None:
            self._comm.destroy()
            self._comm = None

    # ------------------------------------------------------------------
    # p2p.  RCCL matching is by (peer, order), so a source-less receive
    # (tuto.md:90) cannot be posted directly: every dist-level send
    # bumps a per-pair sequence counter in the TCP store (a commutative
    # ADD, so creation order does not matter) and an any-source
    # receiver polls the peers' next-sequence keys, learns which rank
    # sent first, then posts the targeted ncclRecv.  Explicit receives
    # advance the same counters, so the two forms can interleave.  The
    # store round-trip (~tens of us) rides on the p2p demo path only —
    # collectives and the hand-tuned algorithms never touch it.
    # ------------------------------------------------------------------
    def _seq_key(self, dst, src, seq):
        return f"p2p:n:{self._ptag}:{dst}:{src}:{seq}"

    def _notify_send(self, dst):
        """Post the any-source sequence notification WITHOUT putting
        the store round-trip on the send's critical path: a single
        daemon thread drains a queue, which preserves per-pair
        sequence order (the only ordering the protocol needs)."""
        import queue as _q
        import threading as _t
        seqs = getattr(self, "_send_seq", None)
        if seqs is None:
            seqs = self._send_seq = {}
            self._notify_q = _q.Queue()

            def _drain():
                while True:
                    key = self._notify_q.get()
                    if key is None:
                        return
                    try:
                        self._store.add(key, 1)
                    except Exception:
                        return
            self._notify_t = _t.Thread(target=_drain, daemon=True)
            self._notify_t.start()
        s = seqs.get(dst, 0)
        seqs[dst] = s + 1
        self._notify_q.put(self._seq_key(dst, self._rank, s))

    def send(self, t, dst, blocking):
        from . import Work
        _check(t)
        if self._store is not None:
            self._notify_send(dst)
        self._comm.send(t.data_ptr(), t.numel(), _DTYPE[t.dtype], dst,
                        _stream())
        if blocking:
            torch.cuda.current_stream().synchronize()
            return None
        ev = self._rx.record_event(_stream())
        return Work(lambda: self._rx.event_wait(ev))

    def _resolve_any_source(self):
        import time as _time
        if self._store is None:
            raise RuntimeError(
                "any-source recv needs the store (sub-groups inherit it)")
        seqs = getattr(self, "_recv_seq", None)
        if seqs is None:
            seqs = self._recv_seq = {}
        deadline = _time.time() + 300.0
        while True:
            for s in range(self._world):
                if s == self._rank:
                    continue
                nxt = seqs.get(s, 0)
                if self._store.add(self._seq_key(self._rank, s, nxt),
                                   0) >= 1:
                    seqs[s] = nxt + 1
                    return s
            if _time.time() > deadline:
                raise TimeoutError("recv(src=None) timed out")
            _time.sleep(0.0002)

    def recv(self, t, src, blocking):
        from . import Work
        _check(t)
        if src is None:
            src = self._resolve_any_source()
        else:
            seqs = getattr(self, "_recv_seq", None)
            if seqs is None:
                seqs = self._recv_seq = {}
            seqs[src] = seqs.get(src, 0) + 1
        self._comm.recv(t.data_ptr(), t.numel(), _DTYPE[t.dtype], src,
                        _stream())
        if blocking:
            torch.cuda.current_stream().synchronize()
            return src
        ev = self._rx.record_event(_stream())
        return Work(lambda: self._rx.event_wait(ev))

    def sendrecv(self, send_t, dst, recv_t, src):
        """Paired exchange in ONE RCCL group — the deadlock-free form
        of the ring step (see dist.sendrecv).  Both transfers are fused
        into a single kernel on the current stream; the call returns
        when they are enqueued (stream-ordered consumers need no
        wait)."""
        _check(send_t)
        _check(recv_t)
        stream = _stream()
        self._comm.group_start()
        self._comm.send(send_t.data_ptr(), send_t.numel(),
                        _DTYPE[send_t.dtype], dst, stream)
        self._comm.recv(recv_t.data_ptr(), recv_t.numel(),
                        _DTYPE[recv_t.dtype], src, stream)
        self._comm.group_end()
        torch.cuda.current_stream().synchronize()

    # ------------------------------------------------------------------
    # collectives — enqueued on the caller's current stream; Work.wait()
    # is a no-op for stream-ordered consumers (matching torch.distributed
    # semantics for async_op=False).
    # ------------------------------------------------------------------
    def all_reduce(self, t, op):
        from . import Work
        _check(t)
        self._comm.all_reduce(t.data_ptr(), t.data_ptr(), t.numel(),
                              _DTYPE[t.dtype], _OP[op], _stream())
        return Work()

    def broadcast(self, t, src):
        from . import Work
        _check(t)
        self._comm.broadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                             _DTYPE[t.dtype], src, _stream())
        return Work()

    def reduce(self, t, dst, op):
        from . import Work
        _check(t)
        self._comm.reduce(t.data_ptr(), t.data_ptr(), t.numel(),
                          _DTYPE[t.dtype], _OP[op], dst, _stream())
        return Work()

    def all_gather(self, tensor_list, t):
        from . import Work
        _check(t)
        flat = torch.empty(self._world * t.numel(), dtype=t.dtype,
                           device=t.device)
        self._comm.all_gather(t.data_ptr(), flat.data_ptr(), t.numel(),
                              _DTYPE[t.dtype], _stream())
        for i, out in enumerate(tensor_list):
            out.copy_(flat[i * t.numel():(i + 1) * t.numel()].view_as(out))
        return Work()

    def all_gather_into_tensor(self, out_flat, t):
        """Zero-copy fast path (perf paths; not in the reference API)."""
        from . import Work
        _check(t)
        self._comm.all_gather(t.data_ptr(), out_flat.data_ptr(), t.numel(),
                              _DTYPE[t.dtype], _stream())
        return Work()

    def reduce_scatter(self, output, input_list, op):
        from . import Work
        _check(output)
        flat = torch.cat([t.reshape(-1) for t in input_list])
        self._comm.reduce_scatter(flat.data_ptr(), output.data_ptr(),
                                  output.numel(), _DTYPE[output.dtype],
                                  _OP[op], _stream())
        return Work()

    def reduce_scatter_tensor(self, output, input_flat, op):
        from . import Work
        _check(output)
        self._comm.reduce_scatter(input_flat.data_ptr(), output.data_ptr(),
                                  output.numel(), _DTYPE[output.dtype],
                                  _OP[op], _stream())
        return Work()

    def gather(self, t, gather_list, dst):
        from . import Work
        _check(t)
        if self._rank == dst:
            flat = torch.empty(self._world * t.numel(), dtype=t.dtype,
                               device=t.device)
            self._comm.gather(t.data_ptr(), flat.data_ptr(), t.numel(),
                              _DTYPE[t.dtype], dst, _stream())
            if gather_list is not None:
                for i, out in enumerate(gather_list):
                    out.copy_(
                        flat[i * t.numel():(i + 1) * t.numel()].view_as(out))
        else:
            self._comm.gather(t.data_ptr(), 0, t.numel(), _DTYPE[t.dtype],
                              dst, _stream())
        return Work()

    def scatter(self, t, scatter_list, src):
        from . import Work
        _check(t)
        if self._rank == src:
            flat = torch.cat([x.reshape(-1) for x in scatter_list])
            self._comm.scatter(flat.data_ptr(), t.data_ptr(), t.numel(),
                               _DTYPE[t.dtype], src, _stream())
        else:
            self._comm.scatter(0, t.data_ptr(), t.numel(), _DTYPE[t.dtype],
                               src, _stream())
        return Work()

    def all_to_all(self, output_list, input_list):
        from . import Work
        for t in input_list:
            _check(t)
        n = input_list[0].numel()
        sflat = torch.cat([x.reshape(-1) for x in input_list])
        rflat = torch.empty_like(sflat)
        self._comm.all_to_all(sflat.data_ptr(), rflat.data_ptr(), n,
                              _DTYPE[input_list[0].dtype], _stream())
        for i, out in enumerate(output_list):
            out.copy_(rflat[i * n:(i + 1) * n].view_as(out))
        return Work()

    def barrier(self):
        self._comm.all_reduce(self._barrier_buf.data_ptr(),
                              self._barrier_buf.data_ptr(), 1, 7, 0,
                              _stream())
        torch.cuda.current_stream().synchronize()

    # expose for hand-rolled algorithms (algorithms/ring.py)
    @property
    def comm(self):
        return self._comm
