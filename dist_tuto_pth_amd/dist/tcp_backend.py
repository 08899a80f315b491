"""Native "tcp" backend — this repo's own CPU plumbing layer.

The reference's ``backend='tcp'`` (ptp.py:30) is THD's CPU-only
channel: per-pair sockets established master/worker-style, p2p
send/recv plus all collectives implemented over them
(tuto.md:367-369, 404-419).  Round 1 delegated this to
torch.distributed's gloo; here the layer is owned end to end:

  * the wire is ``csrc/rcclx.cpp``'s ``TcpMesh`` — one TCP connection
    per rank pair (rank i accepts from j > i, connects to j < i;
    ports travel through the same C++ TCP store used for rendezvous),
    length-framed zero-copy reads/writes on tensor memory;
  * any-source receive (``recv(src=None)``, tuto.md:90) polls all pair
    sockets in C++ (``recv_any``);
  * the six collectives (tuto.md:197-202) are composed from p2p in
    Python: binomial-tree broadcast/reduce, reduce+broadcast
    all-reduce, ring all-gather, pairwise all-to-all — the textbook
    algorithms the tutorial's layer-3 chapter teaches;
  * sub-groups build a fresh sub-mesh over the member ranks
    (tuto.md:182-184).

torch.distributed is never imported on this path; the ``gloo``
backend remains available separately as a cross-check.
"""

from __future__ import annotations

import threading

import torch

from ..utils.native import load_native


def _check_cpu(t: torch.Tensor):
    if t.is_cuda:
        raise RuntimeError(
            "the tcp backend is CPU-only (tuto.md:367-369); use "
            "backend='rccl' for GPU tensors")
    if not t.is_contiguous():
        raise RuntimeError("tcp backend requires contiguous tensors")


def _nbytes(t: torch.Tensor) -> int:
    return t.numel() * t.element_size()


def _combine(acc: torch.Tensor, other: torch.Tensor, op: str):
    if op in ("sum", "avg"):
        acc.add_(other)
    elif op == "product":
        acc.mul_(other)
    elif op == "max":
        torch.maximum(acc, other, out=acc)
    elif op == "min":
        torch.minimum(acc, other, out=acc)
    else:
        raise ValueError(f"unknown reduce op {op!r}")


class _TcpBackend:
    """Backend implementation object held by a ProcessGroup (peer ranks
    arriving here are group-local; the mesh is built with the same
    local numbering)."""

    _split_seq = 0

    def __init__(self, addr, port, world_size, rank, _store=None,
                 _tag="mesh:world"):
        rx = load_native("_rcclx")
        self._rx = rx
        if _store is not None:
            self._store = _store
        else:
            # MASTER_PORT belongs to the launcher's own rendezvous when
            # run under torch.distributed.run; the native store binds
            # port+1 (same convention as the rccl backend).
            self._store = rx.TcpStore(addr, port + 1, rank, world_size,
                                      rank == 0, 120_000)
        self._mesh = rx.TcpMesh(self._store, _tag, rank, world_size)
        self._rank = rank
        self._world = world_size

    def destroy(self):
        # drain in lockstep so no rank closes its pair sockets while a
        # peer is still mid-collective (the gloo delegation does the
        # same with its barrier)
        if self._mesh is not None and self._world > 1:
            try:
                self.barrier()
            except Exception:
                pass
        self._mesh = None

    # ------------------------------------------------------------------
    # p2p (tuto.md:87-112)
    # ------------------------------------------------------------------
    def send(self, t, dst, blocking):
        from . import Work
        _check_cpu(t)
        if blocking:
            self._mesh.send(dst, t.data_ptr(), _nbytes(t))
            return None
        th = threading.Thread(
            target=self._mesh.send, args=(dst, t.data_ptr(), _nbytes(t)),
            daemon=True)
        th.start()
        return Work(th.join)

    def recv(self, t, src, blocking):
        from . import Work
        _check_cpu(t)
        if blocking:
            if src is None:
                return self._mesh.recv_any(t.data_ptr(), _nbytes(t))
            self._mesh.recv(src, t.data_ptr(), _nbytes(t))
            return src
        res = {}

        def _do():
            if src is None:
                res["src"] = self._mesh.recv_any(t.data_ptr(), _nbytes(t))
            else:
                self._mesh.recv(src, t.data_ptr(), _nbytes(t))
                res["src"] = src
        th = threading.Thread(target=_do, daemon=True)
        th.start()
        return Work(th.join)

    def sendrecv(self, send_t, dst, recv_t, src):
        req = self.send(send_t, dst, blocking=False)
        self.recv(recv_t, src, blocking=True)
        req.wait()

    # internal raw helpers (zero-copy: tensors must be contiguous so
    # the mesh reads/writes their memory directly)
    def _send_raw(self, t, peer):
        _check_cpu(t)
        self._mesh.send(peer, t.data_ptr(), _nbytes(t))

    def _recv_raw(self, t, peer):
        _check_cpu(t)
        self._mesh.recv(peer, t.data_ptr(), _nbytes(t))

    # ------------------------------------------------------------------
    # collectives (tuto.md:197-202), composed from p2p
    # ------------------------------------------------------------------
    def broadcast(self, t, src):
        from . import Work
        _check_cpu(t)
        size, rank = self._world, self._rank
        if size > 1:
            vrank = (rank - src) % size
            # binomial tree: recv once at the lowest set bit, then
            # relay downward
            mask = 1
            while mask < size:
                if vrank & mask:
                    self._recv_raw(t, (vrank - mask + src) % size)
                    break
                mask <<= 1
            mask >>= 1
            while mask:
                if vrank + mask < size:
                    self._send_raw(t, (vrank + mask + src) % size)
                mask >>= 1
        return Work()

    def _reduce_to(self, acc, dst, op):
        """Binomial-tree reduction of ``acc`` (modified in place on the
        root's path) toward ``dst``."""
        size, rank = self._world, self._rank
        vrank = (rank - dst) % size
        tmp = torch.empty_like(acc)
        mask = 1
        while mask < size:
            if vrank & mask:
                self._send_raw(acc, (vrank - mask + dst) % size)
                break
            if vrank + mask < size:
                self._recv_raw(tmp, (vrank + mask + dst) % size)
                _combine(acc, tmp, op)
            mask <<= 1

    def reduce(self, t, dst, op):
        from . import Work
        _check_cpu(t)
        if self._world == 1:
            return Work()
        acc = t.clone()
        self._reduce_to(acc, dst, op)
        if self._rank == dst:
            if op == "avg":
                acc /= self._world
            t.copy_(acc)
        return Work()

    def all_reduce(self, t, op):
        from . import Work
        _check_cpu(t)
        if self._world == 1:
            return Work()
        acc = t.clone()
        self._reduce_to(acc, 0, op)
        if self._rank == 0:
            if op == "avg":
                acc /= self._world
            t.copy_(acc)
        self.broadcast(t, 0)
        return Work()

    def gather(self, t, gather_list, dst):
        from . import Work
        _check_cpu(t)
        if self._rank == dst:
            for i in range(self._world):
                if i == self._rank:
                    if gather_list is not None:
                        gather_list[i].copy_(t)
                elif gather_list is not None:
                    _check_cpu(gather_list[i])
                    self._recv_raw(gather_list[i], i)
        else:
            self._send_raw(t, dst)
        return Work()

    def scatter(self, t, scatter_list, src):
        from . import Work
        _check_cpu(t)
        if self._rank == src:
            for i in range(self._world):
                if i == self._rank:
                    t.copy_(scatter_list[i])
                else:
                    self._send_raw(scatter_list[i].contiguous(), i)
        else:
            self._recv_raw(t, src)
        return Work()

    def all_gather(self, tensor_list, t):
        from . import Work
        _check_cpu(t)
        size, rank = self._world, self._rank
        tensor_list[rank].copy_(t)
        if size == 1:
            return Work()
        # ring all-gather: size-1 forwarding steps, each a safe
        # isend+recv pair over buffered sockets
        left = (rank - 1) % size
        right = (rank + 1) % size
        for i in range(size - 1):
            s_idx = (rank - i) % size
            r_idx = (rank - i - 1) % size
            req = self.send(tensor_list[s_idx].contiguous(), right,
                            blocking=False)
            self._recv_raw(tensor_list[r_idx], left)
            req.wait()
        return Work()

    def all_to_all(self, output_list, input_list):
        from . import Work
        size, rank = self._world, self._rank
        output_list[rank].copy_(input_list[rank])
        for d in range(1, size):
            dst = (rank + d) % size
            src = (rank - d) % size
            req = self.send(input_list[dst].contiguous(), dst,
                            blocking=False)
            self._recv_raw(output_list[src], src)
            req.wait()
        return Work()

    def reduce_scatter(self, output, input_list, op):
        from . import Work
        _check_cpu(output)
        size, rank = self._world, self._rank
        if size == 1:
            output.copy_(input_list[0])
            return Work()
        # reduce each destination's slice toward its owner
        for i in range(size):
            acc = input_list[i].reshape(-1).clone()
            self._reduce_to(acc, i, op)
            if rank == i:
                if op == "avg":
                    acc /= size
                output.copy_(acc.view_as(output))
        return Work()

    def barrier(self):
        t = torch.zeros(1)
        self.all_reduce(t, "sum")

    # ------------------------------------------------------------------
    def split(self, ranks):
        from .rccl_backend import _InactiveBackend
        _TcpBackend._split_seq += 1
        tag = f"mesh:sub{_TcpBackend._split_seq}"
        if self._rank not in ranks:
            return _InactiveBackend()
        b = _TcpBackend.__new__(_TcpBackend)
        b._rx = self._rx
        b._store = self._store
        b._mesh = self._rx.TcpMesh(self._store, tag,
                                   ranks.index(self._rank), len(ranks))
        b._rank = ranks.index(self._rank)
        b._world = len(ranks)
        return b
