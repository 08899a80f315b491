"""The ``torch.distributed``-style API surface (L2 of SURVEY.md §1).

This is the interface the reference tutorial consumes everywhere
(call sites: ptp.py:23-26, gloo.py:10-44, allreduce.py:10-32,
train_dist.py:84-99; catalogued at tuto.md:77-122 for point-to-point and
tuto.md:176-202 for the six collectives).  The reference inherits its
implementation from PyTorch 0.x THD (tuto.md:404-419); here the layer is
owned by this package:

  * backend ``"rccl"`` — the MI355X path: a native C++ extension
    (``csrc/rcclx.cpp``) driving RCCL over xGMI, one rank per GPU,
    rendezvous through a C++ TCP store (the master/worker handshake the
    tutorial describes at tuto.md:409-419, rebuilt natively).
  * backend ``"tcp"`` — the reference's CPU THD channel, owned natively:
    a C++ per-pair socket mesh with collectives composed from p2p
    (tcp_backend.py).
  * backend ``"gloo"`` — torch.distributed delegation, kept as a CPU
    cross-check.

Exposed (modern forms of the tutorial's 0.x API):
``init_process_group``, ``get_rank``, ``get_world_size``, ``new_group``,
``send``/``recv``/``isend``/``irecv`` (tuto.md:87-112), the six
collectives ``broadcast``/``reduce``/``all_reduce``/``scatter``/
``gather``/``all_gather`` (tuto.md:197-202), ``ReduceOp`` with
SUM/PRODUCT/MAX/MIN (+AVG) (tuto.md:190-193), request objects with
``.wait()`` (tuto.md:97), ``group.WORLD``, ``barrier``, and the legacy
root-split pair ``gather_send``/``gather_recv`` (ptp.py:17-19).
"""

from __future__ import annotations

import os
import socket
import time
from typing import List, Optional, Sequence

import torch

__all__ = [
    "ReduceOp",
    "reduce_op",
    "Work",
    "group",
    "init_process_group",
    "destroy_process_group",
    "is_initialized",
    "get_backend",
    "get_rank",
    "get_world_size",
    "new_group",
    "send",
    "recv",
    "isend",
    "irecv",
    "sendrecv",
    "broadcast",
    "reduce",
    "all_reduce",
    "scatter",
    "gather",
    "all_gather",
    "all_to_all",
    "reduce_scatter",
    "gather_send",
    "gather_recv",
    "barrier",
]


# --------------------------------------------------------------------------
# Reduce ops (tuto.md:190-193: SUM / PRODUCT / MAX / MIN; +AVG for the
# fused grad-average path)
# --------------------------------------------------------------------------
class ReduceOp:
    SUM = "sum"
    PRODUCT = "product"
    MAX = "max"
    MIN = "min"
    AVG = "avg"


# Legacy 0.x alias: the scripts write ``dist.reduce_op.SUM``
# (gloo.py:44, train_dist.py:99).
reduce_op = ReduceOp


class Work:
    """A pending asynchronous operation (``DistributedRequest`` of
    tuto.md:97).  ``wait()`` blocks until the buffer may be used."""

    def __init__(self, waiter=None):
        self._waiter = waiter
        self._done = waiter is None

    def wait(self):
        if not self._done:
            self._waiter()
            self._done = True
        return True

    def is_completed(self) -> bool:
        return self._done


class ProcessGroup:
    """A communicator over a subset of ranks (``dist.new_group`` of
    tuto.md:182-184)."""

    def __init__(self, ranks: Sequence[int], backend_impl, name: str):
        self.ranks = list(ranks)
        self._impl = backend_impl
        self.name = name

    def size(self) -> int:
        return len(self.ranks)

    def rank(self) -> int:
        """This process's rank *within the group* (-1 if not a member)."""
        g = _state.global_rank
        return self.ranks.index(g) if g in self.ranks else -1

    def global_rank(self, group_rank: int) -> int:
        return self.ranks[group_rank]


class group:
    """Namespace holding the default group, mirroring
    ``dist.group.WORLD`` (ptp.py:14)."""

    WORLD: Optional[ProcessGroup] = None


class _State:
    def __init__(self):
        self.backend: Optional[str] = None
        self.global_rank: int = -1
        self.world_size: int = -1
        self.groups: List[ProcessGroup] = []
        self.device: Optional[torch.device] = None


_state = _State()


# --------------------------------------------------------------------------
# Rendezvous (tuto.md:421-457): env:// , tcp://host:port , file://path
# --------------------------------------------------------------------------
def _parse_init(init_method: str, world_size: int, rank: int):
    """Resolve (master_addr, master_port, world_size, rank) from the init
    method, honoring the env contract MASTER_ADDR/MASTER_PORT/WORLD_SIZE/
    RANK (tuto.md:425-428)."""
    if init_method is None or init_method == "env://":
        addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
        port = int(os.environ.get("MASTER_PORT", "29500"))
        if world_size in (None, -1):
            world_size = int(os.environ["WORLD_SIZE"])
        if rank in (None, -1):
            rank = int(os.environ["RANK"])
        return addr, port, world_size, rank
    if init_method.startswith("tcp://"):
        hostport = init_method[len("tcp://"):]
        host, port = hostport.rsplit(":", 1)
        return host, int(port), world_size, rank
    if init_method.startswith("file://"):
        # file rendezvous (tuto.md:430-437): ranks agree through a shared
        # file; we still need a TCP endpoint for the store, so rank 0
        # writes its address into the file under an fcntl lock.
        import fcntl

        path = init_method[len("file://"):]
        if rank == 0:
            host = "127.0.0.1"
            port = _free_port()
            with open(path, "w") as f:
                fcntl.lockf(f, fcntl.LOCK_EX)
                f.write(f"{host}:{port}\n")
                f.flush()
                fcntl.lockf(f, fcntl.LOCK_UN)
            return host, port, world_size, rank
        deadline = time.time() + 300.0
        while time.time() < deadline:
            try:
                with open(path) as f:
                    fcntl.lockf(f, fcntl.LOCK_SH)
                    line = f.readline().strip()
                    fcntl.lockf(f, fcntl.LOCK_UN)
                if line:
                    host, port = line.rsplit(":", 1)
                    return host, int(port), world_size, rank
            except FileNotFoundError:
                pass
            time.sleep(0.05)
        raise TimeoutError(f"file:// rendezvous timed out on {path}")
    raise ValueError(f"unsupported init_method {init_method!r}")


def _free_port() -> int:
    """A port P such that BOTH P and P+1 were bindable at probe time.
    The native backends put their store on MASTER_PORT+1 (the launcher
    rendezvous owns MASTER_PORT itself under torchrun), and probing
    only P let a live ephemeral connection on P+1 fail rank 0's store
    bind roughly once per ~15 heavy multi-process test runs — the
    peers then sat in connect timeouts, which looked like a hang in
    the any-source test.  Still TOCTOU-racy (launch() retries on a
    fresh pair for that), but the window is now the bind gap, not a
    whole unprobed port."""
    for _ in range(64):
        a = socket.socket()
        a.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        a.bind(("127.0.0.1", 0))
        port = a.getsockname()[1]
        b = socket.socket()
        b.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        try:
            b.bind(("127.0.0.1", port + 1))
        except OSError:
            a.close()
            b.close()
            continue
        b.close()
        a.close()
        return port
    raise RuntimeError("no free adjacent port pair found")


# --------------------------------------------------------------------------
# init / teardown
# --------------------------------------------------------------------------
def init_process_group(
    backend: str = "rccl",
    init_method: Optional[str] = None,
    world_size: int = -1,
    rank: int = -1,
    device_id: Optional[int] = None,
    group_name: str = "",
):
    """Initialize the default process group (``dist.init_process_group``
    of tuto.md:404-419; call sites ptp.py:34, gloo.py:54,
    train_dist.py:134).

    backend:
      * ``"rccl"`` (alias ``"nccl"``): native RCCL-over-xGMI backend,
        one rank per MI355X.  Device defaults to ``rank % device_count``.
      * ``"tcp"``: the native CPU socket-mesh backend (config 1).
      * ``"gloo"``: torch.distributed delegation (CPU cross-check).
    """
    if _state.backend is not None:
        raise RuntimeError("process group already initialized")
    backend = backend.lower()
    if backend == "nccl":
        backend = "rccl"

    addr, port, world_size, rank = _parse_init(init_method, world_size, rank)

    if backend == "tcp":
        # the reference's 'tcp' THD backend (ptp.py:30), owned natively:
        # per-pair sockets + p2p-composed collectives (tcp_backend.py).
        impl = _TcpBackend(addr, port, world_size, rank)
    elif backend == "gloo":
        # torch.distributed delegation, kept as a CPU cross-check only
        impl = _GlooBackend(addr, port, world_size, rank)
    elif backend == "rccl":
        impl = _RcclBackend(addr, port, world_size, rank, device_id)
    else:
        raise ValueError(f"unknown backend {backend!r}")

    _state.backend = backend
    _state.global_rank = rank
    _state.world_size = world_size
    group.WORLD = ProcessGroup(list(range(world_size)), impl, "world")
    _state.groups = [group.WORLD]


def destroy_process_group():
    if _state.backend is None:
        return
    for g in _state.groups:
        g._impl.destroy()
    _state.backend = None
    _state.global_rank = -1
    _state.world_size = -1
    _state.groups = []
    group.WORLD = None


def is_initialized() -> bool:
    return _state.backend is not None


def get_backend() -> str:
    _require_init()
    return _state.backend


def get_rank(g: Optional[ProcessGroup] = None) -> int:
    _require_init()
    return _state.global_rank if g is None else g.rank()


def get_world_size(g: Optional[ProcessGroup] = None) -> int:
    _require_init()
    return _state.world_size if g is None else g.size()


def _require_init():
    if _state.backend is None:
        raise RuntimeError("default process group not initialized")


def _resolve(g) -> ProcessGroup:
    _require_init()
    if g is None or g == 0:  # legacy positional group=0 (train_dist.py:99)
        return group.WORLD
    if isinstance(g, ProcessGroup):
        return g
    raise TypeError(f"not a process group: {g!r}")


def new_group(ranks: Sequence[int]) -> ProcessGroup:
    """Create a sub-group communicator (tuto.md:182-184).  Collective:
    every rank of the default group must call with the same ``ranks``."""
    _require_init()
    ranks = sorted(ranks)
    impl = group.WORLD._impl.split(ranks)
    g = ProcessGroup(ranks, impl, f"sub{len(_state.groups)}")
    _state.groups.append(g)
    return g


# --------------------------------------------------------------------------
# point-to-point (tuto.md:77-122)
# --------------------------------------------------------------------------
def send(tensor: torch.Tensor, dst: int, g=None, tag: int = 0):
    """Blocking send (tuto.md:87)."""
    gr = _resolve(g)
    gr._impl.send(tensor, dst, blocking=True)


def recv(tensor: torch.Tensor, src: Optional[int] = None, g=None, tag: int = 0) -> int:
    """Blocking receive (tuto.md:90).  Returns the source rank."""
    gr = _resolve(g)
    return gr._impl.recv(tensor, src, blocking=True)


def isend(tensor: torch.Tensor, dst: int, g=None, tag: int = 0) -> Work:
    """Non-blocking send returning a request (tuto.md:108)."""
    gr = _resolve(g)
    return gr._impl.send(tensor, dst, blocking=False)


def irecv(tensor: torch.Tensor, src: Optional[int] = None, g=None, tag: int = 0) -> Work:
    """Non-blocking receive returning a request (tuto.md:112)."""
    gr = _resolve(g)
    return gr._impl.recv(tensor, src, blocking=False)


def sendrecv(send_tensor: torch.Tensor, dst: int,
             recv_tensor: torch.Tensor, src: int, g=None):
    """Paired exchange: send ``send_tensor`` to ``dst`` while receiving
    ``recv_tensor`` from ``src``, completing both before returning.

    This is the safe form of the reference's ring step
    (``isend(right)`` + blocking ``recv(left)``, allreduce.py:24-25).
    On RCCL the un-paired pattern is a latent deadlock: every rank's
    send kernel sits on the stream ahead of its recv kernel, and each
    send waits for the *peer's* recv — a cycle once the message exceeds
    RCCL's internal buffering.  Here both ops are posted inside one
    ``ncclGroupStart``/``End`` so RCCL fuses them into a single
    deadlock-free kernel; on gloo (CPU sockets, buffered) it is the
    plain isend+recv pair."""
    gr = _resolve(g)
    return gr._impl.sendrecv(send_tensor, dst, recv_tensor, src)


# --------------------------------------------------------------------------
# the six collectives (tuto.md:197-202)
# --------------------------------------------------------------------------
def broadcast(tensor, src: int, g=None, async_op: bool = False):
    gr = _resolve(g)
    w = gr._impl.broadcast(tensor, src)
    return w if async_op else w.wait()


def reduce(tensor, dst: int, op=ReduceOp.SUM, g=None, async_op: bool = False):
    gr = _resolve(g)
    w = gr._impl.reduce(tensor, dst, op)
    return w if async_op else w.wait()


def all_reduce(tensor, op=ReduceOp.SUM, g=None, async_op: bool = False):
    """In-place all-reduce (gloo.py:44, train_dist.py:99)."""
    gr = _resolve(g)
    w = gr._impl.all_reduce(tensor, op)
    return w if async_op else w.wait()


def scatter(tensor, scatter_list=None, src: int = 0, g=None, async_op: bool = False):
    gr = _resolve(g)
    w = gr._impl.scatter(tensor, scatter_list, src)
    return w if async_op else w.wait()


def gather(tensor, gather_list=None, dst: int = 0, g=None, async_op: bool = False):
    """Gather to root (ptp.py:26 — note the reference passes
    ``gather_list`` only on the root)."""
    gr = _resolve(g)
    w = gr._impl.gather(tensor, gather_list, dst)
    return w if async_op else w.wait()


def all_gather(tensor_list, tensor, g=None, async_op: bool = False):
    gr = _resolve(g)
    w = gr._impl.all_gather(tensor_list, tensor)
    return w if async_op else w.wait()


def all_to_all(output_list, input_list, g=None, async_op: bool = False):
    gr = _resolve(g)
    w = gr._impl.all_to_all(output_list, input_list)
    return w if async_op else w.wait()


def reduce_scatter(output, input_list, op=ReduceOp.SUM, g=None, async_op: bool = False):
    gr = _resolve(g)
    w = gr._impl.reduce_scatter(output, input_list, op)
    return w if async_op else w.wait()


def barrier(g=None):
    gr = _resolve(g)
    gr._impl.barrier()


# --------------------------------------------------------------------------
# legacy asymmetric gather pair (ptp.py:17-19): root calls gather_recv
# with the output list, non-roots call gather_send.
# --------------------------------------------------------------------------
def gather_recv(tensor_list, tensor, g=None):
    """Root side of the pair: the caller is the destination (ptp.py:17)."""
    gr = _resolve(g)
    return gr._impl.gather(tensor, tensor_list, gr.rank()).wait()


def gather_send(tensor, root: int = 0, g=None):
    gr = _resolve(g)
    return gr._impl.gather(tensor, None, root).wait()


# --------------------------------------------------------------------------
# gloo delegation backend (CPU plumbing; BASELINE config 1)
# --------------------------------------------------------------------------
class _GlooBackend:
    def __init__(self, addr, port, world_size, rank, _tdist_group=None):
        import torch.distributed as tdist

        self._tdist = tdist
        self._group = _tdist_group
        if _tdist_group is None:
            if not tdist.is_initialized():
                tdist.init_process_group(
                    "gloo",
                    init_method=f"tcp://{addr}:{port}",
                    world_size=world_size,
                    rank=rank,
                )
            self._group = tdist.group.WORLD
        self._op_map = {
            ReduceOp.SUM: tdist.ReduceOp.SUM,
            ReduceOp.PRODUCT: tdist.ReduceOp.PRODUCT,
            ReduceOp.MAX: tdist.ReduceOp.MAX,
            ReduceOp.MIN: tdist.ReduceOp.MIN,
        }

    # -- helpers ----------------------------------------------------------
    def _op(self, op):
        if op == ReduceOp.AVG:
            return None  # emulated: SUM then divide
        return self._op_map[op]

    def split(self, ranks):
        sub = self._tdist.new_group(ranks=ranks)
        b = _GlooBackend(None, None, None, None, _tdist_group=sub)
        b._is_sub = True
        return b

    def destroy(self):
        if getattr(self, "_is_sub", False):
            return
        if self._tdist.is_initialized():
            # drain in lockstep so no rank exits mid-collective
            self._tdist.barrier(group=self._tdist.group.WORLD)
            self._tdist.destroy_process_group()

    # -- p2p --------------------------------------------------------------
    def send(self, tensor, dst, blocking):
        d = self._tdist.get_global_rank(self._group, dst) \
            if self._group is not self._tdist.group.WORLD else dst
        if blocking:
            self._tdist.send(tensor, d, group=self._group)
            return None
        req = self._tdist.isend(tensor, d, group=self._group)
        return Work(req.wait)

    def recv(self, tensor, src, blocking):
        d = None
        if src is not None:
            d = self._tdist.get_global_rank(self._group, src) \
                if self._group is not self._tdist.group.WORLD else src
        if blocking:
            return self._tdist.recv(tensor, d, group=self._group)
        req = self._tdist.irecv(tensor, d, group=self._group)
        return Work(req.wait)

    def sendrecv(self, send_tensor, dst, recv_tensor, src):
        # gloo's socket transport buffers sends, so the plain pair is
        # already deadlock-free here
        req = self.send(send_tensor, dst, blocking=False)
        self.recv(recv_tensor, src, blocking=True)
        req.wait()

    # -- collectives ------------------------------------------------------
    def _gr(self, rank_in_group):
        """group-rank -> global rank for torch.distributed calls."""
        if self._group is self._tdist.group.WORLD:
            return rank_in_group
        return self._tdist.get_global_rank(self._group, rank_in_group)

    def broadcast(self, tensor, src):
        self._tdist.broadcast(tensor, self._gr(src), group=self._group)
        return Work()

    def reduce(self, tensor, dst, op):
        if op == ReduceOp.AVG:
            self._tdist.reduce(tensor, self._gr(dst),
                               op=self._tdist.ReduceOp.SUM, group=self._group)
            if self._tdist.get_rank(self._group) == dst:
                tensor.div_(self._tdist.get_world_size(self._group))
        else:
            self._tdist.reduce(tensor, self._gr(dst), op=self._op(op),
                               group=self._group)
        return Work()

    def all_reduce(self, tensor, op):
        if op == ReduceOp.AVG:
            self._tdist.all_reduce(tensor, op=self._tdist.ReduceOp.SUM,
                                   group=self._group)
            tensor.div_(self._tdist.get_world_size(self._group))
        else:
            self._tdist.all_reduce(tensor, op=self._op(op), group=self._group)
        return Work()

    def scatter(self, tensor, scatter_list, src):
        self._tdist.scatter(tensor, scatter_list, self._gr(src),
                            group=self._group)
        return Work()

    def gather(self, tensor, gather_list, dst):
        self._tdist.gather(tensor, gather_list, self._gr(dst),
                           group=self._group)
        return Work()

    def all_gather(self, tensor_list, tensor):
        self._tdist.all_gather(tensor_list, tensor, group=self._group)
        return Work()

    def all_to_all(self, output_list, input_list):
        # gloo has no alltoall: emulate with non-blocking p2p
        rank = self._tdist.get_rank(self._group)
        world = self._tdist.get_world_size(self._group)
        reqs = []
        for i in range(world):
            if i == rank:
                output_list[i].copy_(input_list[i])
            else:
                reqs.append(self._tdist.isend(input_list[i].contiguous(),
                                              self._gr(i),
                                              group=self._group))
                reqs.append(self._tdist.irecv(output_list[i], self._gr(i),
                                              group=self._group))
        for r in reqs:
            r.wait()
        return Work()

    def reduce_scatter(self, output, input_list, op):
        # gloo has no reduce_scatter: emulate with all_reduce of the concat
        # then slice — correctness path only (CPU tests).
        rank = self._tdist.get_rank(self._group)
        world = self._tdist.get_world_size(self._group)
        flat = torch.stack([t.reshape(-1) for t in input_list])
        if op == ReduceOp.AVG:
            self._tdist.all_reduce(flat, op=self._tdist.ReduceOp.SUM,
                                   group=self._group)
            flat.div_(world)
        else:
            self._tdist.all_reduce(flat, op=self._op(op), group=self._group)
        output.copy_(flat[rank].view_as(output))
        return Work()

    def barrier(self):
        self._tdist.barrier(group=self._group)


# --------------------------------------------------------------------------
# native backends (csrc/rcclx.cpp): RCCL-over-xGMI (the MI355X path) and
# the self-owned CPU "tcp" mesh
# --------------------------------------------------------------------------
from .rccl_backend import _RcclBackend  # noqa: E402  (needs Work/ReduceOp above)
from .tcp_backend import _TcpBackend  # noqa: E402
