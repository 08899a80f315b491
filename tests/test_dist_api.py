"""API-surface tests over the CPU backends, fork-local-ranks over
loopback (the reference's own harness shape): every L2
primitive the tutorial catalogues (tuto.md:77-202), run on BOTH the
native "tcp" backend (this repo's own sockets, tcp_backend.py) and the
gloo cross-check delegation."""

import os

import pytest
import torch

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.dist.launcher import launch


# ---- child functions (module-level: spawn requires picklability) --------

def _fn_ptp_blocking(rank, size):
    # tuto.md:87-95 blocking send/recv of one fp32 scalar
    t = torch.zeros(1)
    if rank == 0:
        t += 1
        dist.send(t, dst=1)
    else:
        src = dist.recv(t, src=0)
        assert src == 0
        assert t.item() == 1.0


def _fn_ptp_nonblocking(rank, size):
    # tuto.md:100-120 isend/irecv + wait
    t = torch.zeros(1)
    if rank == 0:
        t += 1
        req = dist.isend(t, dst=1)
    else:
        req = dist.irecv(t, src=0)
    req.wait()
    if rank == 1:
        assert t.item() == 1.0


def _fn_all_reduce(rank, size):
    t = torch.ones(2, 2) * (rank + 1)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    assert torch.allclose(t, torch.full((2, 2), 3.0))
    t = torch.ones(3) * (rank + 2)
    dist.all_reduce(t, op=dist.ReduceOp.PRODUCT)
    assert torch.allclose(t, torch.full((3,), 6.0))
    t = torch.tensor([float(rank)])
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    assert t.item() == 1.0
    t = torch.tensor([float(rank)])
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    assert t.item() == 0.0
    # legacy alias spelling (gloo.py:44)
    t = torch.ones(1)
    dist.all_reduce(t, op=dist.reduce_op.SUM)
    assert t.item() == 2.0


def _fn_broadcast_reduce(rank, size):
    t = torch.arange(4.0) if rank == 0 else torch.zeros(4)
    dist.broadcast(t, src=0)
    assert torch.equal(t, torch.arange(4.0))
    t = torch.ones(2) * (rank + 1)
    dist.reduce(t, dst=0, op=dist.ReduceOp.SUM)
    if rank == 0:
        assert torch.allclose(t, torch.full((2,), 3.0))


def _fn_gather_scatter(rank, size):
    # gather (ptp.py:26): root receives everyone's tensor
    t = torch.ones(1) * (rank + 1)
    glist = [torch.zeros(1) for _ in range(size)] if rank == 0 else None
    dist.gather(t, gather_list=glist, dst=0)
    if rank == 0:
        assert sum(x.item() for x in glist) == 3.0
    # scatter
    out = torch.zeros(2)
    slist = [torch.full((2,), float(i)) for i in range(size)] \
        if rank == 0 else None
    dist.scatter(out, scatter_list=slist, src=0)
    assert torch.allclose(out, torch.full((2,), float(rank)))


def _fn_all_gather(rank, size):
    t = torch.full((2,), float(rank))
    outs = [torch.zeros(2) for _ in range(size)]
    dist.all_gather(outs, t)
    for i, o in enumerate(outs):
        assert torch.allclose(o, torch.full((2,), float(i)))


def _fn_reduce_scatter_alltoall(rank, size):
    ins = [torch.full((3,), float(rank + 1 + i)) for i in range(size)]
    out = torch.zeros(3)
    dist.reduce_scatter(out, ins, op=dist.ReduceOp.SUM)
    # rank r gets sum over ranks of ins[r] = sum_k (k+1+r)
    expect = sum(k + 1 + rank for k in range(size))
    assert torch.allclose(out, torch.full((3,), float(expect)))
    a2a_in = [torch.full((1,), float(rank * 10 + i)) for i in range(size)]
    a2a_out = [torch.zeros(1) for _ in range(size)]
    dist.all_to_all(a2a_out, a2a_in)
    for i, o in enumerate(a2a_out):
        assert o.item() == i * 10 + rank


def _fn_gather_pair(rank, size):
    # legacy asymmetric pair (ptp.py:17-19)
    t = torch.ones(1)
    if rank == 0:
        tl = [torch.zeros(1) for _ in range(size)]
        dist.gather_recv(tl, t)
        assert sum(x.item() for x in tl) == float(size)
    else:
        dist.gather_send(t, root=0)


def _fn_new_group(rank, size):
    g = dist.new_group([0, 1])
    t = torch.ones(1) * (rank + 1)
    dist.all_reduce(t, op=dist.ReduceOp.SUM, g=g)
    assert t.item() == 3.0
    assert dist.get_world_size(g) == 2
    assert dist.get_rank(g) == rank
    dist.barrier()


def _fn_rank_world(rank, size):
    assert dist.get_rank() == rank
    assert dist.get_world_size() == size
    assert dist.is_initialized()
    assert dist.get_backend() in ("gloo", "tcp")
    dist.barrier()


def _fn_ptp_demo(rank, size):
    # the ptp.py:21-28 demo end-to-end: gather ones to root, sum == size
    t = torch.ones(1)
    tl = [torch.zeros(1) for _ in range(size)] if rank == 0 else None
    dist.gather(t, gather_list=tl, dst=0)
    if rank == 0:
        assert sum(x.item() for x in tl)== float(size)


# ---- drivers ------------------------------------------------------------

@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_ptp_blocking(backend):
    launch(_fn_ptp_blocking, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_ptp_nonblocking(backend):
    launch(_fn_ptp_nonblocking, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_all_reduce_ops(backend):
    launch(_fn_all_reduce, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_broadcast_reduce(backend):
    launch(_fn_broadcast_reduce, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_gather_scatter(backend):
    launch(_fn_gather_scatter, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_all_gather(backend):
    launch(_fn_all_gather, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_reduce_scatter_alltoall(backend):
    launch(_fn_reduce_scatter_alltoall, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_gather_send_recv_pair(backend):
    launch(_fn_gather_pair, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_new_group(backend):
    launch(_fn_new_group, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_rank_world_backend(backend):
    launch(_fn_rank_world, 2, backend=backend)


@pytest.mark.parametrize("backend", ["gloo", "tcp"])
def test_ptp_demo_world2(backend):
    launch(_fn_ptp_demo, 2, backend=backend)


# ---------------------------------------------------------------------------
# init methods beyond env:// (tuto.md:421-457: file:// with fcntl
# locking, tcp://host:port)
# ---------------------------------------------------------------------------
def _fn_initmethod_worker(rank, size, init_method):
    from dist_tuto_pth_amd import dist as d
    d.init_process_group("gloo", init_method=init_method, world_size=size,
                         rank=rank)
    try:
        t = torch.full((4,), float(rank + 1))
        d.all_reduce(t, op=d.ReduceOp.SUM)
        expect = float(sum(r + 1 for r in range(size)))
        assert torch.allclose(t, torch.full((4,), expect))
    finally:
        d.destroy_process_group()


def _spawn_initmethod(init_method, size=2):
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_fn_initmethod_worker,
                         args=(r, size, init_method)) for r in range(size)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
    for r, p in enumerate(procs):
        assert p.exitcode == 0, (r, p.exitcode)


def test_init_method_file():
    import tempfile
    d = tempfile.mkdtemp()
    path = os.path.join(d, "rdzv")
    _spawn_initmethod(f"file://{path}")


def test_init_method_tcp():
    from dist_tuto_pth_amd.dist import _free_port
    port = _free_port()
    _spawn_initmethod(f"tcp://127.0.0.1:{port}")
