"""Native "tcp" backend specifics (dist/tcp_backend.py +
csrc/rcclx.cpp TcpMesh): the self-owned CPU plumbing layer.
torch.distributed is never on this path — _fn_assert_no_tdist proves
it stays unimported in the child processes."""

import sys

import torch

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.dist.launcher import launch


# ---- child functions (module-level: spawn requires picklability) --------

def _fn_any_source(rank, size):
    # tuto.md:90 source-less receive: root collects from whoever
    # arrives first
    if rank == 0:
        seen = set()
        for _ in range(size - 1):
            t = torch.zeros(2)
            src = dist.recv(t, src=None)
            assert t[0].item() == float(src * 10)
            seen.add(src)
        assert seen == set(range(1, size))
    else:
        dist.send(torch.full((2,), float(rank * 10)), dst=0)


def _fn_large_message(rank, size):
    # 4 MB p2p: exercises framing + zero-copy reads across multiple
    # socket buffers
    n = 1 << 20
    if rank == 0:
        t = torch.arange(n, dtype=torch.float32)
        dist.send(t, dst=1)
        r = torch.zeros(n)
        dist.recv(r, src=1)
        assert torch.equal(r, t * 2)
    else:
        t = torch.zeros(n)
        dist.recv(t, src=0)
        dist.send(t * 2, dst=0)


def _fn_world3_collectives(rank, size):
    t = torch.ones(5) * (rank + 1)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    assert torch.allclose(t, torch.full((5,), 6.0))
    t = torch.full((3,), float(rank))
    dist.all_reduce(t, op=dist.ReduceOp.AVG)
    assert torch.allclose(t, torch.full((3,), 1.0))
    b = torch.arange(4.0) if rank == 1 else torch.zeros(4)
    dist.broadcast(b, src=1)
    assert torch.equal(b, torch.arange(4.0))
    outs = [torch.zeros(2) for _ in range(size)]
    dist.all_gather(outs, torch.full((2,), float(rank)))
    for i, o in enumerate(outs):
        assert o[0].item() == float(i)
    dist.barrier()


def _fn_world4_new_group(rank, size):
    g = dist.new_group([1, 3])
    if rank in (1, 3):
        t = torch.ones(1) * (rank + 1)
        dist.all_reduce(t, op=dist.ReduceOp.SUM, g=g)
        assert t.item() == 6.0
        assert dist.get_world_size(g) == 2
    dist.barrier()


def _fn_sendrecv_ring(rank, size):
    # the paired-exchange primitive around a ring
    right = (rank + 1) % size
    left = (rank - 1) % size
    s = torch.full((8,), float(rank))
    r = torch.zeros(8)
    dist.sendrecv(s, right, r, left)
    assert r[0].item() == float(left)


def _fn_assert_no_tdist(rank, size):
    t = torch.ones(1)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    assert t.item() == float(size)
    assert "torch.distributed.distributed_c10d" not in sys.modules or \
        not getattr(sys.modules.get("torch.distributed"),
                    "is_initialized", lambda: False)(), \
        "torch.distributed got initialized on the native tcp path"


def _fn_ring_algorithms(rank, size):
    # the L3 hand-rolled algorithms over the native tcp p2p surface
    from dist_tuto_pth_amd.algorithms import (chunked_ring_all_reduce,
                                              ring_all_reduce)
    t = torch.ones(10) * (rank + 1)
    out = torch.zeros(10)
    ring_all_reduce(t, out)
    assert torch.allclose(out, torch.full((10,), 6.0))
    t2 = torch.ones(17) * (rank + 1)
    chunked_ring_all_reduce(t2)
    assert torch.allclose(t2, torch.full((17,), 6.0))


# ---- drivers ------------------------------------------------------------

def test_any_source_recv():
    launch(_fn_any_source, 3, backend="tcp")


def test_large_message():
    launch(_fn_large_message, 2, backend="tcp")


def test_world3_collectives():
    launch(_fn_world3_collectives, 3, backend="tcp")


def test_world4_new_group():
    launch(_fn_world4_new_group, 4, backend="tcp")


def test_sendrecv_ring_world4():
    launch(_fn_sendrecv_ring, 4, backend="tcp")


def test_no_torch_distributed():
    launch(_fn_assert_no_tdist, 2, backend="tcp")


def test_ring_algorithms_on_tcp():
    launch(_fn_ring_algorithms, 3, backend="tcp")


def _fn_any_source_skips_foreign_sizes(rank, size):
    # regression (rcclx.cpp recv_any): an any-source receive must not
    # consume a queued frame of a DIFFERENT size — that frame belongs
    # to a later targeted recv (the original bug ate a peer's 4 B
    # destroy-barrier token mid-wait and corrupted the barrier).
    import time
    if rank == 0:
        t8, t4 = torch.zeros(2), torch.zeros(1)
        src = dist.recv(t8, src=None)   # rank 1's 4 B frame is already
        assert src == 2                 # queued; must be skipped
        assert t8[0].item() == 20.0
        dist.recv(t4, src=1)
        assert t4[0].item() == 10.0
    elif rank == 1:
        dist.send(torch.full((1,), 10.0), dst=0)
    else:
        time.sleep(0.3)
        dist.send(torch.full((2,), 20.0), dst=0)


def test_any_source_skips_foreign_sizes():
    launch(_fn_any_source_skips_foreign_sizes, 3, backend="tcp")
