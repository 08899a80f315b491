"""Native RCCL backend tests on a single MI355X: store rendezvous, comm
init, the six collectives + p2p (world=1 exercises the full native path;
multi-rank is covered by the gloo-path wrapper tests plus the driver's
multi-GPU bench — SURVEY.md §4)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.utils.native import load_native

DEV = "cuda:0"


@pytest.fixture(scope="module")
def world1():
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    dist.init_process_group("rccl", world_size=1, rank=0, device_id=0)
    yield
    dist.destroy_process_group()


def test_native_so_loaded():
    rx = load_native("_rcclx")
    assert rx.device_count() >= 1
    assert len(rx.get_unique_id()) == 128


def test_world1_all_reduce(world1):
    t = torch.randn(1 << 16, device=DEV)
    ref = t.clone()
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    torch.cuda.synchronize()
    assert torch.equal(t, ref)


def test_world1_all_reduce_ops(world1):
    for op in (dist.ReduceOp.SUM, dist.ReduceOp.PRODUCT,
               dist.ReduceOp.MAX, dist.ReduceOp.MIN, dist.ReduceOp.AVG):
        t = torch.randn(257, device=DEV)
        ref = t.clone()
        dist.all_reduce(t, op=op)
        torch.cuda.synchronize()
        assert torch.allclose(t, ref), op


def test_world1_broadcast_gather_scatter(world1):
    t = torch.randn(100, device=DEV)
    ref = t.clone()
    dist.broadcast(t, src=0)
    torch.cuda.synchronize()
    assert torch.equal(t, ref)

    glist = [torch.zeros(100, device=DEV)]
    dist.gather(t, gather_list=glist, dst=0)
    torch.cuda.synchronize()
    assert torch.equal(glist[0], ref)

    out = torch.zeros(100, device=DEV)
    dist.scatter(out, scatter_list=[ref.clone()], src=0)
    torch.cuda.synchronize()
    assert torch.equal(out, ref)


def test_world1_all_gather_reduce_scatter(world1):
    t = torch.randn(64, device=DEV)
    outs = [torch.zeros(64, device=DEV)]
    dist.all_gather(outs, t)
    torch.cuda.synchronize()
    assert torch.equal(outs[0], t)

    rs_out = torch.zeros(64, device=DEV)
    dist.reduce_scatter(rs_out, [t.clone()], op=dist.ReduceOp.SUM)
    torch.cuda.synchronize()
    assert torch.allclose(rs_out, t)


def test_world1_bf16_allreduce(world1):
    t = torch.randn(1 << 12, device=DEV, dtype=torch.bfloat16)
    ref = t.clone()
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    torch.cuda.synchronize()
    assert torch.equal(t.float(), ref.float())


def test_world1_barrier(world1):
    dist.barrier()


def test_world1_sendrecv_loopback(world1):
    """Grouped paired exchange (the deadlock-free ring step) —
    world-1 loopback proves the native group_start/send/recv/group_end
    path end to end."""
    t = torch.randn(4096, device=DEV)
    r = torch.zeros_like(t)
    dist.sendrecv(t, 0, r, 0)
    assert torch.equal(r, t)


def test_world1_reduce_columns_streams(world1):
    """The pipelined fullmesh building block: reduce_columns on the
    compute stream consuming data ordered by events from a second
    stream (the schedule of fullmesh_all_reduce at depth>1)."""
    k = load_native("_kernels")
    n, P = 1 << 18, 7
    dst = torch.randn(n, device=DEV)
    expect = dst.clone()
    scratch = torch.randn(P, n, device=DEV)
    expect += scratch.sum(0)
    comm_s = torch.cuda.Stream()
    with torch.cuda.stream(comm_s):
        scratch.mul_(1.0)  # some comm-stream work producing scratch
        ev = torch.cuda.Event()
        ev.record(comm_s)
    torch.cuda.current_stream().wait_event(ev)
    k.reduce_columns(dst.data_ptr(), scratch.data_ptr(), P, n, n, 1.0,
                     7, torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    assert torch.allclose(dst, expect, rtol=1e-5, atol=1e-5)


def test_world1_training_step(world1):
    """One full DP training step through the native backend + HIP ops."""
    from dist_tuto_pth_amd import ops
    from dist_tuto_pth_amd.models import Net
    from dist_tuto_pth_amd.optim import FusedSGD
    from dist_tuto_pth_amd.parallel import average_gradients

    torch.manual_seed(0)
    model = Net().to(DEV)
    opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.5)
    x = torch.randn(64, 1, 28, 28, device=DEV)
    tgt = torch.randint(0, 10, (64,), device=DEV)
    for _ in range(3):
        opt.zero_grad()
        loss = ops.log_softmax_nll(model.forward_logits(x), tgt)
        loss.backward()
        average_gradients(model)
        opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
