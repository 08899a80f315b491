"""GPU tests for the IPC peer-copy transport (algorithms/ipc.py +
csrc/rcclx.cpp DeviceBuffer/ipc_open/memcpy_async).

The slot/offset math is CPU-tested in test_ipc_logic.py; here the real
handle plumbing runs on-device: hipIpcGetMemHandle export, child-
process hipIpcOpenMemHandle, one-sided writes, and a full 2-rank
fullmesh all-reduce with BOTH ranks on cuda:0 (IPC works between any
two processes on the node, so a 1-GPU box exercises the whole
transport — only the link under the copy differs at world>1)."""

import multiprocessing as mp
import os
import socket

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

from dist_tuto_pth_amd.utils.native import load_native  # noqa: E402


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_device_buffer_roundtrip():
    rx = load_native("_rcclx")
    buf = rx.DeviceBuffer(1 << 16, 0)
    assert buf.nbytes() == 1 << 16
    t = torch.arange(1024, dtype=torch.float32, device="cuda:0")
    out = torch.zeros_like(t)
    s = torch.cuda.current_stream().cuda_stream
    rx.memcpy_async(buf.ptr(), t.data_ptr(), t.numel() * 4, s)
    rx.memcpy_async(out.data_ptr(), buf.ptr(), t.numel() * 4, s)
    torch.cuda.synchronize()
    assert torch.equal(out, t)


def test_ipc_handle_is_exportable():
    rx = load_native("_rcclx")
    buf = rx.DeviceBuffer(4096, 0)
    h = buf.ipc_handle()
    assert isinstance(h, bytes) and len(h) == 64


def _child_open_and_write(handle, nbytes, q):
    try:
        torch.cuda.set_device(0)
        rx = load_native("_rcclx")
        ptr = rx.ipc_open(handle)
        n = nbytes // 4
        t = torch.empty(n, dtype=torch.float32, device="cuda:0")
        s = torch.cuda.current_stream().cuda_stream
        rx.memcpy_async(t.data_ptr(), ptr, nbytes, s)
        torch.cuda.synchronize()
        ok = torch.equal(t, torch.full((n,), 7.0, device="cuda:0"))
        t.fill_(11.0)
        rx.memcpy_async(ptr, t.data_ptr(), nbytes, s)
        rx.stream_sync(s)
        rx.ipc_close(ptr)
        q.put(("ok", bool(ok)))
    except Exception as e:  # noqa: BLE001
        q.put(("err", repr(e)))


def test_ipc_handle_cross_process():
    """Parent exports, child opens + verifies + writes back, parent
    sees the write — the one-sided push the transport is built on."""
    rx = load_native("_rcclx")
    nbytes = 1 << 14
    buf = rx.DeviceBuffer(nbytes, 0)
    n = nbytes // 4
    t = torch.full((n,), 7.0, device="cuda:0")
    s = torch.cuda.current_stream().cuda_stream
    rx.memcpy_async(buf.ptr(), t.data_ptr(), nbytes, s)
    torch.cuda.synchronize()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_child_open_and_write,
                    args=(buf.ipc_handle(), nbytes, q))
    p.start()
    kind, val = q.get(timeout=120)
    p.join(timeout=30)
    assert kind == "ok", val
    assert val, "child did not see the parent's pattern"
    out = torch.zeros(n, device="cuda:0")
    rx.memcpy_async(out.data_ptr(), buf.ptr(), nbytes, s)
    torch.cuda.synchronize()
    assert torch.equal(out, torch.full((n,), 11.0, device="cuda:0")), \
        "parent did not see the child's one-sided write"


def _rank_fn(rank, world, port, numel, q):
    try:
        torch.cuda.set_device(0)
        rx = load_native("_rcclx")
        from dist_tuto_pth_amd.algorithms.ipc import (
            IpcTransport, fullmesh_all_reduce_ipc)
        store = rx.TcpStore("127.0.0.1", port, rank, world, rank == 0,
                            120_000)
        k = load_native("_kernels")
        g = torch.Generator().manual_seed(123 + rank)
        t = torch.randn(numel, generator=g).cuda()
        expect = sum(torch.randn(numel,
                                 generator=torch.Generator()
                                 .manual_seed(123 + r))
                     for r in range(world)).cuda()
        chunk_cap = ((numel + world - 1) // world + 16) * 4
        tp = IpcTransport(store, rank, world, chunk_cap, device=0,
                          tag="t2")
        fullmesh_all_reduce_ipc(t, tp, k, rank, world)
        torch.cuda.synchronize()
        ok = torch.allclose(t, expect, rtol=1e-5, atol=1e-5)
        tp.close()
        q.put(("ok", bool(ok), rank))
    except Exception as e:  # noqa: BLE001
        q.put(("err", repr(e), rank))


def test_ipc_fullmesh_two_ranks_one_device():
    """Full end-to-end: store handle exchange, IPC mesh, one-sided
    exchange/gather pushes, reduce_columns — 2 ranks sharing cuda:0."""
    world, numel = 2, 100_000
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_fn,
                         args=(r, world, port, numel, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for kind, val, rank in results:
        assert kind == "ok", f"rank {rank}: {val}"
        assert val, f"rank {rank}: all-reduce mismatch"
