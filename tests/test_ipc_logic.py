"""Fake-transport tests for the one-sided IPC fullmesh all-reduce
(algorithms/ipc.py): the exact production slot/offset math at worlds
2-8 on CPU, with pushes as ctypes.memmove into the target rank's mesh
and barriers as threading.Barrier.  The real handle plumbing is proved
on-device by tests/test_ipc_gpu.py."""

import ctypes
import threading

import pytest
import torch

from dist_tuto_pth_amd.algorithms.ipc import (IpcTransport,
                                              fullmesh_all_reduce_ipc)
from tests.test_xgmi_logic import FakeKernels, _per_rank_buf


class FakeMeshFabric:
    def __init__(self, size, row_bytes):
        self.size = size
        self.row_bytes = (row_bytes + 255) // 256 * 256
        nrows = max(size - 1, 1)
        # one mesh tensor per rank (uint8 so raw byte math applies)
        self.mesh = [torch.zeros(nrows * self.row_bytes,
                                 dtype=torch.uint8)
                     for _ in range(size)]
        self.gate = threading.Barrier(size, timeout=30)


class FakeTransport:
    """Same surface as IpcTransport, in-memory."""

    def __init__(self, fabric: FakeMeshFabric, rank: int):
        self.fabric = fabric
        self.rank = rank
        self.size = fabric.size
        self.row_bytes = fabric.row_bytes

    slot = staticmethod(IpcTransport.slot)

    def row_ptr(self, slot_idx):
        return self.fabric.mesh[self.rank].data_ptr() \
            + slot_idx * self.row_bytes

    def push(self, peer, dst_off, src_ptr, nbytes, stream):
        assert dst_off + nbytes <= self.row_bytes
        slot_idx = self.slot(self.rank, peer, self.size)
        dst = self.fabric.mesh[peer].data_ptr() \
            + slot_idx * self.row_bytes + dst_off
        ctypes.memmove(dst, src_ptr, nbytes)

    def copy_local(self, dst_ptr, src_ptr, nbytes, stream):
        ctypes.memmove(dst_ptr, src_ptr, nbytes)

    def barrier(self, stream):
        self.fabric.gate.wait()


def _run_world(size, fn):
    errs = [None] * size

    def worker(r):
        try:
            fn(r)
        except BaseException as e:  # noqa: BLE001
            errs[r] = e

    threads = [threading.Thread(target=worker, args=(r,), daemon=True)
               for r in range(size)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
        assert not t.is_alive(), "rank thread hung"
    for e in errs:
        if e is not None:
            raise e


def test_slot_math_dense():
    for size in range(2, 9):
        for r in range(size):
            slots = sorted(IpcTransport.slot(s, r, size)
                           for s in range(size) if s != r)
            assert slots == list(range(size - 1))


@pytest.mark.parametrize("size", [2, 3, 5, 8])
@pytest.mark.parametrize("numel", [64, 1000, 12345])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_ipc_fullmesh_worlds(size, numel, dtype):
    torch.manual_seed(size * 31 + numel)
    inputs = [torch.randn(numel).to(dtype) for _ in range(size)]
    expect = torch.stack([t.float() for t in inputs]).sum(0)
    outs = [t.clone() for t in inputs]
    kern = FakeKernels()
    esz = inputs[0].element_size()
    chunk_cap = (((numel + size - 1) // size) + 16) * esz + 64
    fabric = FakeMeshFabric(size, chunk_cap)

    def fn(r):
        tp = FakeTransport(fabric, r)
        fullmesh_all_reduce_ipc(outs[r], tp, kern, r, size,
                                buf=_per_rank_buf())

    _run_world(size, fn)
    tol = 1e-5 if dtype == torch.float32 else 0.15
    for r in range(size):
        torch.testing.assert_close(outs[r].float(), expect, rtol=tol,
                                   atol=tol)


@pytest.mark.parametrize("size", [4, 8])
def test_ipc_fullmesh_average(size):
    numel = 4096
    torch.manual_seed(9)
    inputs = [torch.randn(numel) for _ in range(size)]
    expect = torch.stack(inputs).mean(0)
    outs = [t.clone() for t in inputs]
    kern = FakeKernels()
    fabric = FakeMeshFabric(size, (numel // size + 16) * 4 + 64)

    def fn(r):
        tp = FakeTransport(fabric, r)
        fullmesh_all_reduce_ipc(outs[r], tp, kern, r, size, average=True,
                                buf=_per_rank_buf())

    _run_world(size, fn)
    for r in range(size):
        torch.testing.assert_close(outs[r], expect, rtol=1e-5, atol=1e-5)
