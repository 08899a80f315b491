"""Hand-rolled collective algorithm tests (allreduce.py:8-34 corrected;
multiple world sizes and non-divisible chunk counts per SURVEY.md §4)."""

import torch

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.algorithms import (chunked_ring_all_reduce,
                                          ring_all_reduce)
from dist_tuto_pth_amd.dist.launcher import launch


def _expected_sum(shape, size, seed_base=100):
    total = torch.zeros(shape)
    for r in range(size):
        g = torch.Generator().manual_seed(seed_base + r)
        total += torch.rand(shape, generator=g)
    return total


def _fn_ring(rank, size):
    g = torch.Generator().manual_seed(100 + rank)
    send = torch.rand(37, generator=g)
    recv = torch.zeros(37)
    ring_all_reduce(send, recv)
    assert torch.allclose(recv, _expected_sum((37,), size), atol=1e-5)
    # send must be untouched
    g2 = torch.Generator().manual_seed(100 + rank)
    assert torch.equal(send, torch.rand(37, generator=g2))


def _fn_chunked(rank, size):
    for numel in (64, 37, 5, 1, 257):   # non-divisible sizes included
        g = torch.Generator().manual_seed(100 + rank + numel)
        t = torch.rand(numel, generator=g)
        expect = torch.zeros(numel)
        for r in range(size):
            gg = torch.Generator().manual_seed(100 + r + numel)
            expect += torch.rand(numel, generator=gg)
        chunked_ring_all_reduce(t)
        assert torch.allclose(t, expect, atol=1e-5), numel


def _fn_chunked_avg(rank, size):
    t = torch.full((10,), float(rank + 1))
    chunked_ring_all_reduce(t, average=True)
    expect = sum(r + 1 for r in range(size)) / size
    assert torch.allclose(t, torch.full((10,), expect))


def _fn_vs_builtin(rank, size):
    g = torch.Generator().manual_seed(7 + rank)
    t = torch.rand(123, generator=g)
    ref = t.clone()
    dist.all_reduce(ref, op=dist.ReduceOp.SUM)
    chunked_ring_all_reduce(t)
    assert torch.allclose(t, ref, atol=1e-5)


def test_ring_world2():
    launch(_fn_ring, 2)


def test_ring_world3():
    launch(_fn_ring, 3)


def test_chunked_world2():
    launch(_fn_chunked, 2)


def test_chunked_world3():
    launch(_fn_chunked, 3)


def test_chunked_world4():
    launch(_fn_chunked, 4)


def test_chunked_average():
    launch(_fn_chunked_avg, 3)


def test_chunked_matches_builtin_allreduce():
    launch(_fn_vs_builtin, 3)


def test_pad_chunks_cpu():
    """_pad_chunks: 16-byte-aligned chunks, zero-fill, round-trip copy
    (the alignment contract the xGMI exchange relies on)."""
    from dist_tuto_pth_amd.algorithms.xgmi import _buf, _pad_chunks
    for numel, size in ((64, 4), (37, 4), (5, 8), (1, 2), (257, 8)):
        t = torch.arange(numel, dtype=torch.float32)
        work, chunk, padded = _pad_chunks(t, size, _buf)
        assert chunk * size >= numel
        assert chunk % (16 // 4) == 0          # 16-byte alignment in fp32
        assert torch.equal(work[:numel], t)
        if padded:
            assert work[numel:chunk * size].abs().sum() == 0
        else:
            assert work.data_ptr() == t.data_ptr()
