"""Unit tests for the rccl backend's any-source receive protocol
(dist/rccl_backend.py): sends bump per-pair sequence keys in the store
(commutative ADD), an any-source receiver polls next-sequence keys and
posts a targeted recv.  Exercised here with a fake store + recording
comm on CPU; the targeted ncclRecv path itself is GPU-tested in
test_rccl_gpu.py."""

import threading

import torch

from dist_tuto_pth_amd.dist import rccl_backend as rb


class FakeStore:
    def __init__(self):
        self.d = {}
        self.mu = threading.Lock()

    def add(self, key, delta):
        with self.mu:
            v = self.d.get(key, 0) + delta
            self.d[key] = v
            return v


class RecComm:
    def __init__(self):
        self.sends = []
        self.recvs = []

    def send(self, ptr, n, dt, peer, stream):
        self.sends.append(peer)

    def recv(self, ptr, n, dt, peer, stream):
        self.recvs.append(peer)


class FakeT:
    is_cuda = True
    dtype = torch.float32

    def is_contiguous(self):
        return True

    def data_ptr(self):
        return 0

    def numel(self):
        return 4


def _mk(rank, world, store, monkeypatch):
    b = rb._RcclBackend.__new__(rb._RcclBackend)
    b._comm = RecComm()
    b._store = store
    b._rank = rank
    b._world = world
    b._ptag = "w"
    b._rx = None
    return b


def _patch_streams(monkeypatch):
    class FakeStream:
        cuda_stream = 0

        def synchronize(self):
            pass
    monkeypatch.setattr(rb.torch.cuda, "current_stream",
                        lambda *a, **k: FakeStream())


def test_any_source_resolves_to_sender(monkeypatch):
    _patch_streams(monkeypatch)
    store = FakeStore()
    world = 4
    b0 = _mk(0, world, store, monkeypatch)
    b2 = _mk(2, world, store, monkeypatch)
    b2.send(FakeT(), 0, blocking=True)           # rank 2 sends to 0
    src = b0.recv(FakeT(), None, blocking=True)  # any-source sees it
    assert src == 2
    assert b0._comm.recvs == [2]


def test_any_source_ordering_and_mixing(monkeypatch):
    """Two sends from one peer + one from another; an explicit recv
    consumes its notification so a later any-source does not re-match
    the same send."""
    _patch_streams(monkeypatch)
    store = FakeStore()
    world = 3
    b0 = _mk(0, world, store, monkeypatch)
    b1 = _mk(1, world, store, monkeypatch)
    b2 = _mk(2, world, store, monkeypatch)

    b1.send(FakeT(), 0, blocking=True)   # seq 0 from 1
    b1.send(FakeT(), 0, blocking=True)   # seq 1 from 1
    b2.send(FakeT(), 0, blocking=True)   # seq 0 from 2

    # explicit recv from 1 consumes 1's first notification
    assert b0.recv(FakeT(), 1, blocking=True) == 1
    # any-source now matches 1's SECOND send (seq 1), not the consumed one
    srcs = {b0.recv(FakeT(), None, blocking=True),
            b0.recv(FakeT(), None, blocking=True)}
    assert srcs == {1, 2}
    assert b0._recv_seq == {1: 2, 2: 1}


def test_any_source_timeout_is_bounded(monkeypatch):
    _patch_streams(monkeypatch)
    store = FakeStore()
    b0 = _mk(0, 2, store, monkeypatch)
    import time as _t
    real = _t.time
    t0 = real()
    monkeypatch.setattr(rb, "_RcclBackend", rb._RcclBackend)
    # shrink the deadline by monkeypatching time.time seen in the module

    calls = {"n": 0}

    def fake_time():
        calls["n"] += 1
        return real() + (400.0 if calls["n"] > 3 else 0.0)
    import time
    monkeypatch.setattr(time, "time", fake_time)
    try:
        b0.recv(FakeT(), None, blocking=True)
        raised = False
    except TimeoutError:
        raised = True
    assert raised
