"""Launcher failure-path behavior (dist/launcher.py): a failed launch
must raise promptly, reap every child it started, and retry once on a
fresh auto-assigned port.  These paths exist because a real flake (a
stolen rendezvous port) used to leave peers blocked in connect
timeouts and leaked live children wedged the interpreter at exit."""

import multiprocessing
import time

import pytest
import torch  # noqa: F401  (imported for spawn-context parity)

from dist_tuto_pth_amd import dist
from dist_tuto_pth_amd.dist import _free_port
from dist_tuto_pth_amd.dist.launcher import launch


def _alive_children():
    return [p for p in multiprocessing.active_children()
            if p.is_alive()]


def _fn_rank1_dies(rank, size):
    if rank == 1:
        raise SystemExit(3)
    # rank 0 would block forever waiting for rank 1's message
    t = torch.zeros(1)
    dist.recv(t, src=1)


def _fn_sleeps(rank, size):
    time.sleep(600)


def _fn_trivial(rank, size):
    dist.barrier()


def test_nonzero_exit_raises_and_reaps():
    with pytest.raises((RuntimeError, TimeoutError)):
        # retries=0: a deterministic child failure should not be retried
        launch(_fn_rank1_dies, 2, backend="tcp", timeout=20, retries=0)
    for _ in range(50):          # children reaped, nothing left alive
        if not _alive_children():
            break
        time.sleep(0.1)
    assert not _alive_children()


def test_timeout_raises_and_reaps():
    t0 = time.time()
    with pytest.raises((TimeoutError, RuntimeError)):
        launch(_fn_sleeps, 2, backend="tcp", timeout=3, retries=0)
    assert time.time() - t0 < 60  # shared deadline, not per-rank
    for _ in range(50):
        if not _alive_children():
            break
        time.sleep(0.1)
    assert not _alive_children()


def test_retry_recovers_from_stolen_port(monkeypatch):
    # first _free_port() answer is a port we then occupy ourselves (the
    # TOCTOU thief); the retry's fresh port must succeed.
    import socket

    import dist_tuto_pth_amd.dist as ddist

    real = ddist._free_port
    stolen = {}

    def stealing_free_port():
        port = real()
        if not stolen:
            s = socket.socket()
            s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            s.bind(("127.0.0.1", port + 1))   # occupy the store port
            s.listen(1)
            stolen["sock"] = s
        return port

    monkeypatch.setattr(ddist, "_free_port", stealing_free_port)
    try:
        launch(_fn_trivial, 2, backend="tcp", timeout=120)
    finally:
        stolen["sock"].close()


def test_free_port_pair_is_bindable():
    import socket
    for _ in range(5):
        p = _free_port()
        for q in (p, p + 1):
            s = socket.socket()
            s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            s.bind(("127.0.0.1", q))
            s.close()
