"""Process launch + rendezvous bootstrap (L1 of SURVEY.md §1).

Mirrors the fork-join ``__main__`` shape shared by all four reference
scripts (ptp.py:30-47, gloo.py:50-68, allreduce.py:49-67,
train_dist.py:130-147): fork ``size`` local processes over loopback,
each sets the env contract and calls ``init_process_group`` then the
user's ``fn(rank, size)``.  One rank maps to one MI355X
(device = rank) on the rccl backend — fixing the reference's
all-ranks-on-GPU-0 bug (gloo.py:40,59).
"""

from __future__ import annotations

import os
from typing import Callable, Optional

import torch.multiprocessing as mp

from . import destroy_process_group, init_process_group


def init_processes(rank: int, size: int, fn: Callable,
                   backend: str = "gloo",
                   master_addr: str = "127.0.0.1",
                   master_port: int = 29500):
    """Rendezvous bootstrap (train_dist.py:130-135): set
    MASTER_ADDR/MASTER_PORT, init the process group, run ``fn``."""
    os.environ["MASTER_ADDR"] = master_addr
    os.environ["MASTER_PORT"] = str(master_port)
    init_process_group(backend, init_method="env://", world_size=size,
                       rank=rank)
    try:
        fn(rank, size)
    finally:
        destroy_process_group()


def _start(fn: Callable, size: int, backend: str, master_port: int):
    ctx = mp.get_context("spawn")
    procs = []
    for rank in range(size):
        p = ctx.Process(target=init_processes,
                        args=(rank, size, fn, backend, "127.0.0.1",
                              master_port))
        p.start()
        procs.append(p)
    return procs


def _reap(procs) -> None:
    """Terminate and join every still-alive child (a failed launch must
    not leak processes: live non-daemon children block interpreter exit
    and hold the rendezvous ports the retry wants)."""
    import time as _time
    for p in procs:
        if p.is_alive():
            p.terminate()
    deadline = _time.time() + 10.0
    for p in procs:
        p.join(max(0.1, deadline - _time.time()))
        if p.is_alive():
            p.kill()
            p.join(5.0)


def launch(fn: Callable, size: int, backend: str = "gloo",
           master_port: Optional[int] = None, join: bool = True,
           timeout: Optional[float] = 300.0, retries: int = 1):
    """Fork-join launcher (train_dist.py:138-147): start ``size``
    processes running ``init_processes(rank, size, fn, backend)`` and
    join them.  Raises if any child exits non-zero or outlives
    ``timeout`` (a shared deadline; None = wait forever), reaping every
    child on the failure path.  When the port was auto-assigned, a
    failed attempt is retried once on a fresh port: the native store
    binds MASTER_PORT+1, and even with a pair-probing ``_free_port`` a
    concurrent process can steal either port between probe and bind."""
    import time as _time
    auto_port = master_port is None
    last_err: Optional[BaseException] = None
    for attempt in range(retries + 1 if auto_port and join else 1):
        if auto_port:
            from . import _free_port
            master_port = _free_port()
        procs = _start(fn, size, backend, master_port)
        if not join:
            return procs
        deadline = None if timeout is None else _time.time() + timeout
        err = None
        for rank, p in enumerate(procs):
            p.join(None if deadline is None
                   else max(0.1, deadline - _time.time()))
            if p.is_alive():
                err = TimeoutError(f"rank {rank} did not finish")
                break
            if p.exitcode != 0:
                err = RuntimeError(f"rank {rank} exited with "
                                   f"{p.exitcode}")
                break
        if err is None:
            return
        _reap(procs)
        last_err = err
    raise last_err
