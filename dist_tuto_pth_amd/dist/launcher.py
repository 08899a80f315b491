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


def launch(fn: Callable, size: int, backend: str = "gloo",
           master_port: Optional[int] = None, join: bool = True,
           timeout: Optional[float] = None):
    """Fork-join launcher (train_dist.py:138-147): start ``size``
    processes running ``init_processes(rank, size, fn, backend)`` and
    join them.  Raises if any child exits non-zero."""
    if master_port is None:
        from . import _free_port
        master_port = _free_port()
    ctx = mp.get_context("spawn")
    procs = []
    for rank in range(size):
        p = ctx.Process(target=init_processes,
                        args=(rank, size, fn, backend, "127.0.0.1",
                              master_port))
        p.start()
        procs.append(p)
    if not join:
        return procs
    for p in procs:
        p.join(timeout)
    for rank, p in enumerate(procs):
        if p.is_alive():
            p.terminate()
            raise TimeoutError(f"rank {rank} did not finish")
        if p.exitcode != 0:
            raise RuntimeError(f"rank {rank} exited with {p.exitcode}")
