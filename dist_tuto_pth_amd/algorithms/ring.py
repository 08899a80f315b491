"""Hand-rolled collective algorithms (L3 of SURVEY.md §1).

The reference builds a double-buffered ring all-reduce from p2p
primitives (allreduce.py:8-34 = gloo.py:8-34 = tuto.md:326-352) — but
the committed version is incorrect: its buffers are initialized to
zeros and never seeded with ``send``, and the accumulation adds the
local tensors instead of the received buffers.
``ring_all_reduce`` here implements the *intended* algorithm (what the
later official tutorial fixes with ``send.clone()`` /
``accum += recv_buff``), same double-buffered step structure.

``chunked_ring_all_reduce`` is the exercise the tutorial assigns at
tuto.md:354 ("implement ... DistributedDataParallel ... using
torch.chunk"): bandwidth-optimal reduce-scatter + all-gather,
``2*(size-1)`` steps on 1/size-sized chunks.

Both run over this package's ``dist`` p2p ops, so they work on the gloo
CPU path (tests) and on RCCL p2p over xGMI.  The *hand-tuned* MI355X
all-reduce (HIP reduction kernels pipelined with p2p across the 7 xGMI
links, K14 of SURVEY.md §2.4) lives in ``algorithms/xgmi.py``.
"""

from __future__ import annotations

import torch

from .. import dist


def ring_all_reduce(send: torch.Tensor, recv: torch.Tensor, g=None):
    """Corrected double-buffered ring all-reduce of the full tensor
    (allreduce.py:8-34 semantics, bugs fixed).  ``recv`` receives the
    sum over ranks of ``send``."""
    rank = dist.get_rank(g)
    size = dist.get_world_size(g)
    if size == 1:
        recv.copy_(send)
        return
    send_buff = send.clone()
    recv_buff = torch.zeros_like(send)
    accum = send.clone()

    left = ((rank - 1) + size) % size
    right = (rank + 1) % size

    for i in range(size - 1):
        # dist.sendrecv is the deadlock-free form of the reference's
        # isend(right) + recv(left) + wait() step (allreduce.py:24-32):
        # on RCCL both transfers are posted inside one group so the
        # send/recv cycle around the ring cannot stall (see
        # dist.sendrecv docstring); the i%2 buffer swap still forwards
        # last step's received buffer without a copy.
        if i % 2 == 0:
            dist.sendrecv(send_buff, right, recv_buff, left, g)
            accum += recv_buff
        else:
            dist.sendrecv(recv_buff, right, send_buff, left, g)
            accum += send_buff
    recv.copy_(accum)


def chunked_ring_all_reduce(tensor: torch.Tensor, g=None,
                            average: bool = False):
    """In-place bandwidth-optimal ring all-reduce: reduce-scatter then
    all-gather over ``size`` chunks (the tuto.md:354 exercise).

    Each of the ``2*(size-1)`` steps moves only ``numel/size`` elements,
    so total bytes on the wire per rank is ``2*(size-1)/size * bytes`` —
    the ring lower bound."""
    rank = dist.get_rank(g)
    size = dist.get_world_size(g)
    if size == 1:
        if average:
            pass
        return tensor
    flat = tensor.reshape(-1)
    # pad so chunks are equal (non-divisible sizes are a required case,
    # SURVEY.md §4 test list)
    n = flat.numel()
    chunk = (n + size - 1) // size
    padded = flat
    if chunk * size != n:
        padded = torch.zeros(chunk * size, dtype=flat.dtype,
                             device=flat.device)
        padded[:n] = flat
    chunks = list(padded.chunk(size))
    left = ((rank - 1) + size) % size
    right = (rank + 1) % size
    tmp = torch.empty_like(chunks[0])

    # reduce-scatter: after step i, rank owns the running sum of chunk
    # (rank - i) mod size; after size-1 steps rank r holds the full sum
    # of chunk (r+1) mod size.
    for i in range(size - 1):
        send_idx = (rank - i) % size
        recv_idx = (rank - i - 1) % size
        dist.sendrecv(chunks[send_idx], right, tmp, left, g)
        chunks[recv_idx] += tmp

    owned = (rank + 1) % size
    if average:
        chunks[owned] /= size

    # all-gather: circulate the reduced chunks around the ring.
    for i in range(size - 1):
        send_idx = (owned - i) % size
        recv_idx = (owned - i - 1) % size
        dist.sendrecv(chunks[send_idx], right, chunks[recv_idx], left, g)

    if padded.data_ptr() != flat.data_ptr():
        flat.copy_(padded[:n])
    return tensor


def gather_to_root(tensor: torch.Tensor, rank: int, tensor_list=None,
                   root: int = 0, g=None):
    """Root-aware gather convenience (the unused helper at ptp.py:9-19):
    the root passes ``tensor_list`` and receives everyone's tensor;
    non-roots just send."""
    if rank == root:
        assert tensor_list is not None, \
            "root must pass a tensor_list of world_size tensors"
        dist.gather(tensor, gather_list=tensor_list, dst=root, g=g)
    else:
        dist.gather(tensor, gather_list=None, dst=root, g=g)
