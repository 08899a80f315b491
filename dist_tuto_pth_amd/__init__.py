"""dist_tuto_pth_amd — an MI355X-native distributed-training primer library.

A from-scratch rebuild of the capabilities of seba-1511/dist_tuto.pth
("Writing Distributed Applications with PyTorch", tuto.md) designed for
AMD Instinct MI355X (gfx950, CDNA4):

  * ``dist``       — the torch.distributed-style API surface the tutorial
                     consumes (reference: tuto.md:77-202), implemented on a
                     native C++ RCCL backend over xGMI (one rank per GPU)
                     plus a gloo CPU path for plumbing/tests
                     (reference backends: tuto.md:363-419).
  * ``algorithms`` — the hand-rolled collectives the tutorial builds at L3
                     (ring all-reduce: allreduce.py:8-34, corrected per the
                     exercise at tuto.md:354; gather helper: ptp.py:9-19).
  * ``parallel``   — data partitioning (train_dist.py:17-50) and the
                     synchronous-SGD data-parallel layer
                     (average_gradients: train_dist.py:94-100), plus a
                     bucketed, backward-overlapped DDP the tutorial points
                     at (tuto.md:216,320).
  * ``models``     — the tutorial's ConvNet (Net: train_dist.py:53-71) and
                     larger models for the benchmark configs.
  * ``ops``        — hand-written CDNA4 HIP kernels (MFMA/LDS) for every
                     op on the training path.

No CUDA shims, no hipify, no Triton: the GPU path is HIP for gfx950 and
RCCL over xGMI only.
"""

__version__ = "0.1.0"

from . import dist  # noqa: F401

__all__ = ["dist", "__version__"]
