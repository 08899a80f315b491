"""Data-partitioning layer tests (train_dist.py:17-50 semantics)."""

import torch

from dist_tuto_pth_amd.parallel import (DataPartitioner, Partition,
                                        SyntheticMNIST)


def test_partition_view():
    data = list(range(100))
    p = Partition(data, [5, 7, 9])
    assert len(p) == 3
    assert p[0] == 5 and p[2] == 9


def test_partitioner_fractions_and_disjoint():
    data = list(range(1000))
    dp = DataPartitioner(data, sizes=[0.7, 0.2, 0.1], seed=1234)
    parts = [dp.use(i) for i in range(3)]
    assert [len(p) for p in parts] == [700, 200, 100]
    seen = set()
    for p in parts:
        for i in p.index:
            assert i not in seen
            seen.add(i)


def test_partitioner_same_seed_same_shuffle():
    data = list(range(256))
    a = DataPartitioner(data, sizes=[0.5, 0.5], seed=1234)
    b = DataPartitioner(data, sizes=[0.5, 0.5], seed=1234)
    assert a.partitions == b.partitions
    c = DataPartitioner(data, sizes=[0.5, 0.5], seed=4321)
    assert a.partitions != c.partitions


def test_synthetic_mnist_shape_and_determinism():
    d1 = SyntheticMNIST(n=64, seed=7)
    d2 = SyntheticMNIST(n=64, seed=7)
    x, y = d1[3]
    assert x.shape == (1, 28, 28)
    assert 0 <= int(y) < 10
    assert torch.equal(d1.images, d2.images)
    assert torch.equal(d1.labels, d2.labels)
