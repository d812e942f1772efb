"""Data pipeline tests."""

import torch

from dist_mnist_amd.data import DataPartitioner, SyntheticMNIST, partition_dataset


def test_synthetic_deterministic():
    d1 = SyntheticMNIST(n=100, seed=1234)
    d2 = SyntheticMNIST(n=100, seed=1234)
    x1, y1 = d1[7]
    x2, y2 = d2[7]
    assert torch.equal(x1, x2) and y1 == y2
    x3, _ = d1[8]
    assert not torch.equal(x1, x3)
    assert x1.shape == (1, 28, 28) and x1.dtype == torch.float32
    assert 0 <= y1 < 10


def test_partitioner_disjoint_and_deterministic():
    data = SyntheticMNIST(n=1000, seed=1234)
    p1 = DataPartitioner(data, [0.5, 0.5], seed=1234)
    p2 = DataPartitioner(data, [0.5, 0.5], seed=1234)
    assert p1.partitions == p2.partitions  # same seed -> same shuffle
    s0, s1 = set(p1.partitions[0]), set(p1.partitions[1])
    assert not (s0 & s1)
    assert len(s0) == len(s1) == 500


def test_global_batch_math():
    for world in (1, 2, 4, 8):
        loader, bsz = partition_dataset(rank=0, world_size=world, global_batch=128, n=1024)
        assert bsz == 128 // world  # reference R4
        x, y = next(iter(loader))
        assert x.shape == (bsz, 1, 28, 28)
        assert y.shape == (bsz,)


def test_shards_cover_rankwise():
    n = 512
    seen = set()
    for rank in range(4):
        data = SyntheticMNIST(n=n, seed=1234)
        part = DataPartitioner(data, [0.25] * 4, seed=1234).use(rank)
        seen |= set(part.index)
    assert len(seen) == n


def test_prefetcher_cpu_passthrough():
    from dist_mnist_amd.data import SyntheticBatchStream
    from dist_mnist_amd.runtime import Prefetcher

    stream = SyntheticBatchStream(batch_size=4, n_pool=2, seed=5)
    pf = Prefetcher(stream, device="cpu")
    x, y = next(pf)
    assert x.shape == (4, 1, 28, 28)
    assert y.shape == (4,)


def test_synthetic_stream_rgb224_layout():
    """ResNet bench input (BASELINE config 5): NHWC bf16 [N,224,224,8] with
    RGB in channels 0-2 and the 5 pad channels exactly zero."""
    from dist_mnist_amd.data import SyntheticBatchStream

    s1 = SyntheticBatchStream(batch_size=4, n_pool=2, seed=7, pin=False,
                              layout="rgb224_bf16", n_classes=1000)
    s2 = SyntheticBatchStream(batch_size=4, n_pool=2, seed=7, pin=False,
                              layout="rgb224_bf16", n_classes=1000)
    x1, y1 = next(iter(s1))
    x2, y2 = next(iter(s2))
    assert x1.shape == (4, 224, 224, 8) and x1.dtype == torch.bfloat16
    assert torch.equal(x1, x2) and torch.equal(y1, y2)
    assert torch.all(x1[..., 3:] == 0)
    assert x1[..., :3].float().abs().sum() > 0
    assert y1.dtype == torch.int64 and y1.max() < 1000 and y1.min() >= 0
