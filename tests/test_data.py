"""MNIST loader + DistributedSampler semantics (reference utils/train.py:24-30)."""
import torch

from ddp_tricks_amd.utils.data import MNIST, DistributedSampler


def test_synthetic_mnist_shapes():
    ds = MNIST(root="/nonexistent", train=True, num_samples=128)
    assert ds.synthetic
    assert len(ds) == 128
    img, lbl = ds[0]
    assert img.shape == (1, 28, 28) and img.dtype == torch.float32
    assert 0.0 <= img.min() and img.max() <= 1.0
    assert 0 <= lbl < 10


def test_sampler_partitions_cover_dataset():
    ds = MNIST(root="/nonexistent", train=True, num_samples=100)
    samplers = [DistributedSampler(ds, num_replicas=4, rank=r, shuffle=True)
                for r in range(4)]
    idxs = [list(iter(s)) for s in samplers]
    assert all(len(ix) == 25 for ix in idxs)
    union = set().union(*[set(ix) for ix in idxs])
    assert union == set(range(100))


def test_sampler_set_epoch_changes_order():
    ds = MNIST(root="/nonexistent", train=True, num_samples=64)
    s = DistributedSampler(ds, num_replicas=2, rank=0, shuffle=True)
    s.set_epoch(0)
    a = list(iter(s))
    s.set_epoch(1)
    b = list(iter(s))
    assert a != b
    s.set_epoch(0)
    assert list(iter(s)) == a  # seeded determinism


def test_sampler_pads_to_divisible():
    ds = MNIST(root="/nonexistent", train=True, num_samples=10)
    samplers = [DistributedSampler(ds, num_replicas=3, rank=r) for r in range(3)]
    lens = [len(list(iter(s))) for s in samplers]
    assert lens == [4, 4, 4]


def test_matches_torch_distributed_sampler():
    ds = MNIST(root="/nonexistent", train=True, num_samples=50)
    for epoch in (0, 3):
        ours = DistributedSampler(ds, num_replicas=2, rank=1, shuffle=True)
        theirs = torch.utils.data.distributed.DistributedSampler(
            ds, num_replicas=2, rank=1, shuffle=True)
        ours.set_epoch(epoch)
        theirs.set_epoch(epoch)
        assert list(iter(ours)) == list(iter(theirs))
