"""Synthetic data: shapes, determinism, DP sharding."""

import torch

from ddlbench_amd.config import BenchConfig
from ddlbench_amd.data import SyntheticImageDataset, make_loaders, synthetic_batch


def test_shapes_match_reference_table():
    for ds, shape in [("mnist", (1, 28, 28)), ("cifar10", (3, 32, 32)),
                      ("imagenet", (3, 224, 224)), ("highres", (3, 512, 512))]:
        d = SyntheticImageDataset(ds, size=4)
        x, y = d[0]
        assert tuple(x.shape) == shape
        assert 0 <= y < d.num_classes


def test_determinism():
    a = SyntheticImageDataset("cifar10", size=8, seed=7)
    b = SyntheticImageDataset("cifar10", size=8, seed=7)
    xa, ya = a[3]
    xb, yb = b[3]
    assert torch.equal(xa, xb) and ya == yb
    c = SyntheticImageDataset("cifar10", size=8, seed=8)
    xc, _ = c[3]
    assert not torch.equal(xa, xc)


def test_train_test_streams_differ():
    tr = SyntheticImageDataset("mnist", train=True, size=4, seed=1)
    te = SyntheticImageDataset("mnist", train=False, size=4, seed=1)
    assert not torch.equal(tr[0][0], te[0][0])


def test_loader_sharding_disjoint():
    cfg = BenchConfig(dataset="mnist", batch_size=4, synthetic_scale=0.001,
                      num_workers=0)
    seen = []
    for rank in range(2):
        loader, _, sampler = make_loaders(cfg, world_size=2, rank=rank,
                                          pin_memory=False)
        sampler.set_epoch(0)
        idx = list(iter(sampler))
        seen.append(set(idx))
    assert seen[0].isdisjoint(seen[1])


def test_synthetic_batch():
    cfg = BenchConfig(dataset="imagenet")
    x, y = synthetic_batch(cfg, batch_size=2)
    assert x.shape == (2, 3, 224, 224) and y.shape == (2,)
