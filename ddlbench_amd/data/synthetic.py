"""Synthetic datasets, tensor-native.

The reference generates random JPEG ImageFolder trees once per node
(/root/reference/benchmark/generate_synthetic_data.py) and then decodes
them through torchvision at train time. On MI355X the decode pipeline is
pure overhead for a throughput benchmark — the idiomatic equivalent is a
deterministic on-the-fly tensor dataset with the same shapes and sizes
(generate_synthetic_data.py:76-107), so the data path never bottlenecks
288-GB-HBM GPUs and any machine can run without a dataset download.

Determinism: sample i of a dataset with seed s is always the same tensor,
generated from torch.Generator(seed=s*1e6+i) — so DistributedSampler-style
sharding gives every rank a disjoint, reproducible stream.
"""

from __future__ import annotations

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler

from ddlbench_amd.config import BenchConfig, DATASET_SHAPES


class SyntheticImageDataset(Dataset):
    """Random-normal images + uniform labels with reference shapes."""

    def __init__(self, dataset: str, train: bool = True, seed: int = 42,
                 scale: float = 1.0, size: int = 0):
        c, h, w, ncls, ntrain, ntest = DATASET_SHAPES[dataset]
        self.shape = (c, h, w)
        self.num_classes = ncls
        if size > 0:
            self.size = size
        else:
            base = ntrain if train else ntest
            self.size = max(1, int(base * scale))
        self.seed = seed + (0 if train else 1_000_003)

    def __len__(self) -> int:
        return self.size

    def __getitem__(self, idx: int):
        g = torch.Generator().manual_seed(self.seed * 1_000_000 + idx)
        x = torch.randn(self.shape, generator=g)
        y = torch.randint(self.num_classes, (1,), generator=g).item()
        return x, y


def synthetic_batch(cfg: BenchConfig, batch_size: int = 0, device="cpu",
                    dtype=torch.float32, seed: int = 0):
    """One random batch of the config's shape — for profiling/benching."""
    bs = batch_size or cfg.batch_size
    c, h, w = cfg.shape
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(bs, c, h, w, generator=g).to(device=device, dtype=dtype)
    y = torch.randint(cfg.num_classes, (bs,), generator=g).to(device)
    return x, y


def make_loaders(cfg: BenchConfig, world_size: int = 1, rank: int = 0,
                 pin_memory: bool = True):
    """(train_loader, test_loader[, samplers]) honouring DP sharding.

    Mirrors the reference's DistributedSampler usage for the horovod path
    (/root/reference/benchmark/mnist/mnist_horovod.py:209-219). Real-data
    mode (cfg.synthetic False + DATADIR, reference `run.sh -s`) loads a
    class-per-directory tree instead (data/real.py)."""
    if not cfg.synthetic and cfg.data_dir:
        from ddlbench_amd.data.real import RealImageDataset
        train_ds = RealImageDataset(cfg.dataset, cfg.data_dir, train=True)
        test_ds = RealImageDataset(cfg.dataset, cfg.data_dir, train=False)
    else:
        train_ds = SyntheticImageDataset(cfg.dataset, train=True,
                                         seed=cfg.seed,
                                         scale=cfg.synthetic_scale)
        test_ds = SyntheticImageDataset(cfg.dataset, train=False,
                                        seed=cfg.seed,
                                        scale=cfg.synthetic_scale)
    train_sampler = test_sampler = None
    if world_size > 1:
        train_sampler = DistributedSampler(train_ds, num_replicas=world_size,
                                           rank=rank, shuffle=True,
                                           seed=cfg.seed, drop_last=True)
        test_sampler = DistributedSampler(test_ds, num_replicas=world_size,
                                          rank=rank, shuffle=False)
    train_loader = DataLoader(
        train_ds, batch_size=cfg.batch_size,
        shuffle=(train_sampler is None), sampler=train_sampler,
        num_workers=cfg.num_workers, pin_memory=pin_memory, drop_last=True,
        persistent_workers=cfg.num_workers > 0)
    test_loader = DataLoader(
        test_ds, batch_size=cfg.batch_size, shuffle=False,
        sampler=test_sampler, num_workers=cfg.num_workers,
        pin_memory=pin_memory,
        persistent_workers=cfg.num_workers > 0)
    return train_loader, test_loader, train_sampler
