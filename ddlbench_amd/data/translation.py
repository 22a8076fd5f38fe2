"""Synthetic translation data: variable-length token pairs, bucketing
sampler, padded collate.

Parity with the reference's WMT pipeline
(/root/reference/pipedream-fork/runtime/translation/seq2seq/data/ —
BucketingSampler buckets by sequence length with a seeded shuffle,
StaticDistributedSampler shards; sampler.py:12-133). There is no
network, so sentences are deterministic random token streams whose
"translation" is a seeded function of the source (length-correlated),
which is all a throughput benchmark needs."""

from __future__ import annotations

from typing import Iterator, List

import torch
from torch.utils.data import Dataset, Sampler

from ddlbench_amd.models.gnmt import BOS, EOS, PAD


class SyntheticTranslationDataset(Dataset):
    def __init__(self, size: int = 10000, vocab_size: int = 32320,
                 min_len: int = 4, max_len: int = 50, seed: int = 42):
        self.size = size
        self.vocab_size = vocab_size
        self.min_len = min_len
        self.max_len = max_len
        self.seed = seed

    def __len__(self) -> int:
        return self.size

    def src_len(self, idx: int) -> int:
        g = torch.Generator().manual_seed(self.seed * 131071 + idx)
        return int(torch.randint(self.min_len, self.max_len + 1, (1,),
                                 generator=g))

    def __getitem__(self, idx: int):
        g = torch.Generator().manual_seed(self.seed * 131071 + idx)
        n = int(torch.randint(self.min_len, self.max_len + 1, (1,),
                              generator=g))
        m = max(self.min_len,
                min(self.max_len,
                    n + int(torch.randint(-3, 4, (1,), generator=g))))
        src = torch.randint(3, self.vocab_size, (n,), generator=g)
        tgt_body = torch.randint(3, self.vocab_size, (m,), generator=g)
        tgt = torch.cat([torch.tensor([BOS]), tgt_body,
                         torch.tensor([EOS])])
        return src, tgt


class BucketingSampler(Sampler):
    """Groups indices of similar source length into batches; seeded
    shuffle of buckets and batches per epoch; optional rank sharding."""

    def __init__(self, dataset: SyntheticTranslationDataset,
                 batch_size: int, world_size: int = 1, rank: int = 0,
                 seed: int = 0, num_buckets: int = 8):
        self.ds = dataset
        self.batch_size = batch_size
        self.world_size = world_size
        self.rank = rank
        self.seed = seed
        self.num_buckets = num_buckets
        self.epoch = 0
        self._lengths = [dataset.src_len(i) for i in range(len(dataset))]

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def _batches(self) -> List[List[int]]:
        g = torch.Generator().manual_seed(self.seed * 7919 + self.epoch)
        order = torch.argsort(torch.tensor(self._lengths, dtype=torch.float)
                              + torch.rand(len(self._lengths), generator=g))
        batches = [order[i:i + self.batch_size].tolist()
                   for i in range(0, len(order), self.batch_size)]
        perm = torch.randperm(len(batches), generator=g)
        batches = [batches[i] for i in perm]
        # shard whole batches across ranks, dropping the ragged tail
        n = (len(batches) // self.world_size) * self.world_size
        return batches[self.rank:n:self.world_size]

    def __iter__(self) -> Iterator[List[int]]:
        return iter(self._batches())

    def __len__(self) -> int:
        return len(self._batches())


def collate_translation(batch):
    """Pad to (T, B); returns src, src_len, tgt_in, tgt_out."""
    srcs, tgts = zip(*batch)
    B = len(srcs)
    src_len = torch.tensor([len(s) for s in srcs], dtype=torch.long)
    Ts = int(src_len.max())
    Tt = max(len(t) for t in tgts)
    src = torch.full((Ts, B), PAD, dtype=torch.long)
    tgt = torch.full((Tt, B), PAD, dtype=torch.long)
    for b, (s, t) in enumerate(zip(srcs, tgts)):
        src[:len(s), b] = s
        tgt[:len(t), b] = t
    return src, src_len, tgt[:-1], tgt[1:]


class StaticDistributedSampler(Sampler):
    """Deterministic contiguous rank shards, no shuffle — the
    reference's eval-time sampler (seq2seq/data/sampler.py
    StaticDistributedSampler): rank r takes indices [r*per, (r+1)*per)
    of the dataset padded up to world*ceil(n/world), padding indices
    dropped. Yields batches like BucketingSampler."""

    def __init__(self, dataset, batch_size: int, world_size: int = 1,
                 rank: int = 0):
        self.n = len(dataset)
        self.batch_size = batch_size
        per = (self.n + world_size - 1) // world_size
        lo = rank * per
        hi = min(lo + per, self.n)
        self.indices = list(range(lo, hi))

    def __len__(self) -> int:
        return (len(self.indices) + self.batch_size - 1) \
            // self.batch_size

    def __iter__(self):
        for i in range(0, len(self.indices), self.batch_size):
            yield self.indices[i:i + self.batch_size]
