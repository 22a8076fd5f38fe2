"""Stage balancing: profile per-module time, partition to minimize the
slowest stage.

MI355X-native replacement for torchgpipe's balance_by_time
(/root/reference/benchmark/mnist/mnist_gpipe.py:215-217): times each
module of a flattened nn.Sequential with device-synchronized wall clocks,
then solves the contiguous linear-partition problem exactly (DP) instead
of torchgpipe's heuristic."""

from __future__ import annotations

import time
from typing import List, Sequence

import torch
import torch.nn as nn


@torch.no_grad()
def profile_module_times(seq: nn.Sequential, sample: torch.Tensor,
                         device=None, iters: int = 4,
                         warmup: int = 1) -> List[float]:
    """Per-module forward time (seconds). Backward cost is roughly
    proportional (~2x) for conv stacks, so forward time is a valid
    balance weight."""
    device = device or sample.device
    times = [0.0] * len(seq)
    x = sample.to(device)
    modules = [m.to(device) for m in seq]
    for it in range(warmup + iters):
        x = sample.to(device)
        for i, m in enumerate(modules):
            if device.type == "cuda":
                torch.cuda.synchronize(device)
            t0 = time.perf_counter()
            x = m(x)
            if device.type == "cuda":
                torch.cuda.synchronize(device)
            if it >= warmup:
                times[i] += time.perf_counter() - t0
    return [t / iters for t in times]


@torch.no_grad()
def profile_unit_times(units, sample_args, iters: int = 3,
                       warmup: int = 1) -> List[float]:
    """Per-unit forward time for a tuple-I/O unit chain (the GNMT
    pipeline units). CPU/GPU device follows the units' parameters."""
    times = [0.0] * len(units)
    dev = None
    for u in units:
        for prm in u.parameters():
            dev = prm.device
            break
        if dev is not None:
            break
    dev = dev or torch.device("cpu")
    for it in range(warmup + iters):
        xs = tuple(t.to(dev) for t in sample_args)
        for i, u in enumerate(units):
            if dev.type == "cuda":
                torch.cuda.synchronize(dev)
            t0 = time.perf_counter()
            out = u(*xs)
            if dev.type == "cuda":
                torch.cuda.synchronize(dev)
            if it >= warmup:
                times[i] += time.perf_counter() - t0
            xs = (out,) if torch.is_tensor(out) else tuple(out)
    return [t / iters for t in times]


def partition_minmax(weights: Sequence[float], k: int) -> List[int]:
    """Split weights into k contiguous groups minimizing the max group
    sum. Returns group sizes (len k, sums to len(weights)). Exact DP."""
    n = len(weights)
    k = min(k, n)
    prefix = [0.0]
    for w in weights:
        prefix.append(prefix[-1] + w)

    INF = float("inf")
    # dp[j][i] = min over partitions of first i items into j groups of max sum
    dp = [[INF] * (n + 1) for _ in range(k + 1)]
    cut = [[0] * (n + 1) for _ in range(k + 1)]
    dp[0][0] = 0.0
    for j in range(1, k + 1):
        for i in range(j, n + 1):
            for t in range(j - 1, i):
                cost = max(dp[j - 1][t], prefix[i] - prefix[t])
                if cost < dp[j][i]:
                    dp[j][i] = cost
                    cut[j][i] = t
    # backtrack
    sizes = []
    i = n
    for j in range(k, 0, -1):
        t = cut[j][i]
        sizes.append(i - t)
        i = t
    sizes.reverse()
    return sizes


def balance_by_time(n_partitions: int, seq: nn.Sequential,
                    sample: torch.Tensor, device=None) -> List[int]:
    times = profile_module_times(seq, sample, device)
    return partition_minmax(times, n_partitions)
