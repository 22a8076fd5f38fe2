"""Metric helpers: running averages, top-k accuracy, device memory stats."""

from __future__ import annotations

import torch


class AverageMeter:
    def __init__(self) -> None:
        self.reset()

    def reset(self) -> None:
        self.val = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val: float, n: int = 1) -> None:
        self.val = float(val)
        self.sum += float(val) * n
        self.count += n

    @property
    def avg(self) -> float:
        return self.sum / max(self.count, 1)


@torch.no_grad()
def accuracy(output: torch.Tensor, target: torch.Tensor, topk=(1,)):
    """Top-k accuracy fractions (the reference reports top-1 —
    /root/reference/benchmark/mnist/mnist_pytorch.py:102-133)."""
    maxk = max(topk)
    _, pred = output.topk(maxk, dim=1, largest=True, sorted=True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    res = []
    for k in topk:
        res.append(correct[:k].reshape(-1).float().sum().item() / target.size(0))
    return res


def gpu_memory_gb(device=None):
    """(allocated_peak, reserved_peak, total) in GB; zeros on CPU.

    Mirrors the reference's use of torch.cuda.memory_stats
    (mnist_pytorch.py:72-83)."""
    if not torch.cuda.is_available():
        return 0.0, 0.0, 0.0
    stats = torch.cuda.memory_stats(device)
    alloc = stats.get("allocated_bytes.all.peak", 0) / 2**30
    reserved = stats.get("reserved_bytes.all.peak", 0) / 2**30
    total = torch.cuda.get_device_properties(device or 0).total_memory / 2**30
    return alloc, reserved, total
