"""LR schedules matching the reference's three trajectories.

* ``step30``   — imagenet step decay: lr × 0.1 every 30 epochs
  (/root/reference/benchmark/imagenet/imagenet_pytorch.py:225-229).
* ``warmup``   — horovod gradual warmup: over the first ``warmup_epochs``
  the per-rank LR ramps linearly from lr₀ to lr₀·world_size, then steps
  0.1×/0.01×/0.001× at epochs 30/60/80
  (imagenet_horovod.py:258-275; ramp evaluated per batch).
* ``constant`` — no adjustment (the reference mnist/cifar10 scripts).

A schedule is a callable ``f(epoch, frac) -> multiplier`` applied to the
OPTIMIZER'S configured base LR (for DDP that base is already lr₀·N, so
the warmup multiplier starts at 1/N). ``epoch`` is 1-based; ``frac`` in
[0,1) is the position inside the epoch (batch_idx / n_batches).
"""

from __future__ import annotations

from typing import Callable

Schedule = Callable[[int, float], float]

SCHEDULES = ("constant", "step30", "warmup")


def make_lr_schedule(name: str, world_size: int = 1,
                     warmup_epochs: int = 5) -> Schedule:
    if name in ("", "constant", None):
        return lambda epoch, frac: 1.0
    if name == "step30":
        # reference: lr * 0.1^(epoch // 30) with 0-based epochs
        return lambda epoch, frac: 0.1 ** ((epoch - 1) // 30)
    if name == "warmup":
        n = max(world_size, 1)
        w = max(warmup_epochs, 1)

        def sched(epoch: int, frac: float) -> float:
            e0 = (epoch - 1) + frac  # 0-based continuous epoch
            if e0 < w:
                # ramps lr_base/N .. lr_base (reference lr_adj:
                # 1/size * (epoch*(size-1)/warmup + 1))
                return (e0 * (n - 1) / w + 1.0) / n
            if e0 < 30:
                return 1.0
            if e0 < 60:
                return 1e-1
            if e0 < 80:
                return 1e-2
            return 1e-3

        return sched
    raise ValueError(f"unknown lr schedule {name!r}")


def apply_lr(optimizer, base_lrs, factor: float) -> float:
    """Set every param group's lr to base*factor; returns the factor."""
    for group, base in zip(optimizer.param_groups, base_lrs):
        group["lr"] = base * factor
    return factor
