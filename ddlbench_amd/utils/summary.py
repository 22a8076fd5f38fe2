"""Model summary: per-layer output shapes and parameter counts.

Our torchsummary equivalent (the reference vendors torchsummary —
/root/reference/pipedream-fork/profiler/torchmodules/torchsummary/
torchsummary.py:33-60 — and benchmark/network_summary.py prints it for
every model x dataset). Forward hooks on leaf modules record output
shape + param count."""

from __future__ import annotations

from typing import List

import torch
import torch.nn as nn


def summarize(model: nn.Module, sample: torch.Tensor) -> List[dict]:
    rows: List[dict] = []
    hooks = []

    def hook(module, inputs, output):
        out = output[0] if isinstance(output, (tuple, list)) else output
        rows.append({
            "name": type(module).__name__,
            "output_shape": tuple(out.shape) if torch.is_tensor(out) else None,
            "params": sum(p.numel() for p in module.parameters(
                recurse=False)),
            "trainable": any(p.requires_grad for p in module.parameters(
                recurse=False)),
        })

    for m in model.modules():
        if len(list(m.children())) == 0:  # leaf
            hooks.append(m.register_forward_hook(hook))
    was_training = model.training
    model.eval()
    try:
        with torch.no_grad():
            model(sample)
    finally:
        for h in hooks:
            h.remove()
        model.train(was_training)
    return rows


def format_summary(rows: List[dict]) -> str:
    lines = [f"{'Layer':<28}{'Output shape':<28}{'Params':>12}"]
    lines.append("-" * 68)
    total = 0
    for r in rows:
        total += r["params"]
        lines.append(f"{r['name']:<28}{str(r['output_shape']):<28}"
                     f"{r['params']:>12,}")
    lines.append("-" * 68)
    lines.append(f"{'Total params':<56}{total:>12,}")
    return "\n".join(lines)
