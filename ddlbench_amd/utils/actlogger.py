"""Activation / gradient capture + sparsity measurement.

Parity with the reference's analysis utilities (SURVEY.md §2.8/§2.12:
torchlogger/activation_gradient_logger.py pickles activations/gradients
every N epochs; utils/sparsity/measure_sparsity.py reports zero
fractions and compressibility)."""

from __future__ import annotations

import os
import pickle
from typing import Dict, Optional

import torch
import torch.nn as nn


class ActivationGradientLogger:
    """Capture leaf-module activations and gradients to disk."""

    def __init__(self, model: nn.Module, out_dir: str,
                 every_n_epochs: int = 1):
        self.model = model
        self.out_dir = out_dir
        self.every = max(1, every_n_epochs)
        self._hooks = []
        self._acts: Dict[str, torch.Tensor] = {}
        self._grads: Dict[str, torch.Tensor] = {}
        self.enabled = False

    def _attach(self) -> None:
        for name, m in self.model.named_modules():
            if len(list(m.children())) > 0:
                continue

            def fwd_hook(mod, inp, out, name=name):
                if self.enabled and torch.is_tensor(out):
                    self._acts[name] = out.detach().cpu()

            def bwd_hook(mod, gin, gout, name=name):
                if self.enabled and gout and torch.is_tensor(gout[0]):
                    self._grads[name] = gout[0].detach().cpu()

            self._hooks.append(m.register_forward_hook(fwd_hook))
            self._hooks.append(m.register_full_backward_hook(bwd_hook))

    def start(self) -> None:
        if not self._hooks:
            self._attach()
        self.enabled = True

    def stop(self) -> None:
        self.enabled = False

    def dump(self, epoch: int) -> Optional[str]:
        if epoch % self.every:
            return None
        os.makedirs(self.out_dir, exist_ok=True)
        path = os.path.join(self.out_dir, f"acts_grads_epoch{epoch}.pkl")
        with open(path, "wb") as f:
            pickle.dump({"activations": self._acts,
                         "gradients": self._grads}, f)
        self._acts.clear()
        self._grads.clear()
        return path

    def close(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()


def measure_sparsity(t: torch.Tensor, threshold: float = 0.0) -> dict:
    """Zero/near-zero fraction and a simple compressibility estimate
    (reference measure_sparsity.py:10-30)."""
    flat = t.detach().float().flatten()
    n = flat.numel()
    zeros = int((flat.abs() <= threshold).sum())
    frac = zeros / max(n, 1)
    # dense vs COO-style (index+value) size ratio
    nnz = n - zeros
    compressed = nnz * (4 + 4)
    dense = n * 4
    return {"numel": n, "zeros": zeros, "sparsity": frac,
            "compression_ratio": dense / max(compressed, 1)}
