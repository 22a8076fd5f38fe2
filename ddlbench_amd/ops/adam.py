"""FusedAdam / FusedAdamW — multi-tensor Adam step as one HIP kernel.

torch.optim.Adam/AdamW semantics (the reference's GNMT optimizer and
AdamWithWeightStashing, pipedream-fork/runtime/adam.py); fp32 m/v state
for bf16 params."""

from __future__ import annotations

import torch
from torch.optim import Optimizer

from ddlbench_amd import ops as _ops
from ddlbench_amd.ops.sgd import _same_dense_layout


class FusedAdam(Optimizer):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0,
                 decoupled_wd: bool = False, backend: str = "auto"):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.backend = backend
        self.decoupled_wd = decoupled_wd
        self._cache = {}

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for gi, group in enumerate(self.param_groups):
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            if params[0].is_cuda and _ops.use_native(params[0].device,
                                                     self.backend):
                self._step_native(gi, group, params)
            else:
                self._step_torch(group, params)
        return loss

    def _state(self, p, dev):
        st = self.state[p]
        if "step" not in st:
            st["step"] = 0
            st["exp_avg"] = torch.zeros(p.shape, dtype=torch.float32,
                                        device=dev)
            st["exp_avg_sq"] = torch.zeros(p.shape, dtype=torch.float32,
                                           device=dev)
        return st

    def _step_torch(self, group, params):
        b1, b2 = group["betas"]
        lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]
        for p in params:
            st = self._state(p, p.device)
            st["step"] += 1
            t = st["step"]
            g = p.grad.float()
            if wd and not self.decoupled_wd:
                g = g.add(p.float(), alpha=wd)
            st["exp_avg"].mul_(b1).add_(g, alpha=1 - b1)
            st["exp_avg_sq"].mul_(b2).addcmul_(g, g, value=1 - b2)
            bc1 = 1 - b1 ** t
            bc2 = 1 - b2 ** t
            upd = (st["exp_avg"] / bc1) / ((st["exp_avg_sq"] / bc2).sqrt()
                                           + eps)
            if wd and self.decoupled_wd:
                upd = upd.add(p.float(), alpha=wd)
            p.data.add_(upd.to(p.dtype), alpha=-lr)

    def _step_native(self, gi, group, params):
        ext = _ops.require_extension()
        b1, b2 = group["betas"]
        lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]
        by_dtype = {}
        for p in params:
            by_dtype.setdefault(p.dtype, []).append(p)
        for dtype, ps in by_dtype.items():
            if dtype not in (torch.float32, torch.bfloat16):
                self._step_torch(group, ps)
                continue
            dev = ps[0].device
            for p in ps:
                self._state(p, dev)
                self.state[p]["step"] += 1
            t = self.state[ps[0]]["step"]
            key = (gi, dtype)
            cached = self._cache.get(key)
            if cached is None or len(cached["params"]) != len(ps):
                prefix = torch.zeros(len(ps), dtype=torch.int64)
                total = 0
                for i, p in enumerate(ps):
                    prefix[i] = total
                    total += p.numel()
                cached = {
                    "params": ps, "total": total, "prefix": prefix.to(dev),
                    "ptr_params": torch.tensor(
                        [p.data_ptr() for p in ps],
                        dtype=torch.int64).to(dev),
                    "ptr_ms": torch.tensor(
                        [self.state[p]["exp_avg"].data_ptr() for p in ps],
                        dtype=torch.int64).to(dev),
                    "ptr_vs": torch.tensor(
                        [self.state[p]["exp_avg_sq"].data_ptr()
                         for p in ps], dtype=torch.int64).to(dev),
                }
                self._cache[key] = cached
            for p in ps:
                if not _same_dense_layout(p, p.grad):
                    p.grad = p.grad.contiguous() if p.is_contiguous() \
                        else p.grad.contiguous(
                            memory_format=torch.channels_last)
                assert _same_dense_layout(p, p.grad)
            ptr_grads = torch.tensor([p.grad.data_ptr() for p in ps],
                                     dtype=torch.int64).to(
                                         dev, non_blocking=True)
            ext.fused_adam(cached["ptr_params"], ptr_grads,
                           cached["ptr_ms"], cached["ptr_vs"],
                           cached["prefix"], cached["total"], lr, b1, b2,
                           eps, wd, 1 - b1 ** t, 1 - b2 ** t,
                           self.decoupled_wd, dtype == torch.bfloat16)


class FusedAdamW(FusedAdam):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2, backend="auto"):
        super().__init__(params, lr, betas, eps, weight_decay,
                         decoupled_wd=True, backend=backend)
