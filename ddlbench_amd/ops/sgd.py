"""FusedSGD — SGD+momentum+weight-decay as one multi-tensor HIP kernel.

torch.optim.SGD semantics (the reference's optimizer —
/root/reference/benchmark/mnist/mnist_pytorch.py:44-50), but the whole
model updates in one kernel launch per dtype bucket instead of a Python
loop over ~160 tensors. Momentum state is fp32 even for bf16 params."""

from __future__ import annotations

import torch
from torch.optim import Optimizer

from ddlbench_amd import ops as _ops


def _same_dense_layout(p, g):
    """True when grad and param walk memory in the same element order.
    Stride values on size-1 dims are arbitrary (a (K,C,1,1) weight is
    simultaneously 'contiguous' and 'channels_last'), so compare strides
    only where the dim extent is > 1."""
    if p.shape != g.shape:
        return False
    return all(sz <= 1 or ps == gs
               for sz, ps, gs in zip(p.shape, p.stride(), g.stride()))



class FusedSGD(Optimizer):
    def __init__(self, params, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0, backend: str = "auto"):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.backend = backend
        self._cache = {}  # per (group_idx, dtype): static ptr tensors

    def _native_ok(self, device: torch.device) -> bool:
        return _ops.use_native(device, self.backend)

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
            if params[0].is_cuda and self._native_ok(params[0].device):
                self._step_native(gi, group, params)
            else:
                self._step_torch(group, params)
        return loss

    # ---- reference (CPU / backend="torch") path -----------------------
    def _step_torch(self, group, params):
        lr, mu, wd = group["lr"], group["momentum"], group["weight_decay"]
        for p in params:
            d = p.grad.float()
            if wd:
                d = d.add(p.float(), alpha=wd)
            if mu:
                st = self.state[p]
                if "momentum_buffer" not in st:
                    st["momentum_buffer"] = d.clone()
                else:
                    st["momentum_buffer"].mul_(mu).add_(d)
                d = st["momentum_buffer"]
            p.data.add_(d.to(p.dtype), alpha=-lr)

    # ---- fused path ----------------------------------------------------
    def _step_native(self, gi, group, params):
        ext = _ops.require_extension()
        lr, mu, wd = group["lr"], group["momentum"], group["weight_decay"]
        by_dtype = {}
        for p in params:
            by_dtype.setdefault(p.dtype, []).append(p)
        for dtype, ps in by_dtype.items():
            if dtype not in (torch.float32, torch.bfloat16):
                self._step_torch(group, ps)
                continue
            dev = ps[0].device
            key = (gi, dtype)
            cached = self._cache.get(key)
            first_step = False
            # invalidate on any identity change, not just count: a same-
            # length set with different members (freeze/unfreeze, grads
            # becoming None) would otherwise reuse stale device pointers
            if (cached is None or len(cached["params"]) != len(ps)
                    or any(a is not b
                           for a, b in zip(cached["params"], ps))):
                numels = [p.numel() for p in ps]
                prefix = torch.zeros(len(ps), dtype=torch.int64)
                total = 0
                for i, n in enumerate(numels):
                    prefix[i] = total
                    total += n
                moms = []
                for p in ps:
                    st = self.state[p]
                    if mu and "momentum_buffer" not in st:
                        st["momentum_buffer"] = torch.zeros(
                            p.shape, dtype=torch.float32, device=dev)
                        first_step = True
                    moms.append(st["momentum_buffer"].data_ptr() if mu else 0)
                cached = {
                    "params": ps,
                    "total": total,
                    "prefix": prefix.to(dev),
                    "ptr_params": torch.tensor(
                        [p.data_ptr() for p in ps], dtype=torch.int64).to(dev),
                    "ptr_moms": torch.tensor(
                        moms, dtype=torch.int64).to(dev),
                }
                self._cache[key] = cached
                cached["first_step"] = first_step
            first_step = cached.pop("first_step", False)
            # the fused kernel walks param/grad as flat buffers in
            # storage order — any dense layout works (NCHW or
            # channels_last) as long as grad strides match param's.
            # On mismatch, hand the kernel a layout-matched COPY without
            # rebinding p.grad: a rebind would orphan persistent DP
            # bucket views (grads would silently stop reaching the
            # all-reduce payload).
            grads = []
            for p in ps:
                g = p.grad
                if not _same_dense_layout(p, g):
                    g = (g.contiguous() if p.is_contiguous()
                         else g.contiguous(
                             memory_format=torch.channels_last))
                    assert _same_dense_layout(p, g), \
                        "FusedSGD needs grads with the param's layout"
                grads.append(g)
            # grads in persistent buckets (DP) keep their pointers —
            # re-upload the pointer array only when one moved
            gptrs = [g.data_ptr() for g in grads]
            if cached.get("grad_ptrs_host") == gptrs:
                ptr_grads = cached["ptr_grads"]
            else:
                ptr_grads = torch.tensor(gptrs, dtype=torch.int64).to(
                    dev, non_blocking=True)
                cached["grad_ptrs_host"] = gptrs
                cached["ptr_grads"] = ptr_grads
            ext.fused_sgd(cached["ptr_params"], ptr_grads,
                          cached["ptr_moms"], cached["prefix"],
                          cached["total"], lr, mu, wd, first_step,
                          dtype == torch.bfloat16)
