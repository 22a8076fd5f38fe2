"""Native NHWC max-pool: autograd bridge + module swap.

The resnet stem's MaxPool2d was the last sizeable torch-library kernel
on the flagship step (at::max_pool_backward_nhwc ~0.8 ms of ~30 ms).
Forward stores a per-channel argmax byte; backward gathers dy per input
pixel from its covering windows (no zero-init + scatter).
Eligibility: bf16 channels_last, square kernel/stride/pad, dilation 1,
C % 8 == 0."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ddlbench_amd import ops as _ops

_CL = torch.channels_last


class _MaxPoolNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, stride, pad):
        ext = _ops.require_extension()
        x = x.contiguous(memory_format=_CL)
        y, idx = ext.maxpool_fwd(x, k, stride, pad)
        ctx.save_for_backward(idx)
        ctx.geom = (x.shape[2], x.shape[3], k, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _ops.require_extension()
        (idx,) = ctx.saved_tensors
        H, W, k, stride, pad = ctx.geom
        dx = ext.maxpool_bwd(dy.contiguous(memory_format=_CL), idx,
                             H, W, k, stride, pad)
        return dx, None, None, None


def maxpool2d_nhwc(x, k, stride, pad):
    return _MaxPoolNHWC.apply(x, k, stride, pad)


def maxpool_eligible(m: nn.MaxPool2d) -> bool:
    def _sq(v):
        return v if isinstance(v, int) else (
            v[0] if v[0] == v[1] else None)
    k = _sq(m.kernel_size)
    s = _sq(m.stride if m.stride is not None else m.kernel_size)
    p = _sq(m.padding)
    d = _sq(m.dilation)
    return None not in (k, s, p) and d in (1, None) \
        and not m.ceil_mode and k <= 5


class MaxPool2dNHWC(nn.Module):
    """Drop-in for an eligible nn.MaxPool2d on the native kernel."""

    def __init__(self, m: nn.MaxPool2d):
        super().__init__()

        def _sq(v):
            return v if isinstance(v, int) else v[0]
        self.k = _sq(m.kernel_size)
        self.stride = _sq(m.stride if m.stride is not None
                          else m.kernel_size)
        self.pad = _sq(m.padding)

    def forward(self, x):
        if x.is_cuda and x.dtype == torch.bfloat16 \
                and x.shape[1] % 8 == 0:
            return maxpool2d_nhwc(x, self.k, self.stride, self.pad)
        return F.max_pool2d(x, self.k, self.stride, self.pad)

    def extra_repr(self):
        return f"k={self.k}, stride={self.stride}, pad={self.pad} [nhwc]"


def convert_maxpools(model: nn.Module) -> int:
    """Swap eligible nn.MaxPool2d children. Returns conversion count."""
    count = 0
    for parent in model.modules():
        for name, child in list(parent.named_children()):
            if isinstance(child, nn.MaxPool2d) and maxpool_eligible(child):
                setattr(parent, name, MaxPool2dNHWC(child))
                count += 1
    return count
