"""MFMA implicit-GEMM convolution: autograd bridge + model converter.

Forward and data-grad run on the hand-written NHWC bf16 kernels
(csrc/conv_mfma*.hip) and the weight-grad on the transpose-read kernel
(csrc/conv_wgrad.hip) — the full conv triple is in-tree by default.

Eligibility: bf16, channels_last, groups=1, dilation=1, C%8==0, K%8==0.
``convert_convs(model)`` swaps every eligible nn.Conv2d for Conv2dMFMA
(stem convs with C=3 stay on the library path)."""

from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from ddlbench_amd import ops as _ops

# weight-grad backend: the native kernel (conv_wgrad.hip) by default —
# DDLB_WGRAD=library opts back into aten's conv backward
_WGRAD = os.environ.get("DDLB_WGRAD", "mfma")

_CL = torch.channels_last


class _ConvMFMA(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, stride, pad):
        ext = _ops.require_extension()
        x = x.contiguous(memory_format=_CL)
        w = weight.contiguous(memory_format=_CL)
        y = ext.conv_igemm_fwd(x, w, stride, pad)
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        ctx.pad = pad
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _ops.require_extension()
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=_CL)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            # (K,C,R,S) logical -> (C,R,S,K) memory for the dgrad B tile
            w_perm = weight.permute(1, 2, 3, 0).contiguous()
            dx = ext.conv_igemm_dgrad(dy, w_perm, x.size(0), x.size(1),
                                      x.size(2), x.size(3), ctx.stride,
                                      ctx.pad)
        if ctx.needs_input_grad[1]:
            if _WGRAD == "mfma":
                R = weight.shape[2]
                dw32 = ext.conv_igemm_wgrad(x, dy, R, weight.shape[3],
                                            ctx.stride, ctx.pad)
                # (K, R*S*C) memory -> logical (K,C,R,S) channels_last
                dw = dw32.view(weight.shape[0], R, weight.shape[3],
                               weight.shape[1]) \
                    .permute(0, 3, 1, 2).to(weight.dtype)
            else:
                dw = torch.ops.aten.convolution_backward(
                    dy, x, weight, None, [ctx.stride, ctx.stride],
                    [ctx.pad, ctx.pad], [1, 1], False, [0, 0], 1,
                    [False, True, False])[1]
        return dx, dw, None, None


def conv2d_mfma(x, weight, stride=1, pad=0):
    return _ConvMFMA.apply(x, weight, stride, pad)


def mfma_eligible(conv: nn.Conv2d, x_dtype=torch.bfloat16) -> bool:
    return (x_dtype == torch.bfloat16
            and conv.groups == 1
            and conv.dilation == (1, 1)
            and conv.bias is None
            and conv.in_channels % 8 == 0
            and conv.out_channels % 8 == 0
            and conv.stride[0] == conv.stride[1]
            and conv.padding[0] == conv.padding[1]
            and conv.kernel_size[0] == conv.kernel_size[1])


class Conv2dMFMA(nn.Module):
    """Drop-in for an eligible nn.Conv2d, running the MFMA kernel."""

    def __init__(self, conv: nn.Conv2d):
        super().__init__()
        self.in_channels = conv.in_channels
        self.out_channels = conv.out_channels
        self.kernel_size = conv.kernel_size
        self.stride = conv.stride[0]
        self.padding = conv.padding[0]
        self.weight = conv.weight

    def forward(self, x):
        if x.is_cuda and x.dtype == torch.bfloat16:
            return conv2d_mfma(x, self.weight, self.stride, self.padding)
        return F.conv2d(x, self.weight, None, self.stride, self.padding)

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, "
                f"k={self.kernel_size}, stride={self.stride}, "
                f"pad={self.padding} [mfma]")


def convert_convs(model: nn.Module, dtype=torch.bfloat16) -> int:
    """Replace eligible nn.Conv2d children with Conv2dMFMA and eligible
    nn.MaxPool2d with the native NHWC pool. Returns the number of
    conversions."""
    count = 0
    for parent in model.modules():
        for name, child in list(parent.named_children()):
            if isinstance(child, nn.Conv2d) and mfma_eligible(child, dtype):
                setattr(parent, name, Conv2dMFMA(child))
                count += 1
    if dtype == torch.bfloat16 and \
            os.environ.get("DDLB_MAXPOOL", "1") != "0":
        from ddlbench_amd.ops.pool import convert_maxpools
        count += convert_maxpools(model)
    return count
