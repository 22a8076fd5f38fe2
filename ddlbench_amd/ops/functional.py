"""Autograd bridges for the native CDNA4 kernels, with plain-PyTorch
reference implementations (the CPU path and the numerics-test oracle).

Dispatch: ``ddlbench_amd.ops.use_native`` — native on HIP devices (hard
error there if the extension is missing), torch composition on CPU."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ddlbench_amd import ops as _ops

_ACT = {"none": 0, "relu": 1, "relu6": 2}


def _is_nhwc(t: torch.Tensor) -> bool:
    return (t.dim() == 4
            and t.is_contiguous(memory_format=torch.channels_last)
            and not t.is_contiguous())


def _layout_contiguous(t: torch.Tensor, nhwc: bool) -> torch.Tensor:
    if nhwc:
        return t.contiguous(memory_format=torch.channels_last)
    return t.contiguous()


def _apply_act(v: torch.Tensor, act: str) -> torch.Tensor:
    if act == "relu":
        return F.relu(v, inplace=True)
    if act == "relu6":
        return F.relu6(v, inplace=True)
    return v


# ---------------------------------------------------------------- BN+act
class _FusedBNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, res, running_mean, running_var,
                training, momentum, eps, act_code):
        ext = _ops.require_extension()
        nhwc = _is_nhwc(x)
        x = _layout_contiguous(x, nhwc)
        res = _layout_contiguous(res, nhwc) if res is not None else None
        outs = ext.bn_act_fwd(x, res, gamma, beta, running_mean,
                              running_var, training, momentum,
                              eps, act_code, nhwc)
        y, mean, invstd = outs[0], outs[1], outs[2]
        if len(outs) > 3:
            # 1-bit activation mask: backward never reads y (saves two
            # full-tensor passes AND the y activation memory)
            ctx.save_for_backward(x, mean, invstd, gamma, outs[3])
            ctx.masked = True
        else:
            ctx.save_for_backward(x, mean, invstd, gamma, y)
            ctx.masked = False
        ctx.training = training
        ctx.act_code = act_code
        ctx.has_res = res is not None
        ctx.nhwc = nhwc
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _ops.require_extension()
        x, mean, invstd, gamma, aux = ctx.saved_tensors
        y, mask = (None, aux) if ctx.masked else (aux, None)
        out = ext.bn_act_bwd(_layout_contiguous(dy, ctx.nhwc), y, x, mean,
                             invstd, gamma, ctx.act_code, ctx.training,
                             ctx.has_res, ctx.nhwc, mask)
        dx, dgamma, dbeta = out[0], out[1], out[2]
        dres = out[3] if ctx.has_res else None
        return (dx, dgamma, dbeta, dres, None, None, None, None, None, None)


def bn_act(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
           running_mean: Optional[torch.Tensor],
           running_var: Optional[torch.Tensor], training: bool,
           momentum: float = 0.1, eps: float = 1e-5, act: str = "relu",
           res: Optional[torch.Tensor] = None,
           backend: str = "auto") -> torch.Tensor:
    """BatchNorm2d + activation (+ residual add), fused on GPU."""
    if _ops.use_native(x, backend):
        return _FusedBNAct.apply(x, gamma.float(), beta.float(), res,
                                 running_mean, running_var, training,
                                 momentum, eps, _ACT[act])
    y = F.batch_norm(x, running_mean, running_var, gamma, beta, training,
                     momentum, eps)
    if res is not None:
        y = y + res
    return _apply_act(y, act)


# ----------------------------------------------------------- cross entropy
class _FusedCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        ext = _ops.require_extension()
        logits = logits.contiguous()
        loss_sum, lse = ext.ce_fwd(logits, target)
        ctx.save_for_backward(logits, target, lse)
        return (loss_sum / logits.size(0)).squeeze(0)

    @staticmethod
    def backward(ctx, grad_out):
        ext = _ops.require_extension()
        logits, target, lse = ctx.saved_tensors
        gscale = (grad_out.detach().float() / logits.size(0)).reshape(1)
        dx = ext.ce_bwd(logits, target, lse, gscale.contiguous())
        return dx, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor,
                  backend: str = "auto") -> torch.Tensor:
    if _ops.use_native(logits, backend):
        return _FusedCrossEntropy.apply(logits, target)
    return F.cross_entropy(logits, target)


# ------------------------------------------------------- depthwise conv3x3
class _FusedDWConv3x3(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, stride):
        ext = _ops.require_extension()
        nhwc = _is_nhwc(x) and x.dtype == torch.bfloat16 \
            and x.size(1) % 8 == 0
        x = _layout_contiguous(x, nhwc)
        w32 = weight.detach().float().contiguous()
        y = ext.dw3x3_fwd(x, w32, stride, nhwc)
        ctx.save_for_backward(x, w32)
        ctx.stride = stride
        ctx.wdtype = weight.dtype
        ctx.w_cl = weight.is_contiguous(memory_format=torch.channels_last) \
            and not weight.is_contiguous()
        ctx.nhwc = nhwc
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _ops.require_extension()
        x, w32 = ctx.saved_tensors
        need_dx, need_dw = ctx.needs_input_grad[0], ctx.needs_input_grad[1]
        dx, dw = ext.dw3x3_bwd(x, w32,
                               _layout_contiguous(dy, ctx.nhwc),
                               ctx.stride, need_dx, need_dw, ctx.nhwc)
        if need_dw:
            dw = dw.to(ctx.wdtype)
            if ctx.w_cl:  # match a channels_last weight's layout
                dw = dw.contiguous(memory_format=torch.channels_last)
        return (dx if need_dx else None), (dw if need_dw else None), None


def depthwise_conv3x3(x: torch.Tensor, weight: torch.Tensor,
                      stride: int = 1, backend: str = "auto") -> torch.Tensor:
    """3x3 depthwise conv, pad 1, groups == channels."""
    if _ops.use_native(x, backend):
        return _FusedDWConv3x3.apply(x, weight, stride)
    return F.conv2d(x, weight, None, stride, 1, 1, groups=x.size(1))
