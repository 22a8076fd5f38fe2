"""nn.Module wrappers over the fused kernels.

``BNAct`` replaces the BatchNorm2d → ReLU (→ residual add) chains of the
model zoo with one fused op; ``DepthwiseConv3x3`` replaces
Conv2d(groups=channels). Both keep plain-PyTorch semantics (state dict
compatible with nn.BatchNorm2d / nn.Conv2d field names)."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from ddlbench_amd.ops import functional as NF

# module-level default backend — engines set this once from the config
_DEFAULT_BACKEND = "auto"


def set_default_backend(backend: str) -> None:
    global _DEFAULT_BACKEND
    assert backend in ("auto", "native", "torch")
    _DEFAULT_BACKEND = backend


def default_backend() -> str:
    return _DEFAULT_BACKEND


class BNAct(nn.Module):
    """BatchNorm2d + activation (+ optional residual add), fused on GPU.

    forward(x, res=None): y = act(bn(x) + res)
    Running stats and affine params stay fp32 regardless of model dtype
    (they are re-cast at dispatch; fp32 optimizer-state discipline)."""

    def __init__(self, num_features: int, act: str = "relu",
                 eps: float = 1e-5, momentum: float = 0.1):
        super().__init__()
        assert act in ("none", "relu", "relu6")
        self.num_features = num_features
        self.act = act
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))
        self._nbt = 0

    def _apply(self, fn, recurse=True):
        # keep BN params/stats fp32 under model.to(bf16): re-cast after
        super()._apply(fn, recurse)
        for name in ("weight", "bias"):
            p = getattr(self, name)
            if p is not None and p.dtype != torch.float32:
                p.data = p.data.float()
                if p.grad is not None:
                    p.grad = p.grad.float()
        for name in ("running_mean", "running_var"):
            b = getattr(self, name)
            if b is not None and b.dtype != torch.float32:
                setattr(self, name, b.float())
        return self

    def forward(self, x: torch.Tensor,
                res: Optional[torch.Tensor] = None) -> torch.Tensor:
        if self.training:
            # python-side counter; synced to the buffer on state_dict()
            # (a per-step GPU add showed up as 51 launches/step)
            self._nbt += 1
        return NF.bn_act(x, self.weight, self.bias, self.running_mean,
                         self.running_var, self.training, self.momentum,
                         self.eps, self.act, res,
                         backend=_DEFAULT_BACKEND)

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        if self._nbt:
            self.num_batches_tracked += self._nbt
            self._nbt = 0
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def extra_repr(self) -> str:
        return f"{self.num_features}, act={self.act}"


class DepthwiseConv3x3(nn.Module):
    """3x3 depthwise conv (groups == channels), pad 1, no bias."""

    def __init__(self, channels: int, stride: int = 1):
        super().__init__()
        assert stride in (1, 2)
        self.channels = channels
        self.stride = stride
        self.weight = nn.Parameter(torch.empty(channels, 1, 3, 3))
        nn.init.kaiming_normal_(self.weight, mode="fan_out")

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return NF.depthwise_conv3x3(x, self.weight, self.stride,
                                    backend=_DEFAULT_BACKEND)

    def extra_repr(self) -> str:
        return f"{self.channels}, stride={self.stride}"


class CrossEntropyLoss(nn.Module):
    """Fused softmax cross-entropy (mean reduction)."""

    def forward(self, logits: torch.Tensor,
                target: torch.Tensor) -> torch.Tensor:
        return NF.cross_entropy(logits, target, backend=_DEFAULT_BACKEND)
