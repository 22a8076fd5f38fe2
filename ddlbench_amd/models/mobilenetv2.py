"""MobileNetV2, parameterized over dataset shape.

Covers the reference's mnistmobilenetv2.py (depthwise conv with
groups=planes — /root/reference/benchmark/mnist/models/mnistmobilenetv2.py:28)
and the torchvision imagenet variant. ``stem="small"`` keeps stride 1 in the
stem (CIFAR-style); ``stem="imagenet"`` uses the standard stride-2 schedule.

The depthwise 3×3 convs run on the hand-written CDNA4 depthwise kernel
(ddlbench_amd/ops/csrc/depthwise_conv.hip); every BN→ReLU6 chain and the
projection-BN + residual-add are fused BNAct calls."""

from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ddlbench_amd.ops.modules import BNAct, DepthwiseConv3x3


class InvertedResidual(nn.Module):
    def __init__(self, cin, cout, stride, expand_ratio):
        super().__init__()
        hidden = cin * expand_ratio
        self.use_res = stride == 1 and cin == cout
        self.expand = None
        if expand_ratio != 1:
            self.expand = nn.Conv2d(cin, hidden, 1, bias=False)
            self.expand_bn = BNAct(hidden, act="relu6")
        self.dw = DepthwiseConv3x3(hidden, stride)
        self.dw_bn = BNAct(hidden, act="relu6")
        self.project = nn.Conv2d(hidden, cout, 1, bias=False)
        self.project_bn = BNAct(cout, act="none")

    def forward(self, x):
        out = x
        if self.expand is not None:
            out = self.expand_bn(self.expand(out))
        out = self.dw_bn(self.dw(out))
        out = self.project(out)
        # fused BN + residual add when shapes allow
        return self.project_bn(out, res=x if self.use_res else None)


# (expansion t, out channels c, repeats n, stride s) — MobileNetV2 paper tbl 2
_IMAGENET_CFG = [
    (1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
    (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]
# CIFAR-style: early strides dropped so 32² doesn't collapse
_SMALL_CFG = [
    (1, 16, 1, 1), (6, 24, 2, 1), (6, 32, 3, 2), (6, 64, 4, 2),
    (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]


class _StemConv(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 3, stride=stride, padding=1,
                              bias=False)
        self.bn = BNAct(cout, act="relu6")

    def forward(self, x):
        return self.bn(self.conv(x))


class _LastConv(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 1, bias=False)
        self.bn = BNAct(cout, act="relu6")

    def forward(self, x):
        return self.bn(self.conv(x))


class _Head(nn.Module):
    def __init__(self, fc):
        super().__init__()
        self.fc = fc

    def forward(self, x):
        return self.fc(F.adaptive_avg_pool2d(x, 1).flatten(1))


class MobileNetV2(nn.Module):
    def __init__(self, in_channels: int = 3, num_classes: int = 1000,
                 stem: str = "imagenet"):
        super().__init__()
        cfg = _IMAGENET_CFG if stem == "imagenet" else _SMALL_CFG
        cin = 32
        features = [_StemConv(in_channels, cin,
                              2 if stem == "imagenet" else 1)]
        for t, c, n, s in cfg:
            for i in range(n):
                features.append(InvertedResidual(cin, c, s if i == 0 else 1, t))
                cin = c
        features.append(_LastConv(cin, 1280))
        self.features = nn.Sequential(*features)
        self.classifier = nn.Linear(1280, num_classes)

    def forward(self, x):
        x = self.features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.classifier(x)

    def to_sequential(self) -> nn.Sequential:
        return nn.Sequential(*self.features, _Head(self.classifier))
