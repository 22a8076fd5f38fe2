"""ResNet family, parameterized over dataset shape.

One implementation covers the reference's three parallel copies
(/root/reference/benchmark/mnist/models/mnistresnet.py,
benchmark/cifar10/pytorchcifargitmodels/resnet.py, torchvision resnet for
imagenet — SURVEY.md §2.6): ``stem="small"`` is the 3×3/stride-1 CIFAR-style
stem used for mnist/cifar10, ``stem="imagenet"`` the 7×7/stride-2 + maxpool
stem used for imagenet/highres.

Every BatchNorm → ReLU (→ residual add) chain is a single fused
``ddlbench_amd.ops.modules.BNAct`` call — one HBM pass on MI355X instead
of three."""

from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ddlbench_amd.ops.modules import BNAct


def conv3x3(cin, cout, stride=1, groups=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False,
                     groups=groups)


def conv1x1(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class Downsample(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.conv = conv1x1(cin, cout, stride)
        self.bn = BNAct(cout, act="none")

    def forward(self, x):
        return self.bn(self.conv(x))


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, planes, stride=1):
        super().__init__()
        self.conv1 = conv3x3(cin, planes, stride)
        self.bn1 = BNAct(planes, act="relu")
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = BNAct(planes, act="relu")  # fused: relu(bn(x) + res)
        self.downsample = None
        if stride != 1 or cin != planes * self.expansion:
            self.downsample = Downsample(cin, planes * self.expansion, stride)

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), res=identity)


class Bottleneck(nn.Module):
    """1x1 -> 3x3 -> 1x1 with fused BN/ReLU/residual. groups/base_width
    give the ResNeXt variant (aggregated transforms — the reference's
    profiler-zoo resnext)."""

    expansion = 4
    groups = 1
    base_width = 64

    def __init__(self, cin, planes, stride=1):
        super().__init__()
        width = int(planes * (self.base_width / 64.0)) * self.groups
        self.conv1 = conv1x1(cin, width)
        self.bn1 = BNAct(width, act="relu")
        self.conv2 = conv3x3(width, width, stride, groups=self.groups)
        self.bn2 = BNAct(width, act="relu")
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = BNAct(planes * self.expansion, act="relu")
        self.downsample = None
        if stride != 1 or cin != planes * self.expansion:
            self.downsample = Downsample(cin, planes * self.expansion, stride)

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), res=identity)


class BottleneckX(Bottleneck):
    groups = 32
    base_width = 4


class Stem(nn.Module):
    def __init__(self, in_channels: int, width: int, kind: str):
        super().__init__()
        self.kind = kind
        if kind == "imagenet":
            self.conv = nn.Conv2d(in_channels, width, 7, stride=2, padding=3,
                                  bias=False)
            self.pool = nn.MaxPool2d(3, stride=2, padding=1)
        else:
            self.conv = conv3x3(in_channels, width)
            self.pool = None
        self.bn = BNAct(width, act="relu")

    def forward(self, x):
        x = self.bn(self.conv(x))
        if self.pool is not None:
            x = self.pool(x)
        return x


class Head(nn.Module):
    def __init__(self, cin: int, num_classes: int):
        super().__init__()
        self.fc = nn.Linear(cin, num_classes)

    def forward(self, x):
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


_LAYERS = {
    "resnet18": (BasicBlock, (2, 2, 2, 2)),
    "resnet34": (BasicBlock, (3, 4, 6, 3)),
    "resnet50": (Bottleneck, (3, 4, 6, 3)),
    "resnet101": (Bottleneck, (3, 4, 23, 3)),
    "resnet152": (Bottleneck, (3, 8, 36, 3)),
    "resnext50_32x4d": (BottleneckX, (3, 4, 6, 3)),
}


class ResNet(nn.Module):
    def __init__(self, arch: str, in_channels: int = 3,
                 num_classes: int = 1000, stem: str = "imagenet"):
        super().__init__()
        block, layers = _LAYERS[arch]
        self.stem = Stem(in_channels, 64, stem)
        self.inplanes = 64
        self.layer1 = self._make_layer(block, 64, layers[0], 1)
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.head = Head(512 * block.expansion, num_classes)

    def _make_layer(self, block, planes, n, stride):
        layers = [block(self.inplanes, planes, stride)]
        self.inplanes = planes * block.expansion
        for _ in range(n - 1):
            layers.append(block(self.inplanes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.stem(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return self.head(x)

    def to_sequential(self) -> nn.Sequential:
        """Flatten into an nn.Sequential of pipeline-partitionable units.

        Residual blocks stay atomic (one module each) — the MI355X-native
        replacement for the reference's torchgpipe @skippable stash/pop
        rebuilds: no cross-partition skip plumbing is
        needed when the block is the partition granule."""
        mods = [self.stem]
        for layer in (self.layer1, self.layer2, self.layer3, self.layer4):
            mods.extend(layer)
        mods.append(self.head)
        return nn.Sequential(*mods)
