"""Extended model zoo: DenseNet, Inception, SqueezeNet, ResNeXt,
MobileNet-v1.

Parity with the reference's pipedream profiler zoo
(/root/reference/pipedream-fork/profiler/image_classification/models/ —
densenet, inception, squeezenet, resnext, mobilenet).
Re-implemented from the architectures on the fused-op stack (BNAct,
DepthwiseConv3x3); every model exposes ``to_sequential()`` with
composite blocks kept atomic so the pipeline engines can partition them
(nasnet is the one profiler-zoo arch not carried — docs/ROADMAP.md)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ddlbench_amd.ops.modules import BNAct, DepthwiseConv3x3


# ------------------------------------------------------------- DenseNet
class DenseLayer(nn.Module):
    """BN-ReLU-Conv1x1 -> BN-ReLU-Conv3x3, output concatenated."""

    def __init__(self, cin: int, growth: int, bn_size: int = 4):
        super().__init__()
        self.bn1 = BNAct(cin, act="relu")
        self.conv1 = nn.Conv2d(cin, bn_size * growth, 1, bias=False)
        self.bn2 = BNAct(bn_size * growth, act="relu")
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)

    def forward(self, x):
        y = self.conv2(self.bn2(self.conv1(self.bn1(x))))
        return torch.cat([x, y], dim=1)


class DenseBlock(nn.Module):
    def __init__(self, cin: int, n_layers: int, growth: int):
        super().__init__()
        self.layers = nn.ModuleList(
            [DenseLayer(cin + i * growth, growth) for i in range(n_layers)])
        self.out_channels = cin + n_layers * growth

    def forward(self, x):
        for l in self.layers:
            x = l(x)
        return x


class Transition(nn.Module):
    def __init__(self, cin: int, cout: int):
        super().__init__()
        self.bn = BNAct(cin, act="relu")
        self.conv = nn.Conv2d(cin, cout, 1, bias=False)

    def forward(self, x):
        return F.avg_pool2d(self.conv(self.bn(x)), 2)


_DENSENET_CFG = {
    "densenet121": (32, (6, 12, 24, 16)),
    "densenet169": (32, (6, 12, 32, 32)),
}


class DenseNet(nn.Module):
    def __init__(self, arch: str = "densenet121", in_channels: int = 3,
                 num_classes: int = 1000, stem: str = "imagenet"):
        super().__init__()
        growth, blocks = _DENSENET_CFG[arch]
        c = 2 * growth
        if stem == "imagenet":
            stem_mods = [nn.Conv2d(in_channels, c, 7, stride=2, padding=3,
                                   bias=False), BNAct(c, act="relu"),
                         nn.MaxPool2d(3, stride=2, padding=1)]
        else:
            stem_mods = [nn.Conv2d(in_channels, c, 3, padding=1,
                                   bias=False), BNAct(c, act="relu")]
        mods = stem_mods
        for i, n in enumerate(blocks):
            blk = DenseBlock(c, n, growth)
            mods.append(blk)
            c = blk.out_channels
            if i != len(blocks) - 1:
                mods.append(Transition(c, c // 2))
                c //= 2
        mods.append(BNAct(c, act="relu"))
        self.features = nn.Sequential(*mods)
        self.classifier = nn.Linear(c, num_classes)

    def forward(self, x):
        x = self.features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.classifier(x)

    def to_sequential(self) -> nn.Sequential:
        return nn.Sequential(*self.features,
                             _PoolFC(self.classifier))


class _PoolFC(nn.Module):
    def __init__(self, fc):
        super().__init__()
        self.fc = fc

    def forward(self, x):
        return self.fc(F.adaptive_avg_pool2d(x, 1).flatten(1))


# ------------------------------------------------------------ Inception
class _CBR(nn.Module):
    def __init__(self, cin, cout, k, stride=1, padding=0):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, k, stride=stride, padding=padding,
                              bias=False)
        self.bn = BNAct(cout, act="relu")

    def forward(self, x):
        return self.bn(self.conv(x))


class InceptionA(nn.Module):
    """The 4-branch 35x35 block of Inception-v3 (1x1 / 5x5 / double-3x3 /
    pool), concatenated."""

    def __init__(self, cin, pool_features):
        super().__init__()
        self.b1 = _CBR(cin, 64, 1)
        self.b5 = nn.Sequential(_CBR(cin, 48, 1), _CBR(48, 64, 5,
                                                       padding=2))
        self.b3 = nn.Sequential(_CBR(cin, 64, 1),
                                _CBR(64, 96, 3, padding=1),
                                _CBR(96, 96, 3, padding=1))
        self.bp = _CBR(cin, pool_features, 1)
        self.out_channels = 64 + 64 + 96 + pool_features

    def forward(self, x):
        pool = self.bp(F.avg_pool2d(x, 3, stride=1, padding=1))
        return torch.cat([self.b1(x), self.b5(x), self.b3(x), pool], 1)


class InceptionB(nn.Module):
    """Grid-reduction block (stride-2 branches + maxpool)."""

    def __init__(self, cin):
        super().__init__()
        self.b3 = _CBR(cin, 384, 3, stride=2)
        self.bd = nn.Sequential(_CBR(cin, 64, 1),
                                _CBR(64, 96, 3, padding=1),
                                _CBR(96, 96, 3, stride=2))
        self.out_channels = cin + 384 + 96

    def forward(self, x):
        return torch.cat([self.b3(x), self.bd(x),
                          F.max_pool2d(x, 3, stride=2)], 1)


class InceptionC(nn.Module):
    """Factorized 7x7 block (1x7/7x1 chains)."""

    def __init__(self, cin, ch7):
        super().__init__()
        self.b1 = _CBR(cin, 192, 1)
        self.b7 = nn.Sequential(
            _CBR(cin, ch7, 1), _CBR(ch7, ch7, (1, 7), padding=(0, 3)),
            _CBR(ch7, 192, (7, 1), padding=(3, 0)))
        self.b7d = nn.Sequential(
            _CBR(cin, ch7, 1), _CBR(ch7, ch7, (7, 1), padding=(3, 0)),
            _CBR(ch7, ch7, (1, 7), padding=(0, 3)),
            _CBR(ch7, ch7, (7, 1), padding=(3, 0)),
            _CBR(ch7, 192, (1, 7), padding=(0, 3)))
        self.bp = _CBR(cin, 192, 1)
        self.out_channels = 192 * 4

    def forward(self, x):
        pool = self.bp(F.avg_pool2d(x, 3, stride=1, padding=1))
        return torch.cat([self.b1(x), self.b7(x), self.b7d(x), pool], 1)


class Inception3(nn.Module):
    """Compact Inception-v3 (no aux head — the reference's profiler copy
    also runs aux-free for throughput). ImageNet/highres shapes only."""

    def __init__(self, in_channels: int = 3, num_classes: int = 1000):
        super().__init__()
        self.stem = nn.Sequential(
            _CBR(in_channels, 32, 3, stride=2), _CBR(32, 32, 3),
            _CBR(32, 64, 3, padding=1), nn.MaxPool2d(3, stride=2),
            _CBR(64, 80, 1), _CBR(80, 192, 3), nn.MaxPool2d(3, stride=2))
        a1 = InceptionA(192, 32)
        a2 = InceptionA(a1.out_channels, 64)
        a3 = InceptionA(a2.out_channels, 64)
        b = InceptionB(a3.out_channels)
        c1 = InceptionC(b.out_channels, 128)
        c2 = InceptionC(c1.out_channels, 160)
        c3 = InceptionC(c2.out_channels, 192)
        self.blocks = nn.Sequential(a1, a2, a3, b, c1, c2, c3)
        self.fc = nn.Linear(c3.out_channels, num_classes)

    def forward(self, x):
        x = self.blocks(self.stem(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)

    def to_sequential(self) -> nn.Sequential:
        return nn.Sequential(self.stem, *self.blocks, _PoolFC(self.fc))


# ------------------------------------------------------------ SqueezeNet
class Fire(nn.Module):
    def __init__(self, cin, squeeze, e1, e3):
        super().__init__()
        self.squeeze = nn.Conv2d(cin, squeeze, 1)
        self.e1 = nn.Conv2d(squeeze, e1, 1)
        self.e3 = nn.Conv2d(squeeze, e3, 3, padding=1)
        self.out_channels = e1 + e3

    def forward(self, x):
        s = F.relu(self.squeeze(x), inplace=True)
        return torch.cat([F.relu(self.e1(s), inplace=True),
                          F.relu(self.e3(s), inplace=True)], 1)


class SqueezeNet(nn.Module):
    """SqueezeNet 1.1."""

    def __init__(self, in_channels: int = 3, num_classes: int = 1000,
                 stem: str = "imagenet"):
        super().__init__()
        stride = 2 if stem == "imagenet" else 1
        self.features = nn.Sequential(
            nn.Conv2d(in_channels, 64, 3, stride=stride), nn.ReLU(True),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
            Fire(64, 16, 64, 64), Fire(128, 16, 64, 64),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
            Fire(128, 32, 128, 128), Fire(256, 32, 128, 128),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
            Fire(256, 48, 192, 192), Fire(384, 48, 192, 192),
            Fire(384, 64, 256, 256), Fire(512, 64, 256, 256))
        self.classifier_conv = nn.Conv2d(512, num_classes, 1)

    def forward(self, x):
        x = F.relu(self.classifier_conv(self.features(x)), inplace=True)
        return F.adaptive_avg_pool2d(x, 1).flatten(1)

    def to_sequential(self) -> nn.Sequential:
        class _Head(nn.Module):
            def __init__(self, conv):
                super().__init__()
                self.conv = conv

            def forward(self, x):
                x = F.relu(self.conv(x), inplace=True)
                return F.adaptive_avg_pool2d(x, 1).flatten(1)

        return nn.Sequential(*self.features, _Head(self.classifier_conv))


# ----------------------------------------------------------- MobileNetV1
class DWSeparable(nn.Module):
    """Depthwise 3x3 (our HIP kernel) + pointwise 1x1, each BN+ReLU."""

    def __init__(self, cin, cout, stride):
        super().__init__()
        self.dw = DepthwiseConv3x3(cin, stride)
        self.bn1 = BNAct(cin, act="relu")
        self.pw = nn.Conv2d(cin, cout, 1, bias=False)
        self.bn2 = BNAct(cout, act="relu")

    def forward(self, x):
        return self.bn2(self.pw(self.bn1(self.dw(x))))


class MobileNetV1(nn.Module):
    def __init__(self, in_channels: int = 3, num_classes: int = 1000,
                 stem: str = "imagenet"):
        super().__init__()
        s0 = 2 if stem == "imagenet" else 1
        cfg = [(64, 1), (128, 2 if stem == "imagenet" else 1), (128, 1),
               (256, 2), (256, 1), (512, 2)] + [(512, 1)] * 5 + \
              [(1024, 2), (1024, 1)]
        mods = [nn.Conv2d(in_channels, 32, 3, stride=s0, padding=1,
                          bias=False), BNAct(32, act="relu")]
        c = 32
        for cout, s in cfg:
            mods.append(DWSeparable(c, cout, s))
            c = cout
        self.features = nn.Sequential(*mods)
        self.fc = nn.Linear(1024, num_classes)

    def forward(self, x):
        x = self.features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)

    def to_sequential(self) -> nn.Sequential:
        return nn.Sequential(*self.features, _PoolFC(self.fc))
