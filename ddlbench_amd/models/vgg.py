"""VGG family (11/13/16/19), parameterized over dataset shape.

Covers the reference's mnistvgg.py / pytorchcifargitmodels/vgg.py /
torchvision VGG. Small inputs use the CIFAR-style single
Linear classifier; imagenet inputs the 4096-wide 3-layer classifier.
Conv → BN → ReLU chains use the fused BNAct op."""

from __future__ import annotations

import torch.nn as nn

from ddlbench_amd.ops.modules import BNAct

_CFG = {
    "vgg11": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "vgg13": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M",
              512, 512, "M"],
    "vgg16": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
              512, 512, 512, "M", 512, 512, 512, "M"],
    "vgg19": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M",
              512, 512, 512, 512, "M", 512, 512, 512, 512, "M"],
}


class ConvBNReLU(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 3, padding=1, bias=False)
        self.bn = BNAct(cout, act="relu")

    def forward(self, x):
        return self.bn(self.conv(x))


class VGG(nn.Module):
    def __init__(self, arch: str, in_channels: int = 3,
                 num_classes: int = 1000, stem: str = "imagenet"):
        super().__init__()
        layers = []
        cin = in_channels
        for v in _CFG[arch]:
            if v == "M":
                # ceil_mode keeps 28x28 (mnist) alive through 5 pools
                layers.append(nn.MaxPool2d(2, 2, ceil_mode=(stem != "imagenet")))
            else:
                layers.append(ConvBNReLU(cin, v))
                cin = v
        self.features = nn.Sequential(*layers)
        self.pool = nn.AdaptiveAvgPool2d(7 if stem == "imagenet" else 1)
        if stem == "imagenet":
            self.classifier = nn.Sequential(
                nn.Flatten(),
                nn.Linear(512 * 49, 4096), nn.ReLU(inplace=True),
                nn.Dropout(0.5),
                nn.Linear(4096, 4096), nn.ReLU(inplace=True),
                nn.Dropout(0.5),
                nn.Linear(4096, num_classes))
        else:
            self.classifier = nn.Sequential(
                nn.Flatten(), nn.Linear(512, num_classes))

    def forward(self, x):
        return self.classifier(self.pool(self.features(x)))

    def to_sequential(self) -> nn.Sequential:
        mods = list(self.features) + [self.pool] + list(self.classifier)
        return nn.Sequential(*mods)
