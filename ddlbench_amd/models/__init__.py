"""Model zoo registry.

One parametric implementation per family replaces the reference's three
parallel per-dataset copies plus the per-dataset gpipemodels rebuilds
(SURVEY.md §2.6): ``build_model(dataset, arch)`` resolves in_channels /
num_classes / stem from the dataset, and every model exposes
``to_sequential()`` for the pipeline engines.
"""

from __future__ import annotations

import torch.nn as nn

from ddlbench_amd.config import DATASET_SHAPES
from ddlbench_amd.models.resnet import ResNet
from ddlbench_amd.models.vgg import VGG
from ddlbench_amd.models.mobilenetv2 import MobileNetV2
from ddlbench_amd.models.extra import (DenseNet, Inception3, MobileNetV1,
                                       SqueezeNet)
from ddlbench_amd.models.nasnet import NASNetAMobile

RESNETS = ("resnet18", "resnet34", "resnet50", "resnet101", "resnet152",
           "resnext50_32x4d")
VGGS = ("vgg11", "vgg13", "vgg16", "vgg19")
DENSENETS = ("densenet121", "densenet169")
ARCHS = RESNETS + VGGS + DENSENETS + (
    "mobilenetv2", "mobilenetv1", "squeezenet", "inception3",
    "nasnetamobile")


def build_model(dataset: str, arch: str) -> nn.Module:
    c, h, w, ncls, _, _ = DATASET_SHAPES[dataset]
    stem = "imagenet" if dataset in ("imagenet", "highres") else "small"
    if arch in RESNETS:
        return ResNet(arch, in_channels=c, num_classes=ncls, stem=stem)
    if arch in VGGS:
        return VGG(arch, in_channels=c, num_classes=ncls, stem=stem)
    if arch in DENSENETS:
        return DenseNet(arch, in_channels=c, num_classes=ncls, stem=stem)
    if arch == "mobilenetv2":
        return MobileNetV2(in_channels=c, num_classes=ncls, stem=stem)
    if arch == "mobilenetv1":
        return MobileNetV1(in_channels=c, num_classes=ncls, stem=stem)
    if arch == "squeezenet":
        return SqueezeNet(in_channels=c, num_classes=ncls, stem=stem)
    if arch == "nasnetamobile":
        return NASNetAMobile(in_channels=c, num_classes=ncls, stem=stem)
    if arch == "inception3":
        if dataset not in ("imagenet", "highres"):
            raise ValueError("inception3 needs imagenet/highres inputs "
                             "(aggressive stem downsampling)")
        return Inception3(in_channels=c, num_classes=ncls)
    raise ValueError(f"unknown arch {arch!r}; choose from {ARCHS}")


def build_sequential(dataset: str, arch: str) -> nn.Sequential:
    """Sequential-flattened variant for pipeline partitioning."""
    return build_model(dataset, arch).to_sequential()
