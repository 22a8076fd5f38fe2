"""NASNet-A (mobile) — the last profiler-zoo architecture
(/root/reference/pipedream-fork/profiler/image_classification/models/
vendors a nasnet copy).

Implemented from the published NASNet-A cell structure (Zoph et al.,
"Learning Transferable Architectures", CVPR'18): a stem, then stacks of
Normal cells with Reduction cells between stacks; every cell combines
its two predecessor feature maps through five op-pairs (separable convs
3x3/5x5/7x7 as depthwise+pointwise twice, avg/max pools, identity) and
concatenates the block outputs. Separable 3x3 depthwise runs on the
in-tree HIP kernel; 5x5/7x7 depthwise use the library grouped conv.
Cell wiring follows the paper's figure; the vendored copy's exact
channel bookkeeping may differ in minor details (documented, not
bit-claimed)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ddlbench_amd.ops.modules import BNAct, DepthwiseConv3x3


class SepConv(nn.Module):
    """NASNet separable conv: ReLU -> (depthwise k, pointwise 1x1, BN)
    applied twice, the second at stride 1."""

    def __init__(self, cin, cout, k, stride):
        super().__init__()
        pad = k // 2
        if k == 3:
            self.dw1 = DepthwiseConv3x3(cin, stride)
        else:
            self.dw1 = nn.Conv2d(cin, cin, k, stride=stride, padding=pad,
                                 groups=cin, bias=False)
        self.pw1 = nn.Conv2d(cin, cout, 1, bias=False)
        self.bn1 = BNAct(cout, act="none")
        if k == 3:
            self.dw2 = DepthwiseConv3x3(cout, 1)
        else:
            self.dw2 = nn.Conv2d(cout, cout, k, padding=pad, groups=cout,
                                 bias=False)
        self.pw2 = nn.Conv2d(cout, cout, 1, bias=False)
        self.bn2 = BNAct(cout, act="none")

    def forward(self, x):
        x = self.bn1(self.pw1(self.dw1(F.relu(x))))
        return self.bn2(self.pw2(self.dw2(F.relu(x))))


class Fit(nn.Module):
    """Project a predecessor map to (cout, H, W) of the cell (1x1 conv,
    with stride-2 when the spatial sizes differ)."""

    def __init__(self, cin, cout, reduce_spatial):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 1,
                              stride=2 if reduce_spatial else 1,
                              bias=False)
        self.bn = BNAct(cout, act="none")

    def forward(self, x):
        return self.bn(self.conv(F.relu(x)))


class NormalCell(nn.Module):
    """NASNet-A normal cell: 5 op-pairs over (h = prev, hp = prev-prev),
    output = concat of the five block sums (5*planes channels)."""

    def __init__(self, c_h, c_hp, planes, hp_reduced):
        super().__init__()
        self.fit_h = Fit(c_h, planes, False)
        self.fit_hp = Fit(c_hp, planes, hp_reduced)
        self.b0_l = SepConv(planes, planes, 3, 1)   # + h (identity)
        self.b1_l = SepConv(planes, planes, 3, 1)
        self.b1_r = SepConv(planes, planes, 5, 1)
        self.b2_r = SepConv(planes, planes, 5, 1)   # avg(h) + sep5(hp)
        self.b4_l = SepConv(planes, planes, 3, 1)
        self.out_channels = 5 * planes

    def forward(self, h, hp):
        x = self.fit_h(h)
        xp = self.fit_hp(hp)
        b0 = self.b0_l(x) + x
        b1 = self.b1_l(xp) + self.b1_r(x)
        b2 = F.avg_pool2d(x, 3, 1, 1) + self.b2_r(xp)
        b3 = F.avg_pool2d(xp, 3, 1, 1) + F.avg_pool2d(xp, 3, 1, 1)
        b4 = self.b4_l(xp) + xp
        return torch.cat([b0, b1, b2, b3, b4], 1)


class ReductionCell(nn.Module):
    """NASNet-A reduction cell (stride-2 first ops), 4 concatenated
    block outputs."""

    def __init__(self, c_h, c_hp, planes, hp_reduced):
        super().__init__()
        self.fit_h = Fit(c_h, planes, False)
        self.fit_hp = Fit(c_hp, planes, hp_reduced)
        self.b0_l = SepConv(planes, planes, 7, 2)
        self.b0_r = SepConv(planes, planes, 5, 2)
        self.b1_r = SepConv(planes, planes, 7, 2)
        self.b2_r = SepConv(planes, planes, 5, 2)
        self.b4_l = SepConv(planes, planes, 3, 1)
        self.out_channels = 4 * planes

    def forward(self, h, hp):
        x = self.fit_h(h)
        xp = self.fit_hp(hp)
        b0 = self.b0_l(xp) + self.b0_r(x)
        b1 = F.max_pool2d(x, 3, 2, 1) + self.b1_r(xp)
        b2 = F.avg_pool2d(x, 3, 2, 1) + self.b2_r(xp)
        b3 = F.max_pool2d(x, 3, 2, 1) + self.b4_l(b0)
        b4 = F.avg_pool2d(b0, 3, 1, 1) + b1
        return torch.cat([b1, b2, b3, b4], 1)


class _CellChainUnit(nn.Module):
    """Carries the (h, hp) pair through one cell so the whole network
    flattens to an nn.Sequential of tuple-passing units."""

    def __init__(self, cell):
        super().__init__()
        self.cell = cell

    def forward(self, state):
        h, hp = state
        return self.cell(h, hp), h


class NASNetAMobile(nn.Module):
    def __init__(self, in_channels: int = 3, num_classes: int = 1000,
                 stem: str = "imagenet", planes: int = 44,
                 cells_per_stack: int = 4):
        super().__init__()
        s0 = 2 if stem == "imagenet" else 1
        self.stem_conv = nn.Conv2d(in_channels, 32, 3, stride=s0,
                                   padding=1, bias=False)
        self.stem_bn = BNAct(32, act="none")
        cells = []
        c_h, c_hp = 32, 32
        p = planes
        hp_reduced = False
        for stack in range(3):
            if stack > 0:
                cell = ReductionCell(c_h, c_hp, p, hp_reduced)
                cells.append(cell)
                c_h, c_hp = cell.out_channels, c_h
                hp_reduced = True
                p *= 2
            for _ in range(cells_per_stack):
                cell = NormalCell(c_h, c_hp, p, hp_reduced)
                cells.append(cell)
                c_h, c_hp = cell.out_channels, c_h
                hp_reduced = False
        self.cells = nn.ModuleList(cells)
        self.fc = nn.Linear(c_h, num_classes)

    def forward(self, x):
        h = self.stem_bn(self.stem_conv(x))
        hp = h
        for cell in self.cells:
            h, hp = cell(h, hp), h
        h = F.relu(h)
        h = F.adaptive_avg_pool2d(h, 1).flatten(1)
        return self.fc(h)

    def to_sequential(self) -> nn.Sequential:
        class _Stem(nn.Module):
            def __init__(self, conv, bn):
                super().__init__()
                self.conv = conv
                self.bn = bn

            def forward(self, x):
                h = self.bn(self.conv(x))
                return h, h

        class _Head(nn.Module):
            def __init__(self, fc):
                super().__init__()
                self.fc = fc

            def forward(self, state):
                h, _ = state
                h = F.adaptive_avg_pool2d(F.relu(h), 1).flatten(1)
                return self.fc(h)

        return nn.Sequential(_Stem(self.stem_conv, self.stem_bn),
                             *[_CellChainUnit(c) for c in self.cells],
                             _Head(self.fc))
