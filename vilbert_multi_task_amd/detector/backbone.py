"""ResNeXt-152-32x8d + FPN backbone, written as plain PyTorch modules.

MI355X-native stance (SURVEY.md §2.3 row "backbone convs"): the detector is
a feeder, not the north-star hot loop — grouped convolutions run on MIOpen
through PyTorch-ROCm; the hand-written HIP effort goes to NMS/RoIAlign and
the ViLBERT transformer. Geometry matches the e2e_faster_rcnn_X-152-32x8d-FPN
config the reference pins at /root/reference/worker.py:68-75: stem 7x7/2,
stages [3, 8, 36, 3], groups=32, width_per_group=8, FrozenBatchNorm, FPN
P2-P6 at 256 channels. `width_mult`/`stage_blocks` shrink the net for CPU
tests.
"""

from __future__ import annotations

from typing import List, Sequence

import torch
import torch.nn as nn
import torch.nn.functional as F


class FrozenBatchNorm2d(nn.Module):
    """BN with fixed affine params folded at inference (detector is
    serving-only here, like the reference's .eval() model)."""

    def __init__(self, n: int):
        super().__init__()
        self.register_buffer("weight", torch.ones(n))
        self.register_buffer("bias", torch.zeros(n))
        self.register_buffer("running_mean", torch.zeros(n))
        self.register_buffer("running_var", torch.ones(n))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        scale = self.weight * (self.running_var + 1e-5).rsqrt()
        bias = self.bias - self.running_mean * scale
        return x * scale.view(1, -1, 1, 1) + bias.view(1, -1, 1, 1)


class Bottleneck(nn.Module):
    expansion = 2  # ResNeXt-8d: width 8*32=256 at stage1 -> out 256? (out = planes*2)

    def __init__(self, in_ch: int, planes: int, stride: int, groups: int, base_width: int):
        super().__init__()
        width = planes * base_width * groups // 64
        width = max((width + groups - 1) // groups, 1) * groups  # divisible by groups
        out_ch = planes * 4
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = FrozenBatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1, groups=groups, bias=False)
        self.bn2 = FrozenBatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = FrozenBatchNorm2d(out_ch)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False),
                FrozenBatchNorm2d(out_ch),
            )

    def forward(self, x):
        idn = x if self.down is None else self.down(x)
        y = F.relu(self.bn1(self.conv1(x)))
        y = F.relu(self.bn2(self.conv2(y)))
        y = self.bn3(self.conv3(y))
        return F.relu(y + idn)


class ResNeXtFPN(nn.Module):
    def __init__(
        self,
        stage_blocks: Sequence[int] = (3, 8, 36, 3),
        groups: int = 32,
        base_width: int = 8,
        stem_ch: int = 64,
        fpn_ch: int = 256,
        width_mult: float = 1.0,
    ):
        super().__init__()
        w = lambda c: max(int(c * width_mult), 8)
        self.stem = nn.Sequential(
            nn.Conv2d(3, w(stem_ch), 7, stride=2, padding=3, bias=False),
            FrozenBatchNorm2d(w(stem_ch)),
            nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1),
        )
        planes = [w(64), w(128), w(256), w(512)]
        in_ch = w(stem_ch)
        self.stages = nn.ModuleList()
        self.out_channels: List[int] = []
        for i, (p, n) in enumerate(zip(planes, stage_blocks)):
            blocks = []
            stride = 1 if i == 0 else 2
            for b in range(n):
                blocks.append(
                    Bottleneck(in_ch, p, stride if b == 0 else 1, groups, base_width)
                )
                in_ch = p * 4
            self.stages.append(nn.Sequential(*blocks))
            self.out_channels.append(in_ch)
        # FPN lateral + output convs over C2..C5 -> P2..P5 (+P6 maxpool)
        self.fpn_ch = fpn_ch
        self.lateral = nn.ModuleList(
            [nn.Conv2d(c, fpn_ch, 1) for c in self.out_channels]
        )
        self.output = nn.ModuleList(
            [nn.Conv2d(fpn_ch, fpn_ch, 3, padding=1) for _ in self.out_channels]
        )

    def forward(self, x: torch.Tensor) -> List[torch.Tensor]:
        x = self.stem(x)
        cs = []
        for stage in self.stages:
            x = stage(x)
            cs.append(x)
        # top-down pathway
        ps = [None] * 4
        last = self.lateral[3](cs[3])
        ps[3] = self.output[3](last)
        for i in (2, 1, 0):
            lat = self.lateral[i](cs[i])
            last = lat + F.interpolate(last, size=lat.shape[-2:], mode="nearest")
            ps[i] = self.output[i](last)
        p6 = F.max_pool2d(ps[3], 1, stride=2)
        return ps + [p6]  # [P2, P3, P4, P5, P6], strides 4,8,16,32,64
