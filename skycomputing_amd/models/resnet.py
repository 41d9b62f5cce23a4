"""ResNet pipeline zoo (secondary model family).

Capability parity with the reference's registered CNN pieces
(reference: scaelum/model/layers.py:6-261): pipeline-splittable ResNet
blocks proving the layer-config IR is model-agnostic. Convs run through
torch (MIOpen on ROCm) — the hand-kernel budget goes to the BERT hot path.
"""

from __future__ import annotations

import torch.nn as nn

from ..registry import LAYER


@LAYER.register_module
class ResHead(nn.Module):
    """Stem: conv + BN + ReLU (reference: scaelum/model/layers.py:166-177)."""

    def __init__(self, in_channels: int = 3, out_channels: int = 64):
        super().__init__()
        self.conv = nn.Conv2d(in_channels, out_channels, kernel_size=3, padding=1, bias=False)
        self.bn = nn.BatchNorm2d(out_channels)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.relu(self.bn(self.conv(x)))


class BasicBlock(nn.Module):
    """(reference: scaelum/model/layers.py:6-59)"""

    expansion = 1

    def __init__(self, in_channels: int, out_channels: int, stride: int = 1):
        super().__init__()
        self.residual = nn.Sequential(
            nn.Conv2d(in_channels, out_channels, 3, stride=stride, padding=1, bias=False),
            nn.BatchNorm2d(out_channels),
            nn.ReLU(inplace=True),
            nn.Conv2d(out_channels, out_channels * self.expansion, 3, padding=1, bias=False),
            nn.BatchNorm2d(out_channels * self.expansion),
        )
        self.shortcut = nn.Sequential()
        if stride != 1 or in_channels != out_channels * self.expansion:
            self.shortcut = nn.Sequential(
                nn.Conv2d(in_channels, out_channels * self.expansion, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_channels * self.expansion),
            )
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.relu(self.residual(x) + self.shortcut(x))


class BottleNeck(nn.Module):
    """(reference: scaelum/model/layers.py:62-107)"""

    expansion = 4

    def __init__(self, in_channels: int, out_channels: int, stride: int = 1):
        super().__init__()
        self.residual = nn.Sequential(
            nn.Conv2d(in_channels, out_channels, 1, bias=False),
            nn.BatchNorm2d(out_channels),
            nn.ReLU(inplace=True),
            nn.Conv2d(out_channels, out_channels, 3, stride=stride, padding=1, bias=False),
            nn.BatchNorm2d(out_channels),
            nn.ReLU(inplace=True),
            nn.Conv2d(out_channels, out_channels * self.expansion, 1, bias=False),
            nn.BatchNorm2d(out_channels * self.expansion),
        )
        self.shortcut = nn.Sequential()
        if stride != 1 or in_channels != out_channels * self.expansion:
            self.shortcut = nn.Sequential(
                nn.Conv2d(in_channels, out_channels * self.expansion, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_channels * self.expansion),
            )
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.relu(self.residual(x) + self.shortcut(x))


_BLOCKS = {"BasicBlock": BasicBlock, "BottleNeck": BottleNeck}


@LAYER.register_module
class ResLayer(nn.Module):
    """A stage of N blocks (reference: scaelum/model/layers.py:110-147)."""

    def __init__(self, block: str, num_blocks: int, in_channels: int, out_channels: int, stride: int):
        super().__init__()
        blk = _BLOCKS[block] if isinstance(block, str) else block
        strides = [stride] + [1] * (num_blocks - 1)
        layers = []
        ch = in_channels
        for s in strides:
            layers.append(blk(ch, out_channels, s))
            ch = out_channels * blk.expansion
        self.layers = nn.Sequential(*layers)
        self.out_channels = ch

    def forward(self, x):
        return self.layers(x)


@LAYER.register_module
class ResTail(nn.Module):
    """Global pool + classifier (reference: scaelum/model/layers.py:150-163)."""

    def __init__(self, in_features: int, num_class: int = 10):
        super().__init__()
        self.avg_pool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(in_features, num_class)

    def forward(self, x):
        x = self.avg_pool(x).flatten(1)
        return self.fc(x)


def resnet_pipeline_config(depth: int = 18, num_class: int = 10) -> list[dict]:
    """Layer-config list for resnet{18,34,50,101,152}
    (reference ctors: scaelum/model/layers.py:180-261)."""
    specs = {
        18: ("BasicBlock", [2, 2, 2, 2]),
        34: ("BasicBlock", [3, 4, 6, 3]),
        50: ("BottleNeck", [3, 4, 6, 3]),
        101: ("BottleNeck", [3, 4, 23, 3]),
        152: ("BottleNeck", [3, 8, 36, 3]),
    }
    block, counts = specs[depth]
    exp = _BLOCKS[block].expansion
    cfg = [dict(layer_type="ResHead", in_channels=3, out_channels=64)]
    ch = 64
    outs = [64, 128, 256, 512]
    strides = [1, 2, 2, 2]
    for n, out, s in zip(counts, outs, strides):
        cfg.append(dict(layer_type="ResLayer", block=block, num_blocks=n,
                        in_channels=ch, out_channels=out, stride=s))
        ch = out * exp
    cfg.append(dict(layer_type="ResTail", in_features=ch, num_class=num_class))
    return cfg
