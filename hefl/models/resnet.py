"""ResNet-18 (NHWC, bf16 activations) for BASELINE.json config #5
(8-client ResNet-18 on 128x128x3, CKKS n=2^15 deep RNS chain).

Standard torchvision-style topology built entirely from hefl's HIP-backed
ops: 7x7 s2 stem conv -> BN+ReLU -> 3x3 s2 maxpool -> 4 stages of 2
BasicBlocks (64/128/256/512, stride-2 downsample at stage entry) ->
global average pool -> fc. The reference has no ResNet (its only model is
the 6-conv CNN, FLPyfhelin.py:118-141); this model exists for the north-star
scale-up config.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import functional as Fx
from ..ops.modules import (BatchNorm2d, Conv2dValid, Dense, GlobalAvgPool,
                           MaxPool)


class BasicBlock(nn.Module):
    def __init__(self, cin: int, cout: int, stride: int = 1, gen=None):
        super().__init__()
        self.conv1 = Conv2dValid(cin, cout, k=3, stride=stride, pad=1,
                                 relu=False, bias=False, gen=gen)
        self.bn1 = BatchNorm2d(cout, relu=True)
        self.conv2 = Conv2dValid(cout, cout, k=3, stride=1, pad=1,
                                 relu=False, bias=False, gen=gen)
        self.bn2 = BatchNorm2d(cout, relu=False)
        self.down_conv = None
        if stride != 1 or cin != cout:
            self.down_conv = Conv2dValid(cin, cout, k=1, stride=stride, pad=0,
                                         relu=False, bias=False, gen=gen)
            self.down_bn = BatchNorm2d(cout, relu=False)

    def forward(self, x):
        idn = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        if self.down_conv is not None:
            idn = self.down_bn(self.down_conv(x))
        return Fx.add_relu(out, idn)


class ResNet18(nn.Module):
    def __init__(self, in_shape=(128, 128, 3), n_classes=10, seed: int = 0):
        super().__init__()
        gen = torch.Generator().manual_seed(seed)
        H, W, C = in_shape
        self.stem = Conv2dValid(C, 64, k=7, stride=2, pad=3, relu=False,
                                bias=False, gen=gen)
        self.stem_bn = BatchNorm2d(64, relu=True)
        self.stem_pool = MaxPool(3, 2, 1)
        stages = []
        cin = 64
        for cout, stride in ((64, 1), (128, 2), (256, 2), (512, 2)):
            stages.append(BasicBlock(cin, cout, stride, gen=gen))
            stages.append(BasicBlock(cout, cout, 1, gen=gen))
            cin = cout
        self.stages = nn.ModuleList(stages)
        self.pool = GlobalAvgPool()
        self.fc = Dense(512, n_classes, relu=False, gen=gen)

    def forward(self, x):
        x = self.stem_pool(self.stem_bn(self.stem(x)))
        for blk in self.stages:
            x = blk(x)
        return self.fc(self.pool(x))

    def n_params(self) -> int:
        return sum(p.numel() for p in self.parameters())
