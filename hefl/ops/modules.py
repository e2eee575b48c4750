"""NHWC layer modules built on hefl.ops.functional.

Parameters are fp32 masters (Adam runs in fp32); the GPU compute path casts
weights/activations to bf16 inside the autograd functions. Initialization
mirrors Keras defaults (glorot_uniform kernels, zero bias), matching the
reference model factory (FLPyfhelin.py:118-141).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from . import functional as Fx


def _glorot_uniform(shape, fan_in, fan_out, gen=None):
    limit = math.sqrt(6.0 / (fan_in + fan_out))
    return (torch.rand(shape, generator=gen) * 2 - 1) * limit


class Conv2dValid(nn.Module):
    """Conv2d, NHWC, 'valid' padding, square kernel/stride, optional fused ReLU."""

    def __init__(self, cin: int, cout: int, k: int = 3, stride: int = 1,
                 relu: bool = True, bias: bool = True, pad: int = 0, gen=None):
        super().__init__()
        self.cin, self.cout, self.k, self.stride, self.relu = cin, cout, k, stride, relu
        self.pad = pad
        fan_in, fan_out = cin * k * k, cout * k * k
        w = _glorot_uniform((cout, k, k, cin), fan_in, fan_out, gen)  # [K,R,S,C]
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(cout)) if bias else None

    def out_hw(self, h: int, w: int):
        return ((h + 2 * self.pad - self.k) // self.stride + 1,
                (w + 2 * self.pad - self.k) // self.stride + 1)

    def forward(self, x):
        return Fx.conv2d(x, self.weight, self.bias, self.stride, self.relu,
                         self.pad)


class MaxPool2x2(nn.Module):
    def forward(self, x):
        return Fx.maxpool2x2(x)


class Flatten(nn.Module):
    def forward(self, x):
        return x.reshape(x.shape[0], -1)


class Dense(nn.Module):
    def __init__(self, cin: int, cout: int, relu: bool = False, bias: bool = True, gen=None):
        super().__init__()
        w = _glorot_uniform((cout, cin), cin, cout, gen)  # [O,I]
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(cout)) if bias else None
        self.relu = relu

    def forward(self, x):
        return Fx.linear(x, self.weight, self.bias, self.relu)
