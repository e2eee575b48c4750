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


class ConvPool(Conv2dValid):
    """Conv2dValid(relu=True) + MaxPool2x2 as one trunk block with a fused
    backward on GPU (one kernel for pool-bwd + ReLU mask + bias grad).
    Same parameters/init stream as Conv2dValid."""

    def __init__(self, cin: int, cout: int, k: int = 3, gen=None):
        super().__init__(cin, cout, k=k, stride=1, relu=True, bias=True,
                         pad=0, gen=gen)

    def out_hw(self, h: int, w: int):
        oh, ow = super().out_hw(h, w)
        return oh // 2, ow // 2

    def forward(self, x):
        return Fx.conv_relu_pool(x, self.weight, self.bias, self.pad)


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


class BatchNorm2d(nn.Module):
    """NHWC batch norm over (N,H,W) per channel, optional fused ReLU.
    fp32 affine params + running stats; bf16 activations on GPU."""

    def __init__(self, c: int, eps: float = 1e-5, momentum: float = 0.1,
                 relu: bool = False):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(c))
        self.bias = nn.Parameter(torch.zeros(c))
        self.register_buffer("running_mean", torch.zeros(c))
        self.register_buffer("running_var", torch.ones(c))
        self.eps, self.momentum, self.relu = eps, momentum, relu
        # [2*c] fp32 raw-sums scratch for the atomic no-finalize BN path
        # (epoch-graph capture only — functional.GRAPH_NO_ZERO). A plain
        # attribute, NOT a buffer: it must stay out of the FedAvg flat
        # vector (weights.py aggregates floating-point buffers) and out of
        # checkpoints. Allocated eagerly on first CUDA forward OUTSIDE a
        # capture (the engine's 3 warmup steps) so the zero-fill is never
        # captured; invariant: zero between steps (bn_bwd_partial<true>
        # block 0 re-zeroes it each backward).
        self._sums = None

    def forward(self, x):
        if x.is_cuda and self._sums is None \
                and not torch.cuda.is_current_stream_capturing():
            self._sums = torch.zeros(2 * self.weight.numel(),
                                     dtype=torch.float32, device=x.device)
        return Fx.batchnorm2d(x, self.weight, self.bias, self.running_mean,
                              self.running_var, self.training, self.momentum,
                              self.eps, self.relu,
                              sums=self._sums if x.is_cuda else None)


class MaxPool(nn.Module):
    def __init__(self, k: int = 2, s: int = 2, p: int = 0):
        super().__init__()
        self.k, self.s, self.p = k, s, p

    def forward(self, x):
        return Fx.maxpool(x, self.k, self.s, self.p)


class GlobalAvgPool(nn.Module):
    def forward(self, x):
        return Fx.global_avgpool(x)
