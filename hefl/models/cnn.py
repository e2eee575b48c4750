"""CNN model families (NHWC, valid padding) for the BASELINE.json configs.

- RefCNN6: the reference's exact architecture — 6x[Conv3x3 valid + ReLU +
  MaxPool2x2] with filters 32,32,32,64,64,128 -> Flatten(512) -> Dense 128
  ReLU -> Dense 64 ReLU -> Dense n_classes. 222,722 params at 256x256x3 / 2
  classes (FLPyfhelin.py:118-141; param count verified in tests).
- CNN2: config #2 headline model (2-conv on 28x28).
- LeNet5: config #3 (32x32x3).
- CNN4: config #4 (4-conv on 224x224).

Models output LOGITS; softmax lives in the fused softmax-CE loss (the
reference's softmax head + categorical cross-entropy, FLPyfhelin.py:138,141).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..config import ModelConfig
from ..ops.modules import ConvPool, Dense, Flatten


class _SeqCNN(nn.Module):
    """Conv/pool trunk + dense head, NHWC."""

    def __init__(self, in_shape, conv_filters, conv_k, dense_units, n_classes,
                 seed: int = 0):
        super().__init__()
        gen = torch.Generator().manual_seed(seed)
        H, W, C = in_shape
        layers = []
        cin = C
        for f in conv_filters:
            block = ConvPool(cin, f, k=conv_k, gen=gen)
            layers.append(block)  # fused conv+relu+pool trunk block
            H, W = block.out_hw(H, W)
            cin = f
        self.trunk = nn.ModuleList(layers)
        self.flatten = Flatten()
        feat = H * W * cin
        self.feat_dim = feat
        head = []
        din = feat
        for u in dense_units:
            head.append(Dense(din, u, relu=True, gen=gen))
            din = u
        head.append(Dense(din, n_classes, relu=False, gen=gen))
        self.head = nn.ModuleList(head)
        # single-hidden heads within the fused-backward envelope CAN
        # dispatch dense_head2 on GPU (one kernel instead of six at the
        # replay floor) — measured SLOWER end-to-end (config #2 36.7 ->
        # 32.6 rounds/s: the single-workgroup kernel serializes ~270 MFMAs
        # + strided w1 reads on ONE CU, worse than six well-overlapped
        # launches). Numerics-validated and kept behind HEFL_HEAD2=1 as the
        # documented probe.
        import os
        self._head2_ok = (os.environ.get("HEFL_HEAD2", "0") == "1"
                          and len(head) == 2 and head[0].relu
                          and not head[1].relu and head[0].bias is not None
                          and head[1].bias is not None
                          and n_classes <= 16
                          and head[0].weight.shape[0] <= 128
                          and head[0].weight.shape[0] % 16 == 0
                          and feat % 16 == 0 and feat <= 1024)
        # fused FORWARD only (row-parallel, one block per sample, backward
        # composed from the standard kernels): measured -3% on config #2
        # (37.5 vs 38.8 rounds/s) — at M=32 the kernel is 32 blocks of
        # serial 800-element VALU dot chains, slower than the two MFMA
        # linear_splitk launches it replaces. Numerics-validated
        # (test_dense_head2_fused_forward_parity); probe HEFL_HEAD2F=1.
        self._head2f_ok = (os.environ.get("HEFL_HEAD2F", "0") == "1"
                           and len(head) == 2 and head[0].relu
                           and not head[1].relu and head[0].bias is not None
                           and head[1].bias is not None
                           and n_classes <= 256
                           and head[0].weight.shape[0] <= 256
                           and feat % 8 == 0 and feat <= 4096)

    def forward(self, x):
        for m in self.trunk:
            x = m(x)
        x = self.flatten(x)
        from ..ops import functional as Fx
        if x.is_cuda and self._head2_ok and x.shape[0] <= 32:
            return Fx.dense_head2(x, self.head[0].weight, self.head[0].bias,
                                  self.head[1].weight, self.head[1].bias)
        if x.is_cuda and self._head2f_ok and x.shape[0] <= 4096:
            return Fx.dense_head2_fwdfused(
                x, self.head[0].weight, self.head[0].bias,
                self.head[1].weight, self.head[1].bias)
        for m in self.head:
            x = m(x)
        return x

    def n_params(self) -> int:
        return sum(p.numel() for p in self.parameters())


class RefCNN6(_SeqCNN):
    def __init__(self, in_shape=(256, 256, 3), n_classes=2, seed=0):
        super().__init__(in_shape, (32, 32, 32, 64, 64, 128), 3, (128, 64),
                         n_classes, seed)


class CNN2(_SeqCNN):
    def __init__(self, in_shape=(28, 28, 1), n_classes=10, seed=0):
        super().__init__(in_shape, (16, 32), 3, (64,), n_classes, seed)


class CNN4(_SeqCNN):
    def __init__(self, in_shape=(224, 224, 1), n_classes=2, seed=0):
        super().__init__(in_shape, (32, 64, 64, 128), 3, (128,), n_classes, seed)


class LeNet5(_SeqCNN):
    def __init__(self, in_shape=(32, 32, 3), n_classes=10, seed=0):
        super().__init__(in_shape, (6, 16), 5, (120, 84), n_classes, seed)


def build_model(cfg: ModelConfig, seed: int = 0) -> _SeqCNN:
    cls = {"cnn2": CNN2, "refcnn6": RefCNN6, "cnn4": CNN4, "lenet5": LeNet5}
    if cfg.name == "resnet18":
        from .resnet import ResNet18
        return ResNet18(cfg.in_shape, cfg.n_classes, seed=seed)
    if cfg.name not in cls:
        raise KeyError(f"unknown model {cfg.name}")
    return cls[cfg.name](cfg.in_shape, cfg.n_classes, seed=seed)
