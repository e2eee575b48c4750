"""Autograd ops for the hefl CNN stack (NHWC layout).

Every op has two paths:
- CUDA (MI355X): hand-written HIP/CDNA4 kernels from `hefl._C` (MFMA
  implicit-GEMM conv, MFMA GEMM linear, fused pool / softmax-CE). Raises if
  the extension is missing — no silent eager fallback on a GPU box.
- CPU: plain fp32 PyTorch reference of the same op. This is the numerics
  oracle the GPU kernels are tested against (tests/test_gpu_ops.py) and what
  runs in the CPU-only plumbing config (BASELINE.json config #1).

Replaces the TF/Keras native surface the reference leans on
(reference model at FLPyfhelin.py:118-141).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

import hefl


def _C():
    return hefl.load_extension()


def _gpu_dtype(t: torch.Tensor) -> torch.Tensor:
    return t if t.dtype == torch.bfloat16 else t.to(torch.bfloat16)


# ---------------------------------------------------------------------------
# Conv2d (valid padding or explicit pre-pad, square stride), NHWC.
# ---------------------------------------------------------------------------

class _Conv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, stride: int, relu: bool, pad: int):
        # x: [N,H,W,C] (bf16 on GPU / f32 on CPU); w: [K,R,S,C] f32 param;
        # b: [K] f32 or None
        ctx.stride = stride
        ctx.relu = relu
        ctx.pad = pad
        ctx.has_bias = b is not None
        if x.is_cuda:
            wb = _gpu_dtype(w.detach())
            bb = b.detach().float() if b is not None else torch.empty(0, device=x.device)
            y = _C().conv2d_fwd(x.contiguous(), wb.contiguous(), bb, stride, relu, pad)
            ctx.save_for_backward(x, wb, y)
        else:
            xf = x.float()
            # torch reference: NHWC -> NCHW
            xn = xf.permute(0, 3, 1, 2)
            wn = w.permute(0, 3, 1, 2)  # [K,C,R,S]
            yn = F.conv2d(xn, wn, b, stride=stride, padding=pad)
            y = yn.permute(0, 2, 3, 1).contiguous()
            if relu:
                y = F.relu(y)
            ctx.save_for_backward(x, w.detach(), y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        stride, relu, pad = ctx.stride, ctx.relu, ctx.pad
        if dy.is_cuda:
            dy = dy.contiguous()
            if relu:
                dy = _C().relu_bwd(dy, y)
            dx = _C().conv2d_dgrad(dy, w, stride, x.shape[1], x.shape[2], pad) \
                if ctx.needs_input_grad[0] else None
            dw = _C().conv2d_wgrad(dy, x.contiguous(), stride, w.shape[1],
                                   w.shape[2], pad)
            db = _C().bias_grad(dy) if ctx.has_bias else None
        else:
            dy = dy.float()
            if relu:
                dy = dy * (y > 0).float()
            xn = x.float().permute(0, 3, 1, 2)
            wn = w.permute(0, 3, 1, 2)
            dyn = dy.permute(0, 3, 1, 2)
            dx = None
            if ctx.needs_input_grad[0]:
                dxn = torch.nn.grad.conv2d_input(xn.shape, wn, dyn, stride=stride,
                                                 padding=pad)
                dx = dxn.permute(0, 2, 3, 1).contiguous()
            dwn = torch.nn.grad.conv2d_weight(xn, wn.shape, dyn, stride=stride,
                                              padding=pad)
            dw = dwn.permute(0, 2, 3, 1).contiguous()  # [K,R,S,C]
            db = dy.sum(dim=(0, 1, 2)) if ctx.has_bias else None
        return dx, dw, db, None, None, None


def conv2d(x, w, b=None, stride: int = 1, relu: bool = False, pad: int = 0):
    return _Conv2dFn.apply(x, w, b, stride, relu, pad)


# ---------------------------------------------------------------------------
# MaxPool 2x2 stride 2 (the only pooling the reference models use,
# FLPyfhelin.py:121-131), NHWC.
# ---------------------------------------------------------------------------

class _MaxPool2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            y, idx = _C().maxpool2x2_fwd(x)
            ctx.save_for_backward(idx)
            ctx.in_shape = x.shape
        else:
            xf = x.float().permute(0, 3, 1, 2)
            yn, idx = F.max_pool2d(xf, 2, 2, return_indices=True)
            y = yn.permute(0, 2, 3, 1).contiguous()
            ctx.save_for_backward(idx)
            ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        N, H, W, C = ctx.in_shape
        if dy.is_cuda:
            dx = _C().maxpool2x2_bwd(dy.contiguous(), idx, H, W)
        else:
            dyn = dy.float().permute(0, 3, 1, 2)
            dxn = F.max_unpool2d(dyn, idx, 2, 2, output_size=(H, W))
            dx = dxn.permute(0, 2, 3, 1).contiguous()
        return dx


def maxpool2x2(x):
    return _MaxPool2x2Fn.apply(x)


# ---------------------------------------------------------------------------
# Linear (Dense): y = x @ w^T + b, optional fused ReLU.
# ---------------------------------------------------------------------------

class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, relu: bool):
        ctx.relu = relu
        ctx.has_bias = b is not None
        if x.is_cuda:
            wb = _gpu_dtype(w.detach())
            bb = b.detach().float() if b is not None else torch.empty(0, device=x.device)
            y = _C().linear_fwd(x, wb, bb, relu)
            ctx.save_for_backward(x, wb, y)
        else:
            y = F.linear(x.float(), w, b)
            if relu:
                y = F.relu(y)
            ctx.save_for_backward(x, w.detach(), y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        if dy.is_cuda:
            dy = dy.contiguous()
            if ctx.relu:
                dy = _C().relu_bwd(dy, y)
            dx = _C().linear_dgrad(dy, w) if ctx.needs_input_grad[0] else None
            dw = _C().linear_wgrad(dy, x).to(torch.float32)
            db = _C().bias_grad(dy.view(-1, dy.shape[-1])) if ctx.has_bias else None
        else:
            dy = dy.float()
            if ctx.relu:
                dy = dy * (y > 0).float()
            dx = dy @ w if ctx.needs_input_grad[0] else None
            dw = dy.t() @ x.float()
            db = dy.sum(0) if ctx.has_bias else None
        return dx, dw, db, None


def linear(x, w, b=None, relu: bool = False):
    return _LinearFn.apply(x, w, b, relu)


# ---------------------------------------------------------------------------
# Fused softmax + categorical cross-entropy (mean over batch) — the
# reference's loss (FLPyfhelin.py:141).
# ---------------------------------------------------------------------------

class _SoftmaxXentFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        if logits.is_cuda:
            loss, probs = _C().softmax_xent_fwd(logits, labels)
        else:
            lf = logits.float()
            probs = F.softmax(lf, dim=-1)
            loss = F.cross_entropy(lf, labels)
        ctx.save_for_backward(probs, labels)
        ctx.out_dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, dloss):
        probs, labels = ctx.saved_tensors
        M = probs.shape[0]
        if probs.is_cuda:
            # dloss stays a device scalar (no host sync: hipGraph-capturable)
            dl = dloss if torch.is_tensor(dloss) else torch.tensor(
                float(dloss), device=probs.device)
            dlogits = _C().softmax_xent_bwd(probs, labels,
                                            dl.to(probs.device, torch.float32))
            dlogits = dlogits.to(ctx.out_dtype)
        else:
            onehot = F.one_hot(labels, probs.shape[-1]).float()
            dlogits = (probs - onehot) * (dloss / M)
            dlogits = dlogits.to(ctx.out_dtype)
        return dlogits, None


def softmax_xent(logits, labels):
    return _SoftmaxXentFn.apply(logits, labels)
