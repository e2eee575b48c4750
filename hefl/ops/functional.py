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
    # prefer the optimizer-maintained bf16 shadow (no per-step cast kernel)
    sh = getattr(t, "_bf16", None)
    if sh is not None:
        return sh
    return t if t.dtype == torch.bfloat16 else t.to(torch.bfloat16)


def _want_cpad(C: int, Kout: int) -> bool:
    """Channel-pad a C % 8 != 0 conv (stem shapes: RefCNN6/ResNet C=3,
    cnn4 C=1) onto the glds MFMA pipeline. The padded lanes multiply zeros,
    but the pipeline is ~5-10x faster per FLOP than the generic-gather
    fallback at these shapes (profiles/r02_refcnn6_kernel_stats.csv:
    stem fwd 25 TF/s, stem wgrad 10 TF/s on the fallback)."""
    return C % 8 != 0 and Kout > 16 and Kout % 8 == 0


def _pad8(t: torch.Tensor) -> torch.Tensor:
    C = t.shape[-1]
    return _C().pad_channels(t.contiguous(), ((C + 7) // 8) * 8)


import os

# BN atomic no-finalize probe, default OFF: measured -35% on config #5
# (hot-word fp32 atomics from ~512 partial blocks per channel word beat
# the 4.7 us finalize launches they replace).
_BN_ATOMIC_MODE = os.getenv("HEFL_BN_ATOMIC", "0")

# PROBE (default OFF, HEFL_BWD_FORK=1 to enable): fork each conv's wgrad
# onto a side stream DURING GRAPH CAPTURE so the recorded fork/join
# become parallel hipGraph branches and wgrad overlaps the dgrad chain.
# Measured a clear LOSS: config2 38.9 -> 29.1 rounds/s (-25%), reference
# 5.0 -> 4.6 (-8%), config4 flat — the per-conv fork/join event nodes
# (cross-stream graph dependencies) cost more at replay than the overlap
# recovers at the ~5 us per-kernel floor. Numerics-correct (all GPU
# tests pass with it on); kept as the documented A/B for the
# "multi-stream backward" idea from the round-1 roadmap.
_BWD_FORK = os.getenv("HEFL_BWD_FORK", "0") == "1"
_SIDE_STREAMS = {}


def _fork_side(dev):
    s = _SIDE_STREAMS.get(dev)
    if s is None:
        s = torch.cuda.Stream(device=dev)
        _SIDE_STREAMS[dev] = s
    return s

# Epoch-graph capture contract (fl/client.py sets this around warmup +
# capture; HEFL_GRAPH_NO_ZERO=0 disables): while True, conv dw / bias db
# outputs skip their zero-init fill launch — the gkey passed to the C++
# wrappers routes the split-K/scatter atomics into a process-lifetime
# hipMalloc buffer (cnn.hip grad_buf) that autograd steals into p.grad
# and the captured multi-tensor Adam consume-and-clears each step (the
# client zeroes the held buffers once right after capture for the first
# replay). Sound because grad_buf memory never belongs to the caching
# allocator, so nothing allocated during a capture can alias it — an
# earlier design that accumulated into capture-pool tensors diverged on
# resnet18 when a grad reused a block freed earlier in the same capture.
# Channel-padded conv dw is SLICED before the steal, so it keeps its
# fill (gkey=0). Measured: fills were 11.5% of config #2 kernel time;
# config2 36.7 -> 38.4-38.9 rounds/s, reference 4.8 -> 4.9-5.0.
GRAPH_NO_ZERO = False


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
        ctx.cpad = False
        # grad_buf keys: the params' CPython ids — stable for the params'
        # lifetime, one accumulation buffer per layer (cnn.hip grad_buf)
        ctx.wkey = id(w)
        ctx.bkey = id(b) if b is not None else 0
        if x.is_cuda:
            wb = _gpu_dtype(w)
            if _want_cpad(x.shape[-1], w.shape[0]):
                ctx.cpad = True
                ctx.in_C = x.shape[-1]
                x = _pad8(x)
                wb = _pad8(wb)
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
            db = None
            bkey = ctx.bkey if GRAPH_NO_ZERO else 0
            if relu and ctx.has_bias:
                dy, db = _C().relu_bias_bwd(dy, y, gkey=bkey)  # one fused pass
            elif relu:
                dy = _C().relu_bwd(dy, y)
            elif ctx.has_bias:
                db = _C().bias_grad(dy, gkey=bkey)
            wg = ctx.wkey if (GRAPH_NO_ZERO and not ctx.cpad) else 0
            fork = (_BWD_FORK and ctx.needs_input_grad[0]
                    and torch.cuda.is_current_stream_capturing())
            if fork:
                cur = torch.cuda.current_stream()
                side = _fork_side(dy.device)
                ev = torch.cuda.Event()
                ev.record(cur)
                with torch.cuda.stream(side):
                    side.wait_event(ev)
                    dw = _C().conv2d_wgrad(dy, x.contiguous(), stride,
                                           w.shape[1], w.shape[2], pad,
                                           gkey=wg)
                    if ctx.cpad:  # drop the zero-padded channel lanes
                        dw = dw[..., :ctx.in_C].contiguous()
                    ev2 = torch.cuda.Event()
                    ev2.record(side)
                dx = _C().conv2d_dgrad(dy, w, stride, x.shape[1], x.shape[2],
                                       pad)
                cur.wait_event(ev2)
            else:
                dx = _C().conv2d_dgrad(dy, w, stride, x.shape[1],
                                       x.shape[2], pad) \
                    if ctx.needs_input_grad[0] else None
                # channel-padded dw sliced before the steal -> classic path
                dw = _C().conv2d_wgrad(dy, x.contiguous(), stride, w.shape[1],
                                       w.shape[2], pad, gkey=wg)
                if ctx.cpad:
                    dw = dw[..., :ctx.in_C].contiguous()
            if ctx.cpad and dx is not None:
                dx = dx[..., :ctx.in_C].contiguous()
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


class _ConvReluPoolFn(torch.autograd.Function):
    """Fused conv(+bias+ReLU) -> maxpool2x2 trunk block (GPU only).

    Forward launches the same two kernels as the composed ops; the win is in
    backward: ONE pool_relu_bias_bwd kernel produces the gated conv-activation
    gradient AND the bias gradient (replacing maxpool2x2_bwd + relu_bias_bwd,
    i.e. two full passes over the activation-gradient tensor)."""

    @staticmethod
    def forward(ctx, x, w, b, pad: int):
        ctx.pad = pad
        ctx.wkey = id(w)
        ctx.bkey = id(b)
        wb = _gpu_dtype(w)
        ctx.cpad = _want_cpad(x.shape[-1], w.shape[0])
        if ctx.cpad:  # stem shapes ride the glds pipeline on padded channels
            ctx.in_C = x.shape[-1]
            x = _pad8(x)
            wb = _pad8(wb)
        bb = b.detach().float()
        y = _C().conv2d_fwd(x.contiguous(), wb.contiguous(), bb, 1, True, pad)
        ctx.conv_hw = (y.shape[1], y.shape[2])
        p, idx = _C().maxpool2x2_fwd(y)
        # backward gates on the POOLED output p (= y at the argmax; relu ran
        # before pool) — the full pre-pool activation y is not retained
        ctx.save_for_backward(x, wb, p, idx)
        return p

    @staticmethod
    def backward(ctx, dy):
        x, w, p, idx = ctx.saved_tensors
        ch, cw = ctx.conv_hw
        bkey = ctx.bkey if GRAPH_NO_ZERO else 0
        dym, db = _C().pool_relu_bias_bwd(dy.contiguous(), idx, p, ch, cw,
                                          gkey=bkey)
        wg = ctx.wkey if (GRAPH_NO_ZERO and not ctx.cpad) else 0
        fork = (_BWD_FORK and ctx.needs_input_grad[0]
                and torch.cuda.is_current_stream_capturing())
        if fork:
            cur = torch.cuda.current_stream()
            side = _fork_side(dy.device)
            ev = torch.cuda.Event()
            ev.record(cur)
            with torch.cuda.stream(side):
                side.wait_event(ev)
                dw = _C().conv2d_wgrad(dym, x.contiguous(), 1, w.shape[1],
                                       w.shape[2], ctx.pad, gkey=wg)
                if ctx.cpad:
                    dw = dw[..., :ctx.in_C].contiguous()
                ev2 = torch.cuda.Event()
                ev2.record(side)
            dx = _C().conv2d_dgrad(dym, w, 1, x.shape[1], x.shape[2], ctx.pad)
            cur.wait_event(ev2)
        else:
            dx = _C().conv2d_dgrad(dym, w, 1, x.shape[1], x.shape[2],
                                   ctx.pad) \
                if ctx.needs_input_grad[0] else None
            dw = _C().conv2d_wgrad(dym, x.contiguous(), 1, w.shape[1],
                                   w.shape[2], ctx.pad, gkey=wg)
            if ctx.cpad:
                dw = dw[..., :ctx.in_C].contiguous()
        if ctx.cpad and dx is not None:
            dx = dx[..., :ctx.in_C].contiguous()
        return dx, dw, db, None


def conv_relu_pool(x, w, b, pad: int = 0):
    """conv3x3(+bias,+relu) then maxpool2x2. GPU: fused-backward Function;
    CPU: composed reference ops (same autograd semantics)."""
    if x.is_cuda:
        return _ConvReluPoolFn.apply(x, w, b, pad)
    return maxpool2x2(conv2d(x, w, b, 1, True, pad))


# ---------------------------------------------------------------------------
# Linear (Dense): y = x @ w^T + b, optional fused ReLU.
# ---------------------------------------------------------------------------

class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, relu: bool):
        ctx.relu = relu
        ctx.has_bias = b is not None
        ctx.bkey = id(b) if b is not None else 0
        if x.is_cuda:
            wb = _gpu_dtype(w)
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
            db = None
            bkey = ctx.bkey if GRAPH_NO_ZERO else 0
            if ctx.relu and ctx.has_bias:
                dy, db = _C().relu_bias_bwd(dy, y, gkey=bkey)
            elif ctx.relu:
                dy = _C().relu_bwd(dy, y)
            elif ctx.has_bias:
                db = _C().bias_grad(dy.view(-1, dy.shape[-1]), gkey=bkey)
            dx = _C().linear_dgrad(dy, w) if ctx.needs_input_grad[0] else None
            dw = _C().linear_wgrad(dy, x).to(torch.float32)
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


class _DenseHead2Fn(torch.autograd.Function):
    """Dense(relu) -> Dense(logits) with a SINGLE-LAUNCH fused backward
    (dW2/db2, relu-gated dh1, dW1/db1, dx in one workgroup) — the six
    separate small-GEMM/bias launches it replaces each ran at the graph
    replay floor and summed to ~23% of the config #2 step. GPU only; the
    CPU path composes the plain linear ops."""

    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        w1b, w2b = _gpu_dtype(w1), _gpu_dtype(w2)
        xc = x.contiguous()
        h1 = _C().linear_fwd(xc, w1b.contiguous(), b1.detach().float(), True)
        logits = _C().linear_fwd(h1, w2b.contiguous(), b2.detach().float(),
                                 False)
        ctx.save_for_backward(xc, h1, w1b, w2b)
        return logits

    @staticmethod
    def backward(ctx, dlogits):
        x, h1, w1b, w2b = ctx.saved_tensors
        dx, dw1, db1, dw2, db2 = _C().dense_head2_bwd(
            dlogits.contiguous(), x, h1, w1b.contiguous(), w2b.contiguous())
        return dx, dw1, db1, dw2, db2


def dense_head2(x, w1, b1, w2, b2):
    return _DenseHead2Fn.apply(x, w1, b1, w2, b2)


class _DenseHead2FwdFn(torch.autograd.Function):
    """Dense(relu) -> Dense(logits) with a row-parallel FUSED FORWARD
    (one block per sample: h1 staged in LDS feeds the logits layer — no
    cross-block dependency) and the backward composed from the SAME
    kernels the two _LinearFn backwards would run (identical count and
    numerics). Replaces two ~7 us linear_splitk launches per step on
    2-dense heads (cnn2's 800->128->10)."""

    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        w1b, w2b = _gpu_dtype(w1), _gpu_dtype(w2)
        xc = x.contiguous()
        h1, logits = _C().dense_head2_fwd(xc, w1b.contiguous(),
                                          b1.detach().float(),
                                          w2b.contiguous(),
                                          b2.detach().float())
        ctx.save_for_backward(xc, h1, w1b, w2b)
        ctx.b1key = id(b1)
        ctx.b2key = id(b2)
        return logits

    @staticmethod
    def backward(ctx, dlogits):
        x, h1, w1b, w2b = ctx.saved_tensors
        dl = dlogits.contiguous()
        k1 = ctx.b1key if GRAPH_NO_ZERO else 0
        k2 = ctx.b2key if GRAPH_NO_ZERO else 0
        db2 = _C().bias_grad(dl, gkey=k2)
        dh1 = _C().linear_dgrad(dl, w2b)
        dw2 = _C().linear_wgrad(dl, h1)
        dh1, db1 = _C().relu_bias_bwd(dh1, h1, gkey=k1)
        dx = _C().linear_dgrad(dh1, w1b) if ctx.needs_input_grad[0] else None
        dw1 = _C().linear_wgrad(dh1, x)
        return dx, dw1, db1, dw2, db2


def dense_head2_fwdfused(x, w1, b1, w2, b2):
    return _DenseHead2FwdFn.apply(x, w1, b1, w2, b2)


# ---------------------------------------------------------------------------
# Fused softmax + categorical cross-entropy (mean over batch) — the
# reference's loss (FLPyfhelin.py:141).
# ---------------------------------------------------------------------------

class _SoftmaxXentFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, acc_loss=None, acc_correct=None):
        if logits.is_cuda:
            empty = torch.empty(0, device=logits.device)
            loss, probs = _C().softmax_xent_fwd(
                logits.contiguous(), labels,
                acc_loss if acc_loss is not None else empty,
                acc_correct if acc_correct is not None else empty)
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
            dlogits = _C().softmax_xent_bwd(
                probs, labels, dl.to(probs.device, torch.float32),
                ctx.out_dtype == torch.bfloat16)
        else:
            onehot = F.one_hot(labels, probs.shape[-1]).float()
            dlogits = (probs - onehot) * (dloss / M)
            dlogits = dlogits.to(ctx.out_dtype)
        return dlogits, None, None, None


def softmax_xent(logits, labels, acc_loss=None, acc_correct=None):
    """Fused softmax + CE; on GPU optionally accumulates mean-loss and
    correct-count into persistent device buffers (stats fused in-kernel)."""
    return _SoftmaxXentFn.apply(logits, labels, acc_loss, acc_correct)


# ---------------------------------------------------------------------------
# BatchNorm2d (NHWC, per-channel over N*H*W) — ResNet-18 support (config #5).
# ---------------------------------------------------------------------------

class _BatchNorm2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training: bool,
                momentum: float, eps: float, relu: bool, sums=None):
        ctx.relu = relu
        ctx.training = training
        C = x.shape[-1]
        if x.is_cuda:
            if training:
                # running stats update fused into the finalize kernel (the
                # torch mul_/add_ chain was 4 kernels per BN layer per step).
                # Under the epoch-graph capture contract (GRAPH_NO_ZERO),
                # the per-layer `sums` buffer switches both passes to the
                # atomic no-finalize kernels: dgamma/dbeta are accumulated
                # straight into the stolen grads (Adam clears them), and
                # the backward partial clears `sums` for the next replay.
                empty = torch.empty(0, device=x.device)
                # HEFL_BN_ATOMIC: 1 = atomic fwd+bwd (no finalize launches),
                # bwd = classic fwd + atomic bwd (bisect aid), 0 = classic
                mode = _BN_ATOMIC_MODE
                use_atomic = (GRAPH_NO_ZERO and sums is not None
                              and mode != "0")
                ctx.sums = sums if use_atomic else None
                ctx.atomic_fwd = use_atomic and mode != "bwd"
                y, mean, invstd = _C().bn_fwd(
                    x.contiguous(), gamma.detach().float(),
                    beta.detach().float(),
                    running_mean if running_mean is not None else empty,
                    running_var if running_var is not None else empty,
                    eps, momentum, relu,
                    sums if ctx.atomic_fwd else empty)
            else:
                ctx.sums = None
                mean = running_mean
                invstd = (running_var + eps).rsqrt()
                y = _C().bn_apply(x.contiguous(), mean, invstd,
                                  gamma.detach().float(), beta.detach().float(),
                                  relu)
        else:
            ctx.sums = None
            xf = x.float().reshape(-1, C)
            if training:
                mean = xf.mean(0)
                var = xf.var(0, unbiased=False)
                invstd = (var + eps).rsqrt()
            else:
                mean = running_mean
                invstd = (running_var + eps).rsqrt()
            y = ((xf - mean) * invstd * gamma + beta).reshape(x.shape)
            if relu:
                y = F.relu(y)
        if training and running_mean is not None and not x.is_cuda:
            with torch.no_grad():
                M = x.numel() // C
                # torch stores the UNBIASED variance in running_var
                var_b = (1.0 / invstd.double() ** 2 - eps).float()
                var_b = var_b * (M / max(M - 1, 1))
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(var_b, alpha=momentum)
        ctx.save_for_backward(x, gamma.detach(), mean, invstd, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, invstd, y = ctx.saved_tensors
        C = x.shape[-1]
        M = x.numel() // C
        if dy.is_cuda:
            dy = dy.contiguous()
            # relu gate fused into the BN backward kernels (no relu_bwd pass)
            ry = y if ctx.relu else torch.empty(0, device=dy.device)
            fs = ctx.sums if ctx.sums is not None \
                else torch.empty(0, device=dy.device)
            dx, dgamma, dbeta = _C().bn_bwd(dy, x.contiguous(), mean, invstd,
                                            gamma.float(), ctx.training, ry,
                                            fs)
        else:
            dy = dy.float()
            if ctx.relu:
                dy = dy * (y > 0).float()
            dyf = dy.reshape(-1, C)
            xf = x.float().reshape(-1, C)
            xhat = (xf - mean) * invstd
            dgamma = (dyf * xhat).sum(0)
            dbeta = dyf.sum(0)
            if ctx.training:
                dx = (gamma * invstd) * (dyf - dbeta / M - xhat * dgamma / M)
            else:
                dx = (gamma * invstd) * dyf
            dx = dx.reshape(x.shape)
        return dx, dgamma, dbeta, None, None, None, None, None, None, None


def batchnorm2d(x, gamma, beta, running_mean, running_var, training=True,
                momentum=0.1, eps=1e-5, relu=False, sums=None):
    return _BatchNorm2dFn.apply(x, gamma, beta, running_mean, running_var,
                                training, momentum, eps, relu, sums)


# ---------------------------------------------------------------------------
# Generic MaxPool k/s/p (ResNet stem 3x3 s2 p1), global avg pool, add+relu.
# ---------------------------------------------------------------------------

class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k: int, s: int, p: int):
        ctx.ksp = (k, s, p)
        ctx.in_shape = x.shape
        if x.is_cuda:
            y, idx = _C().maxpool_fwd(x.contiguous(), k, s, p)
            ctx.save_for_backward(idx)
        else:
            xf = x.float().permute(0, 3, 1, 2)
            yn, idx = F.max_pool2d(xf, k, s, p, return_indices=True)
            y = yn.permute(0, 2, 3, 1).contiguous()
            ctx.save_for_backward(idx)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        k, s, p = ctx.ksp
        N, H, W, C = ctx.in_shape
        if dy.is_cuda:
            dx = _C().maxpool_bwd(dy.contiguous(), idx, H, W, k, s, p)
        else:
            # scatter-ADD (max_unpool2d overwrites, wrong for overlapping
            # windows like the ResNet stem's 3x3 s2)
            dyn = dy.float().permute(0, 3, 1, 2)
            dxn = torch.zeros(N, C, H * W, dtype=dyn.dtype)
            dxn.scatter_add_(2, idx.reshape(N, C, -1), dyn.reshape(N, C, -1))
            dx = dxn.reshape(N, C, H, W).permute(0, 2, 3, 1).contiguous()
        return dx, None, None, None


def maxpool(x, k: int = 2, s: int = 2, p: int = 0):
    if k == 2 and s == 2 and p == 0:
        return maxpool2x2(x)
    return _MaxPoolFn.apply(x, k, s, p)


class _GlobalAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.in_shape = x.shape
        if x.is_cuda:
            return _C().avgpool_global_fwd(x.contiguous())
        return x.float().mean(dim=(1, 2))

    @staticmethod
    def backward(ctx, dy):
        N, H, W, C = ctx.in_shape
        if dy.is_cuda:
            return _C().avgpool_global_bwd(dy.contiguous(), H, W)
        return (dy / (H * W)).reshape(N, 1, 1, C).expand(N, H, W, C).contiguous()


def global_avgpool(x):
    return _GlobalAvgPoolFn.apply(x)


class _AddReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        if a.is_cuda:
            y = _C().add_relu(a.contiguous(), b.contiguous())
        else:
            y = F.relu(a.float() + b.float())
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        if dy.is_cuda:
            d = _C().relu_bwd(dy.contiguous(), y)
        else:
            d = dy * (y > 0).float()
        return d, d


def add_relu(a, b):
    return _AddReluFn.apply(a, b)
