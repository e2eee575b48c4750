"""GPU numerics tests: HIP CNN kernels vs plain PyTorch fp32 CPU references.

bf16 compute => tolerances sized to bf16 rounding (rel ~1e-2 on
accumulated results).
"""
import pytest
import torch
import torch.nn.functional as F

from hefl.ops import functional as Fx
from hefl.ops.adam import FusedAdam

pytestmark = pytest.mark.gpu


def _close(a, b, atol=None, rel=3e-2):
    """Max-norm check scaled to the reference tensor: bf16-rounded inputs
    accumulated in fp32 stay within a few percent of the fp32 reference."""
    a = a.detach().float().cpu()
    b = b.detach().float().cpu()
    scale = b.abs().max().item() + 1e-8
    err = (a - b).abs().max().item()
    bound = max(atol if atol is not None else 0.0, rel * scale)
    assert err < bound, f"max abs err {err:.4g} vs scale {scale:.4g} (bound {bound:.4g})"


@pytest.mark.parametrize("shape", [
    # (N, H, W, C, K, ksize, stride, pad)
    (4, 28, 28, 1, 16, 3, 1, 0),     # cnn2 conv1
    (4, 13, 13, 16, 32, 3, 1, 0),    # cnn2 conv2
    (2, 32, 32, 3, 6, 5, 1, 0),      # lenet5 conv1
    (2, 16, 16, 32, 64, 3, 2, 1),    # resnet-style strided+padded
    (2, 32, 32, 3, 32, 3, 1, 0),     # RefCNN6 conv1 shape class (C=3 -> pad8)
    (2, 34, 34, 3, 64, 7, 2, 3),     # ResNet stem 7x7 s2 (C=3 -> pad8)
    (2, 28, 28, 1, 32, 3, 1, 0),     # cnn4 conv1 (C=1 -> pad8)
])
def test_conv2d_fwd_bwd(shape):
    N, H, W, C, K, ks, stride, pad = shape
    torch.manual_seed(0)
    x = torch.randn(N, H, W, C)
    w = torch.randn(K, ks, ks, C) * 0.2
    b = torch.randn(K) * 0.1

    # CPU fp32 reference through the same autograd op
    xc = x.clone().requires_grad_(True)
    wc = w.clone().requires_grad_(True)
    bc = b.clone().requires_grad_(True)
    # relu=False for the gradient check: near-zero pre-activations flip the
    # relu mask under bf16 rounding, injecting O(1) one-element diffs into
    # accumulated grads that say nothing about the GEMM kernels. The fused
    # relu path is covered by the fwd check below + the training tests.
    yc = Fx.conv2d(xc, wc, bc, stride=stride, relu=False, pad=pad)
    g = torch.randn_like(yc)
    yc.backward(g)

    xg = x.to("cuda", torch.bfloat16).requires_grad_(True)
    wg = w.cuda().requires_grad_(True)
    bg = b.cuda().requires_grad_(True)
    yg = Fx.conv2d(xg, wg, bg, stride=stride, relu=False, pad=pad)
    yg.backward(g.to("cuda", torch.bfloat16))

    _close(yg, yc)
    _close(wg.grad, wc.grad, rel=5e-2)
    _close(bg.grad, bc.grad, rel=5e-2)
    _close(xg.grad, xc.grad, rel=5e-2)
    # fused-relu forward parity
    yr = Fx.conv2d(xg.detach(), wg.detach(), bg.detach(), stride=stride,
                   relu=True, pad=pad)
    _close(yr, torch.relu(yc))


def test_linear_fwd_bwd():
    torch.manual_seed(1)
    M, K, N = 32, 800, 64
    x = torch.randn(M, K)
    w = torch.randn(N, K) * 0.05
    b = torch.randn(N) * 0.1
    xc = x.clone().requires_grad_(True)
    wc = w.clone().requires_grad_(True)
    bc = b.clone().requires_grad_(True)
    yc = Fx.linear(xc, wc, bc, relu=False)
    g = torch.randn_like(yc)
    yc.backward(g)

    xg = x.to("cuda", torch.bfloat16).requires_grad_(True)
    wg = w.cuda().requires_grad_(True)
    bg = b.cuda().requires_grad_(True)
    yg = Fx.linear(xg, wg, bg, relu=False)
    yg.backward(g.to("cuda", torch.bfloat16))
    _close(yg, yc)
    _close(wg.grad, wc.grad, rel=5e-2)
    _close(xg.grad, xc.grad, rel=5e-2)
    _close(bg.grad, bc.grad, rel=5e-2)


@pytest.mark.parametrize("hw", [14, 13])  # 13: odd tail rows must get 0 grad
def test_maxpool_gpu(hw):
    torch.manual_seed(2)
    # quantize to bf16 first so CPU and GPU see identical values (otherwise
    # near-ties argmax differently and the backward scatter lands elsewhere)
    x = torch.randn(3, hw, hw, 8).to(torch.bfloat16).float()
    xc = x.clone().requires_grad_(True)
    yc = Fx.maxpool2x2(xc)
    g = torch.randn_like(yc)
    yc.backward(g)

    xg = x.to("cuda", torch.bfloat16).requires_grad_(True)
    yg = Fx.maxpool2x2(xg)
    yg.backward(g.to("cuda", torch.bfloat16))
    _close(yg, yc)
    _close(xg.grad, xc.grad)


def test_softmax_xent_gpu():
    torch.manual_seed(3)
    M, C = 64, 10
    logits = torch.randn(M, C) * 3
    labels = torch.randint(0, C, (M,))
    lc = logits.clone().requires_grad_(True)
    loss_c = Fx.softmax_xent(lc, labels)
    loss_c.backward()

    lg = logits.to("cuda", torch.bfloat16).requires_grad_(True)
    loss_g = Fx.softmax_xent(lg, labels.cuda())
    loss_g.backward()
    assert abs(loss_g.item() - loss_c.item()) < 0.02
    _close(lg.grad, lc.grad)


def test_fused_adam_gpu():
    torch.manual_seed(4)
    p0 = torch.randn(1000)
    g0 = torch.randn(1000)

    pc = torch.nn.Parameter(p0.clone())
    oc = FusedAdam([pc], lr=1e-3, decay=1e-4)
    pg = torch.nn.Parameter(p0.clone().cuda())
    og = FusedAdam([pg], lr=1e-3, decay=1e-4)
    for _ in range(5):
        pc.grad = g0.clone()
        pg.grad = g0.clone().cuda()
        oc.step()
        og.step()
    _close(pg.data, pc.data, rel=1e-4)


def test_model_step_gpu_matches_cpu_direction():
    """One full fwd+bwd+Adam step of CNN2 on GPU: loss decreases over steps
    and weights stay close to the CPU fp32 trajectory after 1 step."""
    from hefl.models import CNN2
    from hefl.ops.functional import softmax_xent

    torch.manual_seed(5)
    x = torch.randn(16, 28, 28, 1)
    y = torch.randint(0, 10, (16,))

    mc = CNN2((28, 28, 1), 10, seed=0)
    oc = FusedAdam(mc.parameters(), lr=1e-3)
    lc = softmax_xent(mc(x), y)
    lc.backward()
    oc.step()

    mg = CNN2((28, 28, 1), 10, seed=0).cuda()
    og = FusedAdam(mg.parameters(), lr=1e-3)
    lg = softmax_xent(mg(x.to("cuda", torch.bfloat16)), y.cuda())
    lg.backward()
    og.step()

    assert abs(lg.item() - lc.item()) < 0.05
    for p_c, p_g in zip(mc.parameters(), mg.parameters()):
        _close(p_g, p_c, atol=3e-3)


def test_training_reduces_loss_gpu():
    from hefl.models import CNN2
    from hefl.ops.functional import softmax_xent

    torch.manual_seed(6)
    m = CNN2((28, 28, 1), 10, seed=1).cuda()
    opt = FusedAdam(m.parameters(), lr=2e-3)
    x = torch.randn(64, 28, 28, 1, device="cuda", dtype=torch.bfloat16)
    y = torch.randint(0, 10, (64,), device="cuda")
    losses = []
    for _ in range(30):
        loss = softmax_xent(m(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.5, losses[::5]


def test_graphed_client_training_converges():
    """LocalClient with hipGraphs ON and a shard whose last batch is PARTIAL
    (two captured graphs share Adam's device-side schedule buffers — a
    re-created buffer would leave graph #1 reading freed memory and silently
    stop learning; regression test for that exact bug)."""
    from hefl.config import preset
    from hefl.fl.client import LocalClient

    cfg = preset("config2")
    cfg.fl.n_clients = 1
    cfg.fl.samples_per_client = 80  # 2 full batches + 1 partial (16)
    cfg.train.hip_graphs = True
    c = LocalClient(cfg, 0, device="cuda:0")
    assert c.use_graphs
    first = c.local_train(epochs=1)
    for _ in range(40):
        last = c.local_train(epochs=1)
    # whole-epoch graph: all 3 steps (2 full + 1 partial batch) in ONE graph
    ent = c._ep_ent
    assert ent["steps"] == 3 and ent["n"] == 80
    assert last.train_loss < first.train_loss * 0.5, (first.train_loss,
                                                      last.train_loss)
    w = c.get_weights()
    assert not torch.isnan(w).any()


@pytest.mark.gpu
def test_graphed_per_step_path_with_callbacks():
    """The callbacks path uses per-step graphs (one per batch shape) with
    Adam applied as direct launches post-replay; both captured graphs share
    Adam's device-side schedule buffers (regression: a re-created buffer
    left graph #1 reading freed memory and silently stopped learning)."""
    from hefl.config import preset
    from hefl.fl.callbacks import EarlyStopping
    from hefl.fl.client import LocalClient

    cfg = preset("config2")
    cfg.fl.n_clients = 1
    cfg.fl.samples_per_client = 80  # 2 full batches + 1 partial (16)
    cfg.train.hip_graphs = True
    c = LocalClient(cfg, 0, device="cuda:0")
    cb = EarlyStopping(c.model, patience=10_000)
    first = c.local_train(epochs=1, callbacks=[cb])
    for _ in range(40):
        last = c.local_train(epochs=1, callbacks=[cb])
    assert len(c._graphs) == 2, "expected full + partial batch graphs"
    assert last.train_loss < first.train_loss * 0.5, (first.train_loss,
                                                      last.train_loss)
    assert not torch.isnan(c.get_weights()).any()


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    (4, 28, 28, 1, 16, 3),   # cnn2 block1
    (4, 13, 13, 16, 32, 3),  # cnn2 block2 (odd H/W: floor pooling)
    (2, 32, 32, 3, 6, 5),    # lenet5 block1 (K=6: scalar fallback path)
    (2, 30, 30, 8, 64, 3),   # K=64 octet 2x2-block path
])
def test_conv_relu_pool_fused_matches_composed(shape):
    """Fused trunk block (one pool+relu+bias backward kernel) against the
    COMPOSED GPU ops on identical bf16 inputs: same forward kernels run in
    both, so relu-gate and pool-argmax decisions are identical and the only
    difference is the fused backward — grads must match near-exactly.
    (CPU-vs-GPU parity is covered by the per-op tests + training tests;
    comparing this block against fp32 CPU would re-test bf16 tie flips.)"""
    N, H, W, C, K, ks = shape
    torch.manual_seed(0)
    x = torch.randn(N, H, W, C, device="cuda", dtype=torch.bfloat16)
    w = (torch.randn(K, ks, ks, C, device="cuda") * 0.1)
    b = torch.randn(K, device="cuda") * 0.05

    def run(fused):
        xd = x.clone().requires_grad_(True)
        wd = torch.nn.Parameter(w.clone())
        bd = torch.nn.Parameter(b.clone())
        if fused:
            y = Fx.conv_relu_pool(xd, wd, bd)
        else:
            y = Fx.maxpool2x2(Fx.conv2d(xd, wd, bd, 1, True, 0))
        y.float().pow(2).sum().backward()
        return y, xd.grad, wd.grad, bd.grad

    yf, dxf, dwf, dbf = run(True)
    yc, dxc, dwc, dbc = run(False)
    _close(yf, yc, rel=1e-5)
    _close(dxf, dxc, rel=1e-3)
    _close(dwf, dwc, rel=1e-3)
    _close(dbf, dbc, rel=1e-3)


@pytest.mark.parametrize("shape", [
    # (N, H, W, C, K, pad): 3x3 s1, C%32==0 -> direct tiled kernel
    (4, 20, 20, 32, 64, 0),
    (4, 21, 21, 32, 32, 1),   # odd output + pad-1 halo
    (2, 10, 18, 64, 64, 1),   # 8x16 tile variant + C slab loop
    (2, 33, 18, 64, 128, 0),  # tail tiles both dims, two Kout tiles
])
def test_conv_tile3_matches_glds(shape):
    """Direct tiled 3x3 kernel vs the implicit-GEMM path on identical bf16
    inputs (HEFL_TILE3=2 forces tiled dispatch on these small grids)."""
    import os
    import hefl
    C_ = hefl.load_extension()
    N, H, W, C, K, pad = shape
    torch.manual_seed(0)
    x = torch.randn(N, H, W, C, device="cuda", dtype=torch.bfloat16)
    w = (torch.randn(K, 3, 3, C, device="cuda") * 0.1).to(torch.bfloat16)
    b = torch.randn(K, device="cuda") * 0.1
    os.environ["HEFL_TILE3"] = "2"
    try:
        y_tile = C_.conv2d_fwd(x, w, b, 1, True, pad)
    finally:
        os.environ["HEFL_TILE3"] = "0"
    y_ref = C_.conv2d_fwd(x, w, b, 1, True, pad)
    os.environ.pop("HEFL_TILE3", None)
    assert torch.equal(y_tile, y_ref) or \
        (y_tile.float() - y_ref.float()).abs().max().item() < 1e-2


def test_dense_head2_fused_matches_composed():
    """Single-launch dense-head backward vs the composed linear ops on
    identical bf16 inputs/weights (cnn2's exact head shape M=32/partial
    M=16, K=800, N1=64, N2=10)."""
    torch.manual_seed(7)
    for M in (32, 16):
        K, N1, N2 = 800, 64, 10
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w1 = (torch.randn(N1, K, device="cuda") * 0.05)
        b1 = torch.randn(N1, device="cuda") * 0.1
        w2 = (torch.randn(N2, N1, device="cuda") * 0.1)
        b2 = torch.randn(N2, device="cuda") * 0.1

        def run(fused):
            xd = x.clone().requires_grad_(True)
            p = [torch.nn.Parameter(t.clone()) for t in (w1, b1, w2, b2)]
            if fused:
                y = Fx.dense_head2(xd, *p)
            else:
                y = Fx.linear(Fx.linear(xd, p[0], p[1], relu=True),
                              p[2], p[3], relu=False)
            y.float().pow(2).sum().backward()
            return y, xd.grad, [q.grad for q in p]

        yf, dxf, gf = run(True)
        yc, dxc, gc = run(False)
        _close(yf, yc, rel=1e-5)
        _close(dxf, dxc, rel=1e-3)
        for a, b in zip(gf, gc):
            _close(a, b, rel=1e-3)


@pytest.mark.parametrize("shape", [
    # (N, H, W, C, K, pad): dgrad via the tiled kernel (rotated weights)
    (4, 20, 20, 32, 32, 0),
    (4, 21, 21, 32, 64, 1),
    (2, 18, 18, 32, 96, 0),
])
def test_conv_tile3_dgrad_matches_glds(shape):
    """Tiled dgrad (dy conv 180-rotated transposed weights, pad'=2-pad) vs
    the implicit-GEMM dgrad on identical inputs."""
    import os
    import hefl
    C_ = hefl.load_extension()
    N, H, W, C, K, pad = shape
    torch.manual_seed(1)
    OH, OW = H + 2 * pad - 2, W + 2 * pad - 2
    dy = torch.randn(N, OH, OW, K, device="cuda", dtype=torch.bfloat16)
    w = (torch.randn(K, 3, 3, C, device="cuda") * 0.1).to(torch.bfloat16)
    os.environ["HEFL_TILE3"] = "2"
    try:
        dx_tile = C_.conv2d_dgrad(dy, w, 1, H, W, pad)
    finally:
        os.environ["HEFL_TILE3"] = "0"
    dx_ref = C_.conv2d_dgrad(dy, w, 1, H, W, pad)
    os.environ.pop("HEFL_TILE3", None)
    # both paths accumulate f32 but in different orders and round to bf16,
    # so one-ulp flips at |dx| >= 1.28 are expected — relative check like
    # the rest of the suite (a real indexing bug is O(max), far outside it)
    _close(dx_tile, dx_ref, rel=1e-2)


def test_splitk_accumulator_pool_stays_clean():
    """The split-K fp32 accumulator is a pooled buffer that consumer kernels
    consume-and-clear (cnn.hip acc_pool): repeated calls must give identical
    results (a broken clear would leak the previous call's partial sums into
    the next), and match the fp32 reference."""
    torch.manual_seed(3)
    # small-M, deep-K shape: tiles < 256 and ksteps >= 16 force k_chunks > 1
    # in both the fwd and dgrad glds paths
    N, H, W, C, K = 2, 8, 8, 64, 64
    x = torch.randn(N, H, W, C)
    w = torch.randn(K, 3, 3, C) * 0.1
    b = torch.randn(K) * 0.1

    xc = x.clone().requires_grad_(True)
    wc = w.clone().requires_grad_(True)
    bc = b.clone().requires_grad_(True)
    yc = Fx.conv2d(xc, wc, bc, stride=1, relu=False, pad=0)
    g = torch.randn_like(yc)
    yc.backward(g)

    outs = []
    for _ in range(3):
        xg = x.to("cuda", torch.bfloat16).requires_grad_(True)
        wg = w.cuda().requires_grad_(True)
        bg = b.cuda().requires_grad_(True)
        yg = Fx.conv2d(xg, wg, bg, stride=1, relu=False, pad=0)
        yg.backward(g.to("cuda", torch.bfloat16))
        outs.append((yg.detach(), xg.grad))
    # repeat-stable: fp32 atomics reorder (ulp wiggle) but a broken clear
    # would leak the whole previous sum (O(1) relative error)
    for i in (1, 2):
        _close(outs[i][0], outs[0][0], rel=1e-3)
        _close(outs[i][1], outs[0][1], rel=1e-3)
    _close(outs[0][0], yc)
    _close(outs[0][1], xc.grad, rel=5e-2)


def test_generic_maxpool_bwd_pooled_accumulator():
    """maxpool_bwd (generic k/s/p path) scatters into the pooled accumulator;
    repeated calls must match each other and the fp32 reference (3x3 stride-2
    pad-1 pool: overlapping windows + padding exercise untouched elements)."""
    torch.manual_seed(4)
    N, H, W, C = 2, 16, 16, 32
    # quantize to bf16 first: CPU and GPU must see identical values or
    # near-ties argmax differently and the scatter lands elsewhere (same
    # caveat as test_maxpool_gpu)
    x = torch.randn(N, H, W, C).to(torch.bfloat16).float()
    xc = x.clone().requires_grad_(True)
    # NHWC -> NCHW for the torch reference
    yc = F.max_pool2d(xc.permute(0, 3, 1, 2), 3, 2, 1)
    g = torch.randn_like(yc)
    yc.backward(g)

    outs = []
    for _ in range(3):
        xg = x.to("cuda", torch.bfloat16).requires_grad_(True)
        yg = Fx.maxpool(xg, k=3, s=2, p=1)
        yg.backward(g.permute(0, 2, 3, 1).contiguous().to("cuda", torch.bfloat16))
        outs.append((yg.detach(), xg.grad))
    for i in (1, 2):
        _close(outs[i][0], outs[0][0], rel=1e-3)
        _close(outs[i][1], outs[0][1], rel=1e-3)
    _close(outs[0][0], yc.permute(0, 2, 3, 1))
    _close(outs[0][1], xc.grad)  # xc is NHWC, so its grad already is too


@pytest.mark.parametrize("preset_name", ["config3", "config4", "reference"])
def test_graphed_client_converges_other_models(preset_name):
    """Graphed-epoch convergence for the remaining model families (lenet5,
    cnn4). Together with the cnn2 and resnet18 graphed tests this guards
    the fill-skip contract (functional.GRAPH_NO_ZERO + cnn.hip grad_buf)
    on every model the presets name — the capture-pool-reuse hazard that
    motivated grad_buf was allocation-layout (i.e. model) dependent."""
    from hefl.config import preset
    from hefl.fl.client import LocalClient

    cfg = preset(preset_name)
    cfg.fl.n_clients = 1
    cfg.fl.samples_per_client = 64
    cfg.train.batch_size = 16
    cfg.train.hip_graphs = True
    c = LocalClient(cfg, 0, device="cuda:0")
    assert c.use_graphs
    first = c.local_train(epochs=1)
    losses = [c.local_train(epochs=1).train_loss for _ in range(25)]
    tail = sum(losses[-5:]) / 5
    assert tail < first.train_loss * 0.7, (first.train_loss, losses[-5:])
    w = c.get_weights()
    assert not torch.isnan(w).any()


def test_dense_head2_fused_forward_parity():
    """dense_head2_fwdfused (row-parallel fused fwd + composed backward)
    must match the composed linear(relu) -> linear pipeline in forward
    values AND all five gradients (its backward runs the same kernels)."""
    torch.manual_seed(9)
    M, K, N1, N2 = 32, 800, 128, 10
    x = torch.randn(M, K)
    w1 = torch.randn(N1, K) * 0.05
    b1 = torch.randn(N1) * 0.1
    w2 = torch.randn(N2, N1) * 0.1
    b2 = torch.randn(N2) * 0.1
    g = torch.randn(M, N2)

    def run(fn):
        xg = x.to("cuda", torch.bfloat16).requires_grad_(True)
        ps = [t.cuda().requires_grad_(True) for t in (w1, b1, w2, b2)]
        y = fn(xg, *ps)
        y.backward(g.to("cuda", torch.bfloat16))
        return (y.detach(), xg.grad, *[p.grad for p in ps])

    ref = run(lambda xg, a, b, c, d: Fx.linear(Fx.linear(xg, a, b, relu=True),
                                               c, d, relu=False))
    out = run(Fx.dense_head2_fwdfused)
    for a, b_ in zip(out, ref):
        _close(a, b_, rel=2e-2, atol=1e-3)
