"""CPU tests for the ResNet-support ops (batchnorm, generic maxpool, global
avgpool, fused add+relu) against independent torch references, plus the
ResNet-18 model itself."""
import torch
import torch.nn.functional as F

from hefl.ops import functional as Fx


def test_batchnorm_matches_torch():
    torch.manual_seed(0)
    N, H, W, C = 4, 6, 6, 8
    x = torch.randn(N, H, W, C, requires_grad=True)
    gamma = torch.randn(C, requires_grad=True) * 0.5 + 1
    beta = torch.randn(C, requires_grad=True) * 0.1
    gamma.retain_grad(), beta.retain_grad()
    rm, rv = torch.zeros(C), torch.ones(C)
    y = Fx.batchnorm2d(x, gamma, beta, rm, rv, training=True, momentum=0.1)
    xn = x.detach().clone().requires_grad_(True)
    gn = gamma.detach().clone().requires_grad_(True)
    bn = beta.detach().clone().requires_grad_(True)
    rm2, rv2 = torch.zeros(C), torch.ones(C)
    yn = F.batch_norm(xn.permute(0, 3, 1, 2), rm2, rv2, gn, bn, training=True,
                      momentum=0.1).permute(0, 2, 3, 1)
    assert torch.allclose(y, yn, atol=1e-5)
    assert torch.allclose(rm, rm2, atol=1e-6)
    assert torch.allclose(rv, rv2, atol=1e-4)  # ours stores biased batch var

    g = torch.randn_like(y)
    y.backward(g)
    yn.backward(g)
    assert torch.allclose(x.grad, xn.grad, atol=1e-5)
    assert torch.allclose(gamma.grad, gn.grad, atol=1e-4)
    assert torch.allclose(beta.grad, bn.grad, atol=1e-4)


def test_batchnorm_eval_uses_running_stats():
    torch.manual_seed(1)
    C = 4
    x = torch.randn(2, 5, 5, C)
    gamma, beta = torch.ones(C), torch.zeros(C)
    rm, rv = torch.randn(C) * 0.1, torch.rand(C) + 0.5
    y = Fx.batchnorm2d(x, gamma, beta, rm, rv, training=False)
    ref = (x - rm) * (rv + 1e-5).rsqrt()
    assert torch.allclose(y, ref, atol=1e-5)


def test_maxpool_3x3s2p1_matches_torch():
    torch.manual_seed(2)
    x = torch.randn(2, 9, 9, 4, requires_grad=True)
    y = Fx.maxpool(x, 3, 2, 1)
    xn = x.detach().clone().requires_grad_(True)
    yn = F.max_pool2d(xn.permute(0, 3, 1, 2), 3, 2, 1).permute(0, 2, 3, 1)
    assert torch.allclose(y, yn)
    g = torch.randn_like(y)
    y.backward(g)
    yn.backward(g)
    assert torch.allclose(x.grad, xn.grad)


def test_global_avgpool():
    torch.manual_seed(3)
    x = torch.randn(3, 7, 5, 6, requires_grad=True)
    y = Fx.global_avgpool(x)
    assert torch.allclose(y, x.mean(dim=(1, 2)))
    y.sum().backward()
    assert torch.allclose(x.grad, torch.full_like(x, 1.0 / 35))


def test_add_relu():
    torch.manual_seed(4)
    a = torch.randn(50, requires_grad=True)
    b = torch.randn(50, requires_grad=True)
    y = Fx.add_relu(a, b)
    assert torch.allclose(y, F.relu(a + b))
    g = torch.randn(50)
    y.backward(g)
    mask = (a + b > 0).float()
    assert torch.allclose(a.grad, g * mask)
    assert torch.allclose(b.grad, g * mask)


def test_resnet18_trains():
    from hefl.models import build_model
    from hefl.config import ModelConfig
    from hefl.ops.adam import FusedAdam
    from hefl.ops.functional import softmax_xent

    torch.manual_seed(5)
    m = build_model(ModelConfig("resnet18", (32, 32, 3), 4), seed=0)
    assert m.n_params() > 11e6
    opt = FusedAdam(m.parameters(), lr=1e-3)
    x = torch.randn(4, 32, 32, 3)
    y = torch.randint(0, 4, (4,))
    losses = []
    for _ in range(3):
        loss = softmax_xent(m(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0]


def test_flat_params_includes_buffers():
    from hefl.config import ModelConfig
    from hefl.fl.weights import flat_params, load_flat_params
    from hefl.models import build_model

    m = build_model(ModelConfig("resnet18", (32, 32, 3), 4), seed=0)
    v = flat_params(m)
    n_par = sum(p.numel() for p in m.parameters())
    n_buf = sum(b.numel() for b in m.buffers() if b.is_floating_point())
    assert v.numel() == n_par + n_buf
    m2 = build_model(ModelConfig("resnet18", (32, 32, 3), 4), seed=9)
    load_flat_params(m2, v)
    for a, b in zip(m.buffers(), m2.buffers()):
        if a.is_floating_point():
            assert torch.equal(a, b)
