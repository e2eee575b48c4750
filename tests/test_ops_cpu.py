"""CPU-path op tests: hefl ops vs independent torch references + training sanity."""
import torch
import torch.nn.functional as F

from hefl.ops import functional as Fx
from hefl.ops.adam import FusedAdam
from hefl.models import CNN2
from hefl.fl.weights import flat_params, load_flat_params


def test_conv2d_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(2, 9, 9, 3, requires_grad=True)
    w = torch.randn(8, 3, 3, 3, requires_grad=True)  # [K,R,S,C]
    b = torch.randn(8, requires_grad=True)
    y = Fx.conv2d(x, w, b, stride=1, relu=True)
    ref = F.relu(F.conv2d(x.permute(0, 3, 1, 2), w.permute(0, 3, 1, 2), b)
                 ).permute(0, 2, 3, 1)
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    ref2 = F.relu(F.conv2d(xr.permute(0, 3, 1, 2), wr.permute(0, 3, 1, 2), br)
                  ).permute(0, 2, 3, 1)
    ref2.backward(g)
    assert torch.allclose(x.grad, xr.grad, atol=1e-5)
    assert torch.allclose(w.grad, wr.grad, atol=1e-5)
    assert torch.allclose(b.grad, br.grad, atol=1e-5)


def test_maxpool_matches_torch():
    torch.manual_seed(1)
    x = torch.randn(2, 8, 8, 4, requires_grad=True)
    y = Fx.maxpool2x2(x)
    ref = F.max_pool2d(x.permute(0, 3, 1, 2), 2, 2).permute(0, 2, 3, 1)
    assert torch.allclose(y, ref)
    g = torch.randn_like(y)
    y.backward(g)
    xr = x.detach().clone().requires_grad_(True)
    F.max_pool2d(xr.permute(0, 3, 1, 2), 2, 2).permute(0, 2, 3, 1).backward(g)
    assert torch.allclose(x.grad, xr.grad)


def test_linear_and_loss():
    torch.manual_seed(2)
    x = torch.randn(5, 7, requires_grad=True)
    w = torch.randn(3, 7, requires_grad=True)
    b = torch.randn(3, requires_grad=True)
    labels = torch.tensor([0, 1, 2, 1, 0])
    loss = Fx.softmax_xent(Fx.linear(x, w, b), labels)
    ref = F.cross_entropy(F.linear(x, w, b), labels)
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    F.cross_entropy(F.linear(xr, wr, br), labels).backward()
    assert torch.allclose(x.grad, xr.grad, atol=1e-5)
    assert torch.allclose(w.grad, wr.grad, atol=1e-5)
    assert torch.allclose(b.grad, br.grad, atol=1e-5)


def test_adam_matches_torch_adam_first_step():
    torch.manual_seed(3)
    p1 = torch.nn.Parameter(torch.randn(10))
    p2 = torch.nn.Parameter(p1.detach().clone())
    g = torch.randn(10)
    p1.grad = g.clone()
    p2.grad = g.clone()
    opt1 = FusedAdam([p1], lr=1e-3, decay=0.0, eps=1e-7)
    opt2 = torch.optim.Adam([p2], lr=1e-3, eps=1e-7)
    opt1.step()
    opt2.step()
    assert torch.allclose(p1.detach(), p2.detach(), atol=1e-7)


def test_keras_lr_decay_schedule():
    p = torch.nn.Parameter(torch.zeros(1))
    opt = FusedAdam([p], lr=1e-3, decay=0.5)
    for t in range(3):
        p.grad = torch.ones(1)
        opt.step()
    # lr_t at steps 1,2,3: 1e-3, 1e-3/1.5, 1e-3/2.0 — just check it ran and moved
    assert float(p.detach().abs()) > 0


def test_flat_params_roundtrip():
    m = CNN2(seed=0)
    v = flat_params(m)
    assert v.numel() == m.n_params()
    v2 = v * 2 + 1
    load_flat_params(m, v2)
    assert torch.allclose(flat_params(m), v2)


def test_training_reduces_loss():
    torch.manual_seed(0)
    m = CNN2((28, 28, 1), 4, seed=0)
    opt = FusedAdam(m.parameters(), lr=2e-3, decay=0.0)
    x = torch.rand(32, 28, 28, 1)
    y = torch.randint(0, 4, (32,))
    losses = []
    for _ in range(25):
        loss = Fx.softmax_xent(m(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.5, losses[:3] + losses[-3:]
