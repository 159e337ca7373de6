"""Updater family + activation set tests."""

import pytest
import torch

from lightctr_amd.utils.updaters import (activate, activate_backward,
                                         make_updater)


@pytest.mark.parametrize("name", ["sgd", "adagrad", "rmsprop", "adadelta",
                                  "adam", "ftrl"])
def test_updater_reduces_quadratic(name):
    """Each updater should minimize f(w) = ||w - target||^2 /2."""
    torch.manual_seed(0)
    target = torch.tensor([1.0, -2.0, 3.0])
    w = torch.zeros(3)
    kw = {"lr": 0.1} if name in ("sgd", "adagrad", "rmsprop") else {}
    if name == "adam":
        kw = {"lr": 0.05}
    if name == "ftrl":
        kw = {"alpha": 0.5, "l1": 0.0, "l2": 1e-6}
    upd = make_updater(name, **kw)
    f0 = float(((w - target) ** 2).sum())
    iters = 3000 if name == "adadelta" else 300  # adadelta warms up slowly
    for _ in range(iters):
        upd.update(w, w - target)
    f1 = float(((w - target) ** 2).sum())
    assert f1 < f0 * 0.1, (name, f0, f1)


def test_adam_matches_torch_optim():
    torch.manual_seed(1)
    w_ref = torch.randn(10, requires_grad=True)
    w = w_ref.detach().clone()
    opt = torch.optim.Adam([w_ref], lr=0.01)
    upd = make_updater("adam", lr=0.01)
    for _ in range(20):
        g = torch.randn(10)
        w_ref.grad = g.clone()
        opt.step()
        upd.update(w, g)
    assert torch.allclose(w, w_ref.detach(), atol=1e-5)


@pytest.mark.parametrize("kind", ["identity", "sigmoid", "tanh", "relu",
                                  "softplus", "softmax"])
def test_activation_backward_matches_autograd(kind):
    g = torch.Generator().manual_seed(2)
    x = torch.randn(6, 5, generator=g).requires_grad_(True)
    y = activate(x, kind)
    dy = torch.randn(6, 5, generator=g)
    y.backward(dy)
    dx = activate_backward(y.detach(), dy, kind)
    assert torch.allclose(dx, x.grad, atol=1e-5), kind


def test_binary_sigmoid():
    x = torch.tensor([-1.0, 0.5, 2.0])
    y = activate(x, "binary_sigmoid")
    assert torch.equal(y, torch.tensor([0.0, 1.0, 1.0]))
    # straight-through estimator passes the gradient
    dy = torch.ones(3)
    assert torch.equal(activate_backward(y, dy, "binary_sigmoid"), dy)
