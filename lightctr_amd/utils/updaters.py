"""The full updater family — parity with the reference
/root/reference/LightCTR/util/{gradientUpdater.h,momentumUpdater.h}:
SGD, Adagrad, RMSprop, FTRL-proximal, Adadelta, Adam (numeric & tensor
forms). One torch implementation each, usable on CPU or GPU tensors;
the hot-path fused variants live in the HIP kernels (fm/misc/nn_kernels).
"""

from __future__ import annotations

import torch


class Updater:
    def update(self, p: torch.Tensor, g: torch.Tensor) -> None:
        raise NotImplementedError


class SGD(Updater):
    def __init__(self, lr=0.05):
        self.lr = lr

    def update(self, p, g):
        p.sub_(self.lr * g)


class Adagrad(Updater):
    def __init__(self, lr=0.05, eps=1e-8):
        self.lr, self.eps = lr, eps
        self.n = None

    def update(self, p, g):
        if self.n is None:
            self.n = torch.zeros_like(p)
        self.n.add_(g * g)
        p.sub_(self.lr * g / (self.n + self.eps).sqrt())


class RMSprop(Updater):
    def __init__(self, lr=0.01, rho=0.9, eps=1e-8):
        self.lr, self.rho, self.eps = lr, rho, eps
        self.n = None

    def update(self, p, g):
        if self.n is None:
            self.n = torch.zeros_like(p)
        self.n.mul_(self.rho).add_((1 - self.rho) * g * g)
        p.sub_(self.lr * g / (self.n + self.eps).sqrt())


class Adadelta(Updater):
    def __init__(self, rho=0.95, eps=1e-6):
        self.rho, self.eps = rho, eps
        self.n = None
        self.d = None

    def update(self, p, g):
        if self.n is None:
            self.n = torch.zeros_like(p)
            self.d = torch.zeros_like(p)
        self.n.mul_(self.rho).add_((1 - self.rho) * g * g)
        step = ((self.d + self.eps) / (self.n + self.eps)).sqrt() * g
        self.d.mul_(self.rho).add_((1 - self.rho) * step * step)
        p.sub_(step)


class Adam(Updater):
    def __init__(self, lr=1e-3, beta1=0.9, beta2=0.999, eps=1e-8):
        self.lr, self.b1, self.b2, self.eps = lr, beta1, beta2, eps
        self.m = None
        self.v = None
        self.t = 0

    def update(self, p, g):
        if self.m is None:
            self.m = torch.zeros_like(p)
            self.v = torch.zeros_like(p)
        self.t += 1
        self.m.mul_(self.b1).add_((1 - self.b1) * g)
        self.v.mul_(self.b2).add_((1 - self.b2) * g * g)
        mh = self.m / (1 - self.b1 ** self.t)
        vh = self.v / (1 - self.b2 ** self.t)
        p.sub_(self.lr * mh / (vh.sqrt() + self.eps))


class FTRL(Updater):
    """FTRL-proximal (reference gradientUpdater.h:235-278)."""

    def __init__(self, alpha=0.05, beta=1.0, l1=1e-4, l2=1e-4):
        self.alpha, self.beta, self.l1, self.l2 = alpha, beta, l1, l2
        self.z = None
        self.n = None

    def update(self, p, g):
        if self.z is None:
            self.z = torch.zeros_like(p)
            self.n = torch.zeros_like(p)
        g2 = g * g
        sigma = ((self.n + g2).sqrt() - self.n.sqrt()) / self.alpha
        self.z.add_(g - sigma * p)
        self.n.add_(g2)
        w = torch.where(
            self.z.abs() <= self.l1,
            torch.zeros_like(p),
            -(self.z - torch.sign(self.z) * self.l1)
            / ((self.beta + self.n.sqrt()) / self.alpha + self.l2))
        p.copy_(w)


def make_updater(name: str, **kw) -> Updater:
    return {"sgd": SGD, "adagrad": Adagrad, "rmsprop": RMSprop,
            "adadelta": Adadelta, "adam": Adam, "ftrl": FTRL}[name](**kw)


# ---- activations (reference util/activations.h full set) ----

def activate(x: torch.Tensor, kind: str) -> torch.Tensor:
    if kind == "identity" or kind == "none":
        return x
    if kind == "sigmoid":
        return torch.sigmoid(torch.clamp(x, -16, 16))
    if kind == "tanh":
        return torch.tanh(x)
    if kind == "relu":
        return torch.relu(x)
    if kind == "softplus":
        return torch.nn.functional.softplus(x)
    if kind == "softmax":
        return torch.softmax(x, dim=-1)
    if kind == "binary_sigmoid":  # BNN-style hard threshold
        return (x > 0).float()
    raise ValueError(kind)


def activate_backward(y: torch.Tensor, dy: torch.Tensor,
                      kind: str) -> torch.Tensor:
    """Backward from the OUTPUT y (matching the reference's layer API)."""
    if kind in ("identity", "none", "binary_sigmoid"):
        return dy  # straight-through for the binary case
    if kind == "sigmoid":
        return dy * y * (1 - y)
    if kind == "tanh":
        return dy * (1 - y * y)
    if kind == "relu":
        return dy * (y > 0).float()
    if kind == "softplus":
        return dy * (1 - torch.exp(-y))
    if kind == "softmax":
        s = (dy * y).sum(dim=-1, keepdim=True)
        return y * (dy - s)
    raise ValueError(kind)
