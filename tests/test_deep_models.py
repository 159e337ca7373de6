"""VAE, CNN, RNN(LSTM+attention) tests — CPU convergence + grad checks."""

import pytest
import torch

from lightctr_amd.models.cnn import CNNHyper, CNNModel
from lightctr_amd.models.rnn import LSTMUnit, RNNHyper, RNNModel
from lightctr_amd.models.vae import VAEHyper, VAEModel


def _digits(n=512, seed=0, size=16):
    """Synthetic 'digit' images: class = which quadrant is bright."""
    g = torch.Generator().manual_seed(seed)
    X = torch.rand(n, 1, size, size, generator=g) * 0.2
    y = torch.randint(0, 4, (n,), generator=g)
    hs = size // 2
    for i in range(n):
        q = int(y[i])
        r, c = (q // 2) * hs, (q % 2) * hs
        X[i, 0, r:r + hs, c:c + hs] += 0.8
    return X, y


def test_vae_reconstruction_improves():
    g = torch.Generator().manual_seed(1)
    X = (torch.rand(256, 64, generator=g) > 0.7).float()
    m = VAEModel(VAEHyper(in_dim=64, hidden=32, z_dim=8, lr=3e-3))
    first = m.train_step(X)
    for _ in range(120):
        last = m.train_step(X)
    assert last["rec"] < first["rec"] * 0.7, (first, last)
    r = m.reconstruct(X)
    assert r.shape == X.shape
    gen = m.generate(5)
    assert gen.shape == (5, 64)
    assert (gen >= 0).all() and (gen <= 1).all()


def test_cnn_learns_quadrants():
    X, y = _digits(n=512, size=16)
    m = CNNModel(CNNHyper(in_shape=(1, 16, 16), n_classes=4, lr=3e-3))
    for ep in range(12):
        perm = torch.randperm(512)
        for s in range(0, 512, 64):
            idx = perm[s:s + 64]
            m.train_step(X[idx], y[idx])
    p = m.predict_proba(X)
    acc = (p.argmax(dim=1) == y).float().mean()
    assert acc > 0.85, float(acc)


def test_cnn_save_load(tmp_path):
    X, y = _digits(n=64, size=16)
    m = CNNModel(CNNHyper(in_shape=(1, 16, 16), n_classes=4))
    m.train_step(X, y)
    p1 = m.predict_proba(X)
    path = str(tmp_path / "cnn.pt")
    m.save(path)
    m2 = CNNModel(CNNHyper(in_shape=(1, 16, 16), n_classes=4))
    m2.load(path)
    assert torch.allclose(m2.predict_proba(X), p1, atol=1e-5)


def test_lstm_backward_matches_autograd():
    torch.manual_seed(3)
    B, T, D, H = 4, 6, 5, 7
    unit = LSTMUnit(D, H, seed=2)
    xs = torch.randn(B, T, D)
    dhs = torch.randn(B, T, H) * 0.1

    # autograd reference with the same parameters
    W = unit.W.clone().requires_grad_(True)
    b = unit.b.clone().requires_grad_(True)
    xr = xs.clone().requires_grad_(True)
    h = torch.zeros(B, H)
    c = torch.zeros(B, H)
    hs_ref = []
    for t in range(T):
        z = torch.cat([xr[:, t, :], h], dim=1)
        a = z @ W.t() + b
        i = torch.sigmoid(a[:, :H])
        f = torch.sigmoid(a[:, H:2 * H])
        gg = torch.tanh(a[:, 2 * H:3 * H])
        o = torch.sigmoid(a[:, 3 * H:])
        c = f * c + i * gg
        h = o * torch.tanh(c)
        hs_ref.append(h)
    hs_ref = torch.stack(hs_ref, dim=1)
    hs_ref.backward(dhs)

    hs = unit.forward(xs)
    assert torch.allclose(hs, hs_ref.detach(), atol=1e-6)
    dxs, dW, db = unit.backward(dhs)
    assert torch.allclose(dW, W.grad, atol=1e-5)
    assert torch.allclose(db, b.grad, atol=1e-5)
    assert torch.allclose(dxs, xr.grad, atol=1e-5)


def test_rnn_learns_sequences():
    """Class = which third of the sequence carries a bright stripe."""
    g = torch.Generator().manual_seed(4)
    B, T, D = 600, 12, 8
    X = torch.rand(B, T, D, generator=g) * 0.2
    y = torch.randint(0, 3, (B,), generator=g)
    for i in range(B):
        t0 = int(y[i]) * 4
        X[i, t0:t0 + 4, :] += 0.8
    m = RNNModel(RNNHyper(in_dim=D, seq_len=T, hidden=32, attn_dim=16,
                          n_classes=3, lr=0.1))
    for ep in range(15):
        perm = torch.randperm(B)
        for s in range(0, B, 64):
            idx = perm[s:s + 64]
            m.train_step(X[idx], y[idx])
    acc = (m.predict_proba(X).argmax(dim=1) == y).float().mean()
    assert acc > 0.9, float(acc)


@pytest.mark.gpu
def test_cnn_gpu_learns():
    X, y = _digits(n=512, size=16)
    m = CNNModel(CNNHyper(in_shape=(1, 16, 16), n_classes=4, lr=3e-3),
                 device="cuda:0")
    X, y = X.cuda(), y.cuda()
    for ep in range(12):
        perm = torch.randperm(512)
        for s in range(0, 512, 64):
            idx = perm[s:s + 64]
            m.train_step(X[idx], y[idx])
    acc = (m.predict_proba(X).argmax(dim=1) == y).float().mean()
    assert acc > 0.8, float(acc)


@pytest.mark.gpu
def test_vae_gpu_trains():
    g = torch.Generator().manual_seed(1)
    X = (torch.rand(256, 64, generator=g) > 0.7).float().cuda()
    m = VAEModel(VAEHyper(in_dim=64, hidden=32, z_dim=8, lr=3e-3),
                 device="cuda:0")
    first = m.train_step(X)
    for _ in range(100):
        last = m.train_step(X)
    assert last["rec"] < first["rec"] * 0.8, (first, last)
