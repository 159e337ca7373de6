"""GBM tests: binary/multiclass learning, NaN handling, histogram parity."""

import pytest
import torch

from lightctr_amd.models.gbm import GBMHyper, GBMModel


def _nonlinear_data(n=2000, d=8, seed=0, with_nan=False):
    g = torch.Generator().manual_seed(seed)
    X = torch.randn(n, d, generator=g)
    y = ((X[:, 0] * X[:, 1] > 0) ^ (X[:, 2] > 0.5)).float()
    if with_nan:
        mask = torch.rand(n, d, generator=g) < 0.1
        X = torch.where(mask, torch.full_like(X, float("nan")), X)
    return X, y


def test_gbm_binary_learns_nonlinear():
    X, y = _nonlinear_data()
    m = GBMModel(GBMHyper(n_rounds=25, max_depth=4, learning_rate=0.3,
                          subsample=0.9, colsample=1.0))
    m.fit(X, y)
    p = m.predict_proba(X)
    acc = ((p > 0.5) == (y > 0.5)).float().mean()
    assert acc > 0.85, float(acc)


def test_gbm_handles_nan():
    X, y = _nonlinear_data(with_nan=True)
    m = GBMModel(GBMHyper(n_rounds=20, max_depth=4))
    m.fit(X, y)
    p = m.predict_proba(X)
    assert torch.isfinite(p).all()
    acc = ((p > 0.5) == (y > 0.5)).float().mean()
    assert acc > 0.7, float(acc)


def test_gbm_multiclass_softmax():
    g = torch.Generator().manual_seed(1)
    X = torch.randn(1500, 6, generator=g)
    y = (X[:, 0] > 0.4).long() + (X[:, 1] > 0.1).long()  # 3 classes
    m = GBMModel(GBMHyper(n_rounds=15, max_depth=4, n_classes=3,
                          colsample=1.0))
    m.fit(X, y.float())
    p = m.predict_proba(X)
    assert p.shape == (1500, 3)
    acc = (p.argmax(dim=1) == y).float().mean()
    assert acc > 0.8, float(acc)


def test_gbm_save_load_roundtrip(tmp_path):
    X, y = _nonlinear_data(n=500)
    m = GBMModel(GBMHyper(n_rounds=5, max_depth=3))
    m.fit(X, y)
    p1 = m.predict_proba(X)
    path = str(tmp_path / "gbm.pt")
    m.save(path)
    m2 = GBMModel(GBMHyper(n_rounds=5, max_depth=3))
    m2.load(path)
    p2 = m2.predict_proba(X)
    assert torch.allclose(p1, p2)


@pytest.mark.gpu
def test_gbm_hist_kernel_parity():
    from lightctr_amd.ops import hip_ops

    g = torch.Generator().manual_seed(2)
    N, D, n_nodes = 5000, 12, 4
    bins = torch.randint(0, 256, (N, D), generator=g,
                         dtype=torch.uint8).cuda()
    grad = torch.randn(N, generator=g).cuda()
    hess = torch.rand(N, generator=g).cuda()
    node = torch.randint(-1, n_nodes, (N,), generator=g,
                         dtype=torch.int32).cuda()
    hist = hip_ops.gbm_hist(bins, grad, hess, node, n_nodes)
    m = GBMModel(GBMHyper(), device="cpu")
    ref = m._hist(bins.cpu(), grad.cpu(), hess.cpu(), node.cpu(), n_nodes)
    assert torch.allclose(hist.cpu(), ref, atol=1e-3, rtol=1e-4), \
        (hist.cpu() - ref).abs().max()


@pytest.mark.gpu
def test_gbm_gpu_learns():
    X, y = _nonlinear_data()
    m = GBMModel(GBMHyper(n_rounds=20, max_depth=4, colsample=1.0),
                 device="cuda:0")
    m.fit(X.cuda(), y.cuda())
    p = m.predict_proba(X.cuda())
    acc = ((p > 0.5) == (y.cuda() > 0.5)).float().mean()
    assert acc > 0.85, float(acc)
