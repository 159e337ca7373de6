"""Sparse logistic regression (the reference's wide/LR part as a
standalone model — distributed_algo_abst.h's wide path and the LR root of
the FM family, fm_algo_abst.h). Implemented on the FM kernel machinery
with the latent factors pinned at zero (V init 0 => pred == sum w*x; V
receives zero gradients through the fused backward because sumVX == 0 and
v == 0), so the full fused step / sorted backward / FTRL-or-Adagrad path
is shared."""

from __future__ import annotations

from .fm import FMHyper, FMModel, FMTrainer


class LRHyper(FMHyper):
    def __init__(self, num_features: int, **kw):
        kw.setdefault("k", 4)
        kw.setdefault("init_sigma", 0.0)  # V == 0 -> pure linear model
        super().__init__(num_features=num_features, **kw)


class LRModel(FMModel):
    def __init__(self, hyper: LRHyper, device: str = "cpu", **kw):
        assert hyper.init_sigma == 0.0
        super().__init__(hyper, device=device, **kw)


class LRTrainer(FMTrainer):
    pass
