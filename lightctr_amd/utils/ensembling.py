"""Ensembling: majority/weighted voting + AdaBoost — parity with
/root/reference/LightCTR/util/ensembling.h."""

from __future__ import annotations

import math

import torch


class VotingEnsemble:
    """Combines binary classifiers (callables x -> proba) by (weighted)
    probability averaging."""

    def __init__(self, models, weights=None):
        self.models = list(models)
        self.weights = weights or [1.0] * len(self.models)

    def predict_proba(self, *args):
        tot, wsum = None, 0.0
        for m, w in zip(self.models, self.weights):
            p = m(*args) * w
            tot = p if tot is None else tot + p
            wsum += w
        return tot / wsum

    def predict(self, *args):
        return (self.predict_proba(*args) > 0.5).float()


class AdaBoost:
    """Discrete AdaBoost over a weak-learner factory
    fit_weak(X, y, sample_weights) -> callable x -> {0,1}."""

    def __init__(self, fit_weak, n_rounds: int = 10):
        self.fit_weak = fit_weak
        self.n_rounds = n_rounds
        self.learners = []
        self.alphas = []

    def fit(self, X: torch.Tensor, y: torch.Tensor):
        n = X.shape[0]
        w = torch.full((n,), 1.0 / n)
        for _ in range(self.n_rounds):
            h = self.fit_weak(X, y, w)
            pred = h(X)
            err = float((w * (pred != y).float()).sum() / w.sum())
            err = min(max(err, 1e-10), 1 - 1e-10)
            if err >= 0.5:
                break
            alpha = 0.5 * math.log((1 - err) / err)
            self.learners.append(h)
            self.alphas.append(alpha)
            sgn = torch.where(pred == y, -1.0, 1.0)
            w = w * torch.exp(alpha * sgn)
            w = w / w.sum()
        return self

    def predict(self, X: torch.Tensor) -> torch.Tensor:
        score = torch.zeros(X.shape[0])
        for h, a in zip(self.learners, self.alphas):
            score += a * (2 * h(X) - 1)
        return (score > 0).float()
