"""Statistical utilities: normal CDF/inverse, significance helpers, random
draws — parity with /root/reference/LightCTR/util/{significance.h,random.h}
(erf/CDF/inverse-CDF; Box-Muller Gauss, shuffle-pick-K)."""

from __future__ import annotations

import math

import torch


def normal_cdf(x):
    if isinstance(x, torch.Tensor):
        return 0.5 * (1 + torch.erf(x / math.sqrt(2)))
    return 0.5 * (1 + math.erf(x / math.sqrt(2)))


def normal_cdf_inv(p):
    if isinstance(p, torch.Tensor):
        return math.sqrt(2) * torch.erfinv(2 * p - 1)
    return math.sqrt(2) * torch.erfinv(torch.tensor(2 * p - 1)).item()


def z_test(mean_a, mean_b, var_a, var_b, n_a, n_b):
    """Two-sample z statistic + two-sided p-value."""
    se = math.sqrt(var_a / n_a + var_b / n_b)
    if se == 0:
        return 0.0, 1.0
    z = (mean_a - mean_b) / se
    p = 2 * (1 - normal_cdf(abs(z)))
    return z, p


def gaussian(shape, mean=0.0, std=1.0, seed=None, device="cpu"):
    """Box-Muller-equivalent Gaussian draws (torch native)."""
    g = torch.Generator().manual_seed(seed) if seed is not None else None
    return (torch.randn(shape, generator=g) * std + mean).to(device)


def shuffle_pick_k(n: int, k: int, seed=None) -> torch.Tensor:
    g = torch.Generator().manual_seed(seed) if seed is not None else None
    return torch.randperm(n, generator=g)[:k]
