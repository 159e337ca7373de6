"""Gradient / parameter compression codecs.

Parity with the reference compression stack, GPU-first:
  * Fp16Codec       — fp16 wire values (reference common/float16.h; CDNA4
                      packed v_cvt via torch dtype conversion)
  * QuantileCodec   — int8 (or any width <= 8 bit) quantile compression with
                      uniform / log / normal-CDF tables (reference
                      util/quantile_compress.h:71-148)
  * LowBitCodec     — 1/2-bit sign(-magnitude) quantization (reference
                      util/product_quantizer.h:24-45 lowbit_quantize)
  * ProductQuantizer— PQ: per-subvector k-means -> uint8 codes (reference
                      util/product_quantizer.h:114-197), k-means distance
                      steps run as batched GPU matmuls.

GPU paths use the codec kernels in ops/csrc/codec_kernels.hip; CPU paths
are the torch oracles used by the tests.
"""

from __future__ import annotations

import math

import torch

from ..ops._extension import require_hip_ops


class Fp16Codec:
    @staticmethod
    def encode(x: torch.Tensor) -> torch.Tensor:
        return x.to(torch.float16)

    @staticmethod
    def decode(h: torch.Tensor) -> torch.Tensor:
        return h.to(torch.float32)


def _normal_cdf_inv(p: torch.Tensor) -> torch.Tensor:
    return math.sqrt(2.0) * torch.erfinv(2 * p - 1)


class QuantileCodec:
    """Table-based scalar quantizer. mode: uniform | log | normal."""

    def __init__(self, levels: int = 256, mode: str = "uniform",
                 lo: float = -1.0, hi: float = 1.0, device: str = "cpu"):
        assert levels <= 256
        self.levels = levels
        p = (torch.arange(levels, dtype=torch.float64) + 0.5) / levels
        if mode == "uniform":
            t = lo + (hi - lo) * p
        elif mode == "log":
            # symmetric log spacing: dense near 0 (gradient-shaped)
            half = levels // 2
            mags = torch.logspace(-6, 0, half, dtype=torch.float64) * hi
            t = torch.cat([-mags.flip(0), mags])[:levels]
            t = torch.sort(t).values
        elif mode == "normal":
            t = _normal_cdf_inv(p) * (hi - lo) / 6.0 + (hi + lo) / 2.0
        else:
            raise ValueError(mode)
        self.table = t.float().to(device)

    def encode(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda:
            return require_hip_ops().quantile_encode(x.contiguous(),
                                                     self.table)
        mids = (self.table[:-1] + self.table[1:]) / 2
        return torch.searchsorted(mids, x.contiguous(),
                                  right=False).to(torch.uint8)

    def decode(self, code: torch.Tensor) -> torch.Tensor:
        if code.is_cuda:
            return require_hip_ops().quantile_decode(code.contiguous(),
                                                     self.table)
        return self.table[code.long()]


class LowBitCodec:
    """1-bit (sign * mean|x|) or 2-bit (sign * {lo,hi} magnitude)."""

    def __init__(self, bits: int = 1):
        assert bits in (1, 2)
        self.bits = bits

    def encode(self, x: torch.Tensor):
        absx = x.abs()
        if self.bits == 1:
            scale = float(absx.mean())
            meta = (scale, scale, float("nan"))
            thresh = 0.0
        else:
            thresh = float(absx.mean())
            small = absx < thresh
            lo = float(absx[small].mean()) if small.any() else 0.0
            big = ~small
            hi = float(absx[big].mean()) if big.any() else thresh
            meta = (lo, hi, thresh)
        if x.is_cuda:
            words = require_hip_ops().lowbit_encode(x.contiguous(), thresh,
                                                    self.bits)
        else:
            n = x.numel()
            per = 32 // self.bits
            c = (x.reshape(-1) >= 0).to(torch.int64)
            if self.bits == 2:
                c |= (absx.reshape(-1) >= thresh).to(torch.int64) << 1
            nw = (n * self.bits + 31) // 32
            pad = nw * per - n
            c = torch.cat([c, torch.zeros(pad, dtype=torch.int64)])
            shifts = (torch.arange(per) * self.bits)
            words = (c.view(nw, per) << shifts).sum(dim=1).to(torch.int32)
        return words, meta, x.numel()

    def decode(self, words: torch.Tensor, meta, n: int) -> torch.Tensor:
        lo, hi, _ = meta
        if words.is_cuda:
            return require_hip_ops().lowbit_decode(words, lo, hi, self.bits,
                                                   n)
        per = 32 // self.bits
        w = words.to(torch.int64).view(-1, 1)
        shifts = torch.arange(per) * self.bits
        codes = ((w >> shifts) & ((1 << self.bits) - 1)).reshape(-1)[:n]
        sign = torch.where(codes & 1 > 0, 1.0, -1.0)
        mag = torch.full((n,), lo)
        if self.bits == 2:
            mag = torch.where(codes & 2 > 0, torch.tensor(hi),
                              torch.tensor(lo))
        return sign * mag


class ProductQuantizer:
    """PQ: split D-dim vectors into n_sub subvectors, k-means each to 256
    centroids, code = uint8 per subvector. K-means distance/argmin steps are
    batched matmuls so the GPU path runs on MFMA via torch."""

    def __init__(self, dim: int, n_sub: int = 4, n_centroids: int = 256,
                 iters: int = 10, seed: int = 0):
        assert dim % n_sub == 0
        self.dim, self.n_sub = dim, n_sub
        self.d_sub = dim // n_sub
        self.n_centroids = n_centroids
        self.iters = iters
        self.seed = seed
        self.centroids = None  # [n_sub, n_centroids, d_sub]

    def fit(self, X: torch.Tensor) -> "ProductQuantizer":
        N = X.shape[0]
        g = torch.Generator().manual_seed(self.seed)
        ncb = min(self.n_centroids, N)
        cents = []
        Xs = X.view(N, self.n_sub, self.d_sub)
        for s in range(self.n_sub):
            xs = Xs[:, s, :]
            idx = torch.randperm(N, generator=g)[:ncb].to(X.device)
            C = xs[idx].clone()
            for _ in range(self.iters):
                d = torch.cdist(xs, C)
                assign = d.argmin(dim=1)
                newC = torch.zeros_like(C)
                cnt = torch.zeros(ncb, device=X.device)
                newC.index_add_(0, assign, xs)
                cnt.index_add_(0, assign,
                               torch.ones(N, device=X.device))
                empty = cnt == 0
                cnt = cnt.clamp(min=1)
                newC /= cnt.unsqueeze(1)
                # empty-cluster splitting (reference :138-186): reseed from
                # the most populated cluster with jitter
                if empty.any():
                    big = cnt.argmax()
                    newC[empty] = C[big] + 1e-3 * torch.randn(
                        int(empty.sum()), self.d_sub, generator=g
                    ).to(X.device)
                C = newC
            cents.append(C)
        self.centroids = torch.stack(cents)
        return self

    def encode(self, X: torch.Tensor) -> torch.Tensor:
        N = X.shape[0]
        Xs = X.view(N, self.n_sub, self.d_sub)
        codes = torch.empty(N, self.n_sub, dtype=torch.uint8,
                            device=X.device)
        for s in range(self.n_sub):
            d = torch.cdist(Xs[:, s, :], self.centroids[s])
            codes[:, s] = d.argmin(dim=1).to(torch.uint8)
        return codes

    def decode(self, codes: torch.Tensor) -> torch.Tensor:
        N = codes.shape[0]
        out = torch.empty(N, self.dim, device=codes.device)
        for s in range(self.n_sub):
            out[:, s * self.d_sub:(s + 1) * self.d_sub] = \
                self.centroids[s][codes[:, s].long()]
        return out
