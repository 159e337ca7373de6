"""RNN — LSTM + additive attention + FC head, manual BPTT.

Capability parity with the reference RNN stack
(/root/reference/LightCTR/train/unit/lstm_unit.h: 4-gate LSTM with full
sequence history and BPTT :152-277; train/unit/attention_unit.h: softmax
additive attention with an inner 2-layer FC scorer :40-75 and exact
backward :77-118; train/train_rnn_algo.h: 28-step LSTM over image rows ->
attention -> FC classifier).

Rebuilt batched: gates for the whole batch are one [B, 4H] GEMM per step
(reference lstm_unit.h:324-334 Matrix::Multiply hot site). On GPU the
gate/attention GEMMs run through the IN-TREE MFMA bf16 kernels
(ops/csrc/gemm_kernels.hip, fp32 master weights, bf16 operands, fused
bias); on CPU (and for the exact-vs-autograd tests) the fp32 torch
matmul path is kept. BPTT is the exact manual reverse pass, and updates
are per-tensor Adagrad (the reference uses 12 AdagradUpdater_Num
instances; here the same optimizer over the packed tensors).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch


class _Adagrad:
    def __init__(self, lr=0.05, eps=1e-8):
        self.lr, self.eps = lr, eps
        self.state = {}

    def step(self, params_grads):
        for name, p, g in params_grads:
            n = self.state.setdefault(name, torch.zeros_like(p))
            n += g * g
            p -= self.lr * g / (n + self.eps).sqrt()


def _hip_gemm_ok(device) -> bool:
    if torch.device(device).type != "cuda":
        return False
    from ..ops._extension import has_hip_ops

    return has_hip_ops()


def _gemm_bf16(A, Bst, bias, M, N, K, transA=0, transB=0):
    """C = op(A) @ op(Bst)^T via the in-tree MFMA GEMM (bf16 operands,
    fp32 accumulate/out)."""
    from ..ops._extension import require_hip_ops

    ops = require_hip_ops()
    return ops.gemm_bf16(A.to(torch.bfloat16).contiguous(),
                         Bst.to(torch.bfloat16).contiguous(), bias,
                         M, N, K, transA, transB, 0, False)


class LSTMUnit:
    """Batched 4-gate LSTM with stored sequence history + exact BPTT."""

    def __init__(self, in_dim: int, hidden: int, seed=0, device="cpu",
                 clip: float = 15.0, backend: str | None = None):
        # elementwise delta clipping at +-clip during BPTT (the reference
        # LSTM's error_clip_threshold = 15, lstm_unit.h:56,177-179)
        self.clip = clip
        self.D, self.H = in_dim, hidden
        g = torch.Generator().manual_seed(seed)
        s = (1.0 / (in_dim + hidden)) ** 0.5
        self.W = (torch.randn(4 * hidden, in_dim + hidden, generator=g)
                  * s).to(device)
        self.b = torch.zeros(4 * hidden, device=device)
        self.b[hidden:2 * hidden] = 1.0  # forget-gate bias
        self.device = device
        # "hip_bf16" routes the gate GEMMs through the in-tree MFMA
        # kernels; "torch" is the fp32 exact path (CPU + autograd tests)
        self.backend = backend or (
            "hip_bf16" if _hip_gemm_ok(device) else "torch")

    def forward(self, xs: torch.Tensor, train=True):
        """xs: [B, T, D]. Returns hs [B, T, H]."""
        B, T, D = xs.shape
        H = self.H
        h = torch.zeros(B, H, device=xs.device)
        c = torch.zeros(B, H, device=xs.device)
        cache = []
        hs = []
        hip = self.backend == "hip_bf16"
        Wbf = self.W.to(torch.bfloat16) if hip else None
        for t in range(T):
            z = torch.cat([xs[:, t, :], h], dim=1)  # [B, D+H]
            if hip:
                a = _gemm_bf16(z, Wbf, self.b, B, 4 * H, D + H)
            else:
                a = z @ self.W.t() + self.b
            i = torch.sigmoid(a[:, :H])
            f = torch.sigmoid(a[:, H:2 * H])
            g = torch.tanh(a[:, 2 * H:3 * H])
            o = torch.sigmoid(a[:, 3 * H:])
            c = f * c + i * g
            tc = torch.tanh(c)
            h = o * tc
            hs.append(h)
            if train:
                cache.append((z, i, f, g, o, c, tc))
        if train:
            self._cache = cache
            self._xs_shape = xs.shape
        return torch.stack(hs, dim=1)

    def backward(self, dhs: torch.Tensor):
        """dhs: [B, T, H] grads on each step's h. Returns (dxs, dW, db)."""
        B, T, H = dhs.shape
        D = self.D
        dW = torch.zeros_like(self.W)
        db = torch.zeros_like(self.b)
        dxs = torch.zeros(B, T, D, device=dhs.device)
        dh_next = torch.zeros(B, H, device=dhs.device)
        dc_next = torch.zeros(B, H, device=dhs.device)
        hip = self.backend == "hip_bf16"
        Wbf = self.W.to(torch.bfloat16) if hip else None
        for t in reversed(range(T)):
            z, i, f, g, o, c, tc = self._cache[t]
            c_prev = self._cache[t - 1][5] if t > 0 else torch.zeros_like(c)
            dh = dhs[:, t, :] + dh_next
            if self.clip > 0.0:
                dh = torch.clamp(dh, -self.clip, self.clip)
            do = dh * tc
            dc = dh * o * (1 - tc * tc) + dc_next
            di = dc * g
            df = dc * c_prev
            dg = dc * i
            dc_next = dc * f
            da = torch.cat([di * i * (1 - i), df * f * (1 - f),
                            dg * (1 - g * g), do * o * (1 - o)], dim=1)
            if hip:
                # dW = da^T @ z: [4H, D+H]; dz = da @ W: [B, D+H]
                dW += _gemm_bf16(da, z, None, 4 * H, self.D + H, B, 1, 1)
                dz = _gemm_bf16(da, Wbf, None, B, self.D + H, 4 * H, 0, 1)
            else:
                dW += da.t() @ z
                dz = da @ self.W
            db += da.sum(dim=0)
            dxs[:, t, :] = dz[:, :D]
            dh_next = dz[:, D:]
        return dxs, dW, db


class AttentionUnit:
    """Additive attention: score_t = v^T tanh(Wa h_t + ba); softmax over
    steps; context = sum alpha_t h_t (reference attention_unit.h)."""

    def __init__(self, hidden: int, attn_dim: int = 32, seed=0,
                 device="cpu", backend: str | None = None):
        g = torch.Generator().manual_seed(seed + 5)
        self.Wa = (torch.randn(attn_dim, hidden, generator=g)
                   * (1.0 / hidden) ** 0.5).to(device)
        self.ba = torch.zeros(attn_dim, device=device)
        self.v = (torch.randn(attn_dim, generator=g)
                  * (1.0 / attn_dim) ** 0.5).to(device)
        self.backend = backend or (
            "hip_bf16" if _hip_gemm_ok(device) else "torch")

    def forward(self, hs: torch.Tensor, train=True):
        """hs: [B, T, H] -> context [B, H]."""
        B, T, H = hs.shape
        A = self.Wa.shape[0]
        if self.backend == "hip_bf16":
            pre = _gemm_bf16(hs.reshape(B * T, H), self.Wa, self.ba,
                             B * T, A, H).view(B, T, A)
            u = torch.tanh(pre)
        else:
            u = torch.tanh(hs @ self.Wa.t() + self.ba)  # [B, T, A]
        scores = u @ self.v  # [B, T]
        alpha = torch.softmax(scores, dim=1)
        ctx = (alpha.unsqueeze(2) * hs).sum(dim=1)
        if train:
            self._cache = (hs, u, alpha)
        return ctx

    def backward(self, dctx: torch.Tensor):
        hs, u, alpha = self._cache
        B, T, H = hs.shape
        dalpha = torch.einsum("bh,bth->bt", dctx, hs)
        dhs = alpha.unsqueeze(2) * dctx.unsqueeze(1)
        # softmax backward
        s = (dalpha * alpha).sum(dim=1, keepdim=True)
        dscore = alpha * (dalpha - s)  # [B, T]
        dv = torch.einsum("bt,bta->a", dscore, u)
        du = dscore.unsqueeze(2) * self.v.view(1, 1, -1)
        dpre = du * (1 - u * u)  # [B, T, A]
        A = self.Wa.shape[0]
        if self.backend == "hip_bf16":
            d2 = dpre.reshape(B * T, A)
            h2 = hs.reshape(B * T, H)
            dWa = _gemm_bf16(d2, h2, None, A, H, B * T, 1, 1)
            dhs = dhs + _gemm_bf16(d2, self.Wa, None, B * T, H, A,
                                   0, 1).view(B, T, H)
        else:
            dWa = torch.einsum("bta,bth->ah", dpre, hs)
            dhs = dhs + dpre @ self.Wa
        dba = dpre.sum(dim=(0, 1))
        return dhs, dWa, dba, dv


@dataclass
class RNNHyper:
    in_dim: int = 28
    seq_len: int = 28
    hidden: int = 64
    attn_dim: int = 32
    n_classes: int = 10
    lr: float = 0.05
    seed: int = 1234


class RNNModel:
    """seq-of-rows classifier (reference train_rnn_algo.h:34-45)."""

    def __init__(self, hyper: RNNHyper, device: str = "cpu"):
        self.h = hyper
        self.device = torch.device(device)
        self.lstm = LSTMUnit(hyper.in_dim, hyper.hidden, seed=hyper.seed,
                             device=device)
        self.attn = AttentionUnit(hyper.hidden, hyper.attn_dim,
                                  seed=hyper.seed, device=device)
        g = torch.Generator().manual_seed(hyper.seed + 9)
        self.Wo = (torch.randn(hyper.n_classes, hyper.hidden, generator=g)
                   * (1.0 / hyper.hidden) ** 0.5).to(device)
        self.bo = torch.zeros(hyper.n_classes, device=device)
        self.opt = _Adagrad(lr=hyper.lr)

    def forward(self, xs, train=True):
        hs = self.lstm.forward(xs, train=train)
        ctx = self.attn.forward(hs, train=train)
        if train:
            self._ctx = ctx
        return ctx @ self.Wo.t() + self.bo

    def train_step(self, xs, y) -> float:
        B = xs.shape[0]
        logits = self.forward(xs, train=True)
        p = torch.softmax(logits, dim=1)
        yk = torch.nn.functional.one_hot(y.long(),
                                         self.h.n_classes).float()
        loss = float(torch.nn.functional.cross_entropy(logits, y.long()))
        dlogits = (p - yk) / B
        dWo = dlogits.t() @ self._ctx
        dbo = dlogits.sum(dim=0)
        dctx = dlogits @ self.Wo
        dhs, dWa, dba, dv = self.attn.backward(dctx)
        dxs, dW, db = self.lstm.backward(dhs)
        self.opt.step([
            ("Wo", self.Wo, dWo), ("bo", self.bo, dbo),
            ("Wa", self.attn.Wa, dWa), ("ba", self.attn.ba, dba),
            ("v", self.attn.v, dv),
            ("W", self.lstm.W, dW), ("b", self.lstm.b, db),
        ])
        return loss

    def predict_proba(self, xs):
        return torch.softmax(self.forward(xs, train=False), dim=1)

    def save(self, path):
        torch.save({"W": self.lstm.W, "b": self.lstm.b,
                    "Wa": self.attn.Wa, "ba": self.attn.ba,
                    "v": self.attn.v, "Wo": self.Wo, "bo": self.bo,
                    "hyper": self.h.__dict__}, path)

    def load(self, path):
        d = torch.load(path, map_location=self.device, weights_only=True)
        self.lstm.W.copy_(d["W"]); self.lstm.b.copy_(d["b"])
        self.attn.Wa.copy_(d["Wa"]); self.attn.ba.copy_(d["ba"])
        self.attn.v.copy_(d["v"])
        self.Wo.copy_(d["Wo"]); self.bo.copy_(d["bo"])
