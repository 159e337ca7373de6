"""Dense MLP built on the hand-written MFMA bf16 GEMM.

Replaces the reference's Fully_Conn_Layer chain
(/root/reference/LightCTR/train/layer/fullyconnLayer.h) in CDNA4 form:
  * GPU: bf16 MFMA GEMM with fused bias+activation epilogue
    (ops/csrc/gemm_kernels.hip), fp32 master weights, bf16 + bf16^T operand
    mirrors refreshed by the fused dense optimizer kernels
  * CPU: the same math in fp32 torch (test oracle)
  * optimizers: Adam (reference momentumUpdater.h:113-215 semantics) and
    Adagrad (reference default), fused on GPU
  * dropout on layer outputs (reference __global_sparse_rate semantics,
    fullyconnLayer.h:199-201) via a torch-generated mask.
"""

from __future__ import annotations

import math

import torch

from ..ops._extension import require_hip_ops

_ACTS = {"none": 0, "relu": 1, "sigmoid": 2}


class DenseLayer:
    def __init__(self, in_dim: int, out_dim: int, act: str = "relu",
                 optimizer: str = "adam", lr: float = 1e-3, l2: float = 0.0,
                 dropout: float = 0.0, seed: int = 0, device: str = "cpu",
                 clip: float = 0.0):
        self.in_dim, self.out_dim = in_dim, out_dim
        self.act = act
        self.act_id = _ACTS[act]
        self.optimizer = optimizer
        self.lr, self.l2 = lr, l2
        self.dropout = dropout
        # elementwise delta clipping to +-clip before the grads are formed
        # (the reference FC/LSTM error_clip_threshold=15 semantics,
        # fullyconnLayer.h:130, matrix.h:152-159); 0 disables
        self.clip = clip
        self.device = torch.device(device)
        g = torch.Generator().manual_seed(seed)
        # He init like the reference FC layer's scaled Gaussian
        std = math.sqrt(2.0 / in_dim)
        self.W = (torch.randn(out_dim, in_dim, generator=g) * std).to(self.device)
        self.b = torch.zeros(out_dim, device=self.device)
        self._gpu = self.device.type == "cuda"
        if self._gpu:
            require_hip_ops()
            self.Wbf = self.W.to(torch.bfloat16).contiguous()
            self.Wtbf = self.W.t().contiguous().to(torch.bfloat16)
        if optimizer == "adam":
            self.m = torch.zeros_like(self.W)
            self.v = torch.zeros_like(self.W)
            self.mb = torch.zeros_like(self.b)
            self.vb = torch.zeros_like(self.b)
            self.beta1, self.beta2, self.eps = 0.9, 0.999, 1e-8
            self.step_count = 0
        else:  # adagrad
            self.n = torch.zeros_like(self.W)
            self.nb = torch.zeros_like(self.b)
            self.eps = 1e-8

    # ---- GPU path (bf16 MFMA) ----
    def forward_gpu(self, x_bf: torch.Tensor, train: bool = True):
        ops = require_hip_ops()
        B = x_bf.shape[0]
        y, y_bf = ops.gemm_bf16_full(x_bf, self.Wbf, self.b, B, self.out_dim,
                                     self.in_dim, 0, 0, self.act_id)
        if train:
            if self.dropout > 0.0:
                mask = (torch.rand_like(y) >= self.dropout).float() / (
                    1.0 - self.dropout)
                y = y * mask
                y_bf = ops.to_bf16(y)
                self._mask = mask
            self._x_bf = x_bf
            self._y = y
        return y, y_bf

    def backward_gpu(self, dy: torch.Tensor):
        ops = require_hip_ops()
        if self.dropout > 0.0:
            dy = dy * self._mask
        if self.clip > 0.0:
            dy = torch.clamp(dy, -self.clip, self.clip)
        dZ, dZbf = ops.act_backward(dy.contiguous(), self._y, self.act_id)
        self._db = ops.colsum(dZ)
        B = dZ.shape[0]
        # dW[out,in] = dZ^T @ x      (A transposed-storage, B row-major)
        self._dW = ops.gemm_bf16(dZbf, self._x_bf, None, self.out_dim,
                                 self.in_dim, B, 1, 1, 0, False)
        # dx[B,in] = dZ @ W          (fast path via the W^T mirror)
        dx = ops.gemm_bf16(dZbf, self.Wtbf, None, B, self.in_dim,
                           self.out_dim, 0, 0, 0, False)
        return dx

    # ---- CPU path (fp32 oracle) ----
    def forward_cpu(self, x: torch.Tensor, train: bool = True):
        y = x @ self.W.t() + self.b
        if self.act == "relu":
            y = torch.relu(y)
        elif self.act == "sigmoid":
            y = torch.sigmoid(torch.clamp(y, -16, 16))
        if train:
            if self.dropout > 0.0:
                mask = (torch.rand_like(y) >= self.dropout).float() / (
                    1.0 - self.dropout)
                y = y * mask
                self._mask = mask
            self._x = x
            self._y = y
        return y, y

    def backward_cpu(self, dy: torch.Tensor):
        if self.dropout > 0.0:
            dy = dy * self._mask
        if self.clip > 0.0:
            dy = torch.clamp(dy, -self.clip, self.clip)
        if self.act == "relu":
            dZ = dy * (self._y > 0).float()
        elif self.act == "sigmoid":
            dZ = dy * self._y * (1 - self._y)
        else:
            dZ = dy
        self._db = dZ.sum(dim=0)
        self._dW = dZ.t() @ self._x
        return dZ @ self.W

    def forward(self, x, train: bool = True):
        return self.forward_gpu(x, train) if self._gpu else self.forward_cpu(x, train)

    def backward(self, dy):
        return self.backward_gpu(dy) if self._gpu else self.backward_cpu(dy)

    def apply_grads(self):
        if self.optimizer == "adam":
            self.step_count += 1
            if self._gpu:
                ops = require_hip_ops()
                ops.dense_adam(self.W, self._dW, self.m, self.v, self.Wbf,
                               self.Wtbf, self.lr, self.beta1, self.beta2,
                               self.eps, self.step_count, self.l2)
                ops.dense_adam(self.b, self._db, self.mb, self.vb, None, None,
                               self.lr, self.beta1, self.beta2, self.eps,
                               self.step_count, 0.0)
            else:
                for p, gr, m, v in ((self.W, self._dW, self.m, self.v),
                                    (self.b, self._db, self.mb, self.vb)):
                    g = gr + (self.l2 * p if p is self.W else 0)
                    m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
                    v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
                    bc1 = 1 - self.beta1 ** self.step_count
                    bc2 = 1 - self.beta2 ** self.step_count
                    p.sub_(self.lr * (m / bc1) / ((v / bc2).sqrt() + self.eps))
        else:
            if self._gpu:
                ops = require_hip_ops()
                ops.dense_adagrad(self.W, self._dW, self.n, self.Wbf,
                                  self.Wtbf, self.lr, self.eps, self.l2)
                ops.dense_adagrad(self.b, self._db, self.nb, None, None,
                                  self.lr, self.eps, 0.0)
            else:
                for p, gr, n in ((self.W, self._dW, self.n),
                                 (self.b, self._db, self.nb)):
                    g = gr + (self.l2 * p if p is self.W else 0)
                    n.add_(g * g)
                    p.sub_(self.lr * g / (n + self.eps).sqrt())

    def state_dict(self):
        d = {"W": self.W, "b": self.b}
        if self.optimizer == "adam":
            d.update(m=self.m, v=self.v, mb=self.mb, vb=self.vb,
                     step=torch.tensor(self.step_count))
        else:
            d.update(n=self.n, nb=self.nb)
        return d

    def load_state_dict(self, d):
        self.W.copy_(d["W"])
        self.b.copy_(d["b"])
        if self.optimizer == "adam":
            self.m.copy_(d["m"]); self.v.copy_(d["v"])
            self.mb.copy_(d["mb"]); self.vb.copy_(d["vb"])
            self.step_count = int(d["step"])
        else:
            self.n.copy_(d["n"]); self.nb.copy_(d["nb"])
        if self._gpu:
            self.Wbf.copy_(self.W.to(torch.bfloat16))
            self.Wtbf.copy_(self.W.t().contiguous().to(torch.bfloat16))


class MLP:
    """Stack of DenseLayers with a scalar head (last layer act='none')."""

    def __init__(self, dims, acts=None, optimizer="adam", lr=1e-3,
                 l2: float = 0.0, dropout: float = 0.0, seed=0, device="cpu",
                 clip: float = 0.0):
        n = len(dims) - 1
        acts = acts or (["relu"] * (n - 1) + ["none"])
        self.layers = [
            DenseLayer(dims[i], dims[i + 1], acts[i], optimizer, lr, l2,
                       dropout if i < n - 1 else 0.0, seed + i, device,
                       clip=clip)
            for i in range(n)
        ]
        self._gpu = self.layers[0]._gpu

    def forward(self, x, train=True):
        """x: bf16 [B,in] on GPU, fp32 on CPU. Returns fp32 [B, out_last]."""
        y = None
        for la in self.layers:
            y, x = la.forward(x, train)  # next layer consumes bf16 mirror
        return y

    def backward(self, dy):
        for la in reversed(self.layers):
            dy = la.backward(dy)
        return dy

    def apply_grads(self):
        for la in self.layers:
            la.apply_grads()

    def state_dict(self):
        return {f"layer{i}": la.state_dict()
                for i, la in enumerate(self.layers)}

    def load_state_dict(self, d):
        for i, la in enumerate(self.layers):
            la.load_state_dict(d[f"layer{i}"])
