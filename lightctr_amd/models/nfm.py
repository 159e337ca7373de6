"""NFM (Neural Factorization Machine / Wide&Deep-style) — MI355X-native.

Capability parity with the reference Train_NFM_Algo
(/root/reference/LightCTR/train/train_nfm_algo.{h,cpp}): wide LR term +
bi-interaction vector 0.5*((sum v x)^2 - sum (v x)^2) fed into an MLP.
GPU path: fused CDNA4 bi-interaction kernel -> bf16 MFMA MLP -> sorted
segment-reduce sparse backward + fused sparse optimizer.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops import fm_ref
from ..ops._extension import require_hip_ops, sort_ids
from ..utils.metrics import auc_score
from .mlp import MLP


@dataclass
class NFMHyper:
    num_features: int
    k: int = 16
    hidden: tuple = (64,)
    optimizer: str = "adagrad"  # sparse side: adagrad | ftrl
    mlp_optimizer: str = "adam"
    lr: float = 0.05
    mlp_lr: float = 1e-3
    eps: float = 1e-8
    l2: float = 1e-5
    ftrl_alpha: float = 0.15
    ftrl_beta: float = 1.0
    ftrl_l1: float = 1e-4
    ftrl_l2: float = 1e-4
    ftrl_v: str = "adagrad"  # latent updater under ftrl (see models/fm.py)
    dropout: float = 0.0
    init_sigma: float = 0.01
    seed: int = 1234


class NFMModel:
    def __init__(self, hyper: NFMHyper, device: str = "cpu",
                 max_batch_nnz: int = 1 << 22):
        self.h = hyper
        self.device = torch.device(device)
        F, K = hyper.num_features, hyper.k
        assert K in (4, 8, 16, 32, 64)
        g = torch.Generator().manual_seed(hyper.seed)
        self.W = torch.zeros(F, device=self.device)
        self.V = (torch.randn(F, K, generator=g) * hyper.init_sigma).to(
            self.device)
        self.gradW = torch.zeros_like(self.W)
        self.gradV = torch.zeros_like(self.V)
        self.nW = torch.zeros_like(self.W)
        self.nV = torch.zeros_like(self.V)
        if hyper.optimizer == "ftrl":
            self.zW = torch.zeros_like(self.W)
            self.zV = torch.zeros_like(self.V)
        dims = [K, *hyper.hidden, 1]
        self.mlp = MLP(dims, optimizer=hyper.mlp_optimizer, lr=hyper.mlp_lr,
                       dropout=hyper.dropout, seed=hyper.seed, device=device)
        nwords = (F + 63) // 64
        self.touched = torch.zeros(nwords, dtype=torch.int64,
                                   device=self.device)
        cap = min(F, max_batch_nnz)
        self.uniq = torch.zeros(cap, dtype=torch.int32, device=self.device)
        self.count = torch.zeros(1, dtype=torch.int32, device=self.device)
        self._gpu = self.device.type == "cuda"
        if self._gpu:
            require_hip_ops()

    # ---- forward ----
    def _forward_gpu(self, row_ptr, fids, vals, train=True):
        ops = require_hip_ops()
        wide, sumVX, vec, vec_bf = ops.nfm_forward(row_ptr, fids, vals,
                                                   self.W, self.V)
        deep = self.mlp.forward(vec_bf, train=train)  # [B,1] fp32
        pred = wide + deep[:, 0]
        return pred, (wide, sumVX, vec)

    def _forward_cpu(self, row_ptr, fids, vals, train=True):
        pred_lin, sumVX = fm_ref.fm_forward_ref(row_ptr, fids, vals, self.W,
                                                self.V)
        # fm_forward_ref's pred = wide + 0.5(||sumVX||^2 - sumV2X2) scalar;
        # we need the per-k vector, recompute pieces:
        B = row_ptr.numel() - 1
        rp = row_ptr.long()
        counts = rp[1:] - rp[:-1]
        row_idx = torch.repeat_interleave(torch.arange(B), counts)
        f = fids.long()
        vx = self.V[f] * vals.unsqueeze(1)
        sumV2X2k = torch.zeros(B, self.h.k)
        sumV2X2k.index_add_(0, row_idx, vx * vx)
        vec = 0.5 * (sumVX * sumVX - sumV2X2k)
        wide = torch.zeros(B)
        wide.index_add_(0, row_idx, self.W[f] * vals)
        deep = self.mlp.forward(vec, train=train)
        pred = wide + deep[:, 0]
        return pred, (wide, sumVX, vec)

    def predict_proba(self, row_ptr, fids, vals):
        if self._gpu:
            pred, _ = self._forward_gpu(row_ptr, fids, vals, train=False)
        else:
            pred, _ = self._forward_cpu(row_ptr, fids, vals, train=False)
        return torch.sigmoid(torch.clamp(pred, -16, 16))

    # ---- train step ----
    def train_step(self, row_ptr, fids, vals, labels):
        B = row_ptr.numel() - 1
        scale = 1.0 / B
        if self._gpu:
            ops = require_hip_ops()
            pred, (wide, sumVX, vec) = self._forward_gpu(row_ptr, fids, vals)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            ddeep = self.mlp.backward(dpred.unsqueeze(1))  # [B,K] fp32
            gw, gv = ops.nfm_backward_emit(row_ptr, fids, vals, self.V, sumVX,
                                           ddeep.contiguous(), dpred)
            sorted_fids, perm = sort_ids(fids, self.h.num_features)
            ops.fm_sorted_apply(sorted_fids, perm, gw, gv, self.gradW,
                                self.gradV, self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            live = self.uniq[: min(self.uniq.numel(), fids.numel())]
            if self.h.optimizer == "ftrl":
                ops.fm_ftrl_apply(live, self.count, self.W, self.V,
                                  self.zW, self.nW, self.zV, self.nV,
                                  self.gradW, self.gradV, self.h.ftrl_alpha,
                                  self.h.ftrl_beta, self.h.ftrl_l1,
                                  self.h.ftrl_l2,
                                  1 if self.h.ftrl_v == "adagrad" else 0,
                                  self.h.lr, self.h.eps, self.h.l2)
            else:
                ops.fm_adagrad_apply(live, self.count, self.W, self.V,
                                     self.nW, self.nV, self.gradW, self.gradV,
                                     self.h.lr, self.h.eps, self.h.l2)
            self.mlp.apply_grads()
            return loss
        # CPU oracle path
        pred, (wide, sumVX, vec) = self._forward_cpu(row_ptr, fids, vals)
        loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
        dvec = self.mlp.backward(dpred.unsqueeze(1))
        # sparse grads: gW = dpred*x ; gV[f,k] = dvec[k]*(sumVX[k]-v x)*x
        Bn = row_ptr.numel() - 1
        rp = row_ptr.long()
        counts = rp[1:] - rp[:-1]
        row_idx = torch.repeat_interleave(torch.arange(Bn), counts)
        f = fids.long()
        x = vals
        self.gradW.index_add_(0, f, dpred[row_idx] * x)
        gv = (sumVX[row_idx] - self.V[f] * x.unsqueeze(1)) * x.unsqueeze(1) \
            * dvec[row_idx]
        self.gradV.index_add_(0, f, gv)
        uniq = torch.unique(f).int()
        if self.h.optimizer == "ftrl":
            fm_ref.ftrl_apply_ref(uniq, self.W, self.V, self.zW, self.nW,
                                  self.zV, self.nV, self.gradW, self.gradV,
                                  self.h.ftrl_alpha, self.h.ftrl_beta,
                                  self.h.ftrl_l1, self.h.ftrl_l2,
                                  v_adagrad=self.h.ftrl_v == "adagrad",
                                  v_lr=self.h.lr, v_eps=self.h.eps,
                                  v_l2=self.h.l2)
        else:
            fm_ref.adagrad_apply_ref(uniq, self.W, self.V, self.nW, self.nV,
                                     self.gradW, self.gradV, self.h.lr,
                                     self.h.eps, self.h.l2)
        self.mlp.apply_grads()
        return loss

    def state_dict(self):
        return {"W": self.W, "V": self.V, "nW": self.nW, "nV": self.nV,
                "mlp": self.mlp.state_dict(), "hyper": self.h.__dict__}

    def save(self, path):
        torch.save(self.state_dict(), path)

    def load(self, path):
        d = torch.load(path, map_location=self.device, weights_only=True)
        self.W.copy_(d["W"]); self.V.copy_(d["V"])
        self.nW.copy_(d["nW"]); self.nV.copy_(d["nV"])
        self.mlp.load_state_dict(d["mlp"])


class NFMTrainer:
    def __init__(self, dataset, hyper: NFMHyper, device="cpu",
                 batch_size=256, epochs=5):
        self.ds = dataset
        self.model = NFMModel(hyper, device=device)
        self.batch_size, self.epochs = batch_size, epochs
        self.device = torch.device(device)

    def train(self, log=print):
        ds = self.ds.to(self.device)
        N = ds.num_rows
        for ep in range(self.epochs):
            tot, nb = 0.0, 0
            for s in range(0, N, self.batch_size):
                b = ds.slice_rows(s, min(s + self.batch_size, N))
                loss = self.model.train_step(b.row_ptr, b.fids, b.vals,
                                             b.labels)
                tot += float(loss.mean()); nb += 1
            if log:
                log(f"epoch {ep}: loss={tot / max(nb, 1):.5f}")
        return self

    def evaluate(self, dataset=None):
        ds = (dataset or self.ds).to(self.device)
        p = self.model.predict_proba(ds.row_ptr, ds.fids, ds.vals)
        loss = torch.nn.functional.binary_cross_entropy(
            p.clamp(1e-7, 1 - 1e-7), ds.labels)
        return {"auc": auc_score(p.cpu(), ds.labels.cpu()),
                "logloss": float(loss),
                "accuracy": float(((p > 0.5) == (ds.labels > 0.5))
                                  .float().mean())}
