"""Wide&Deep — MI355X-native trainer.

Capability parity with the reference's distributed sparse-LR + dense-MLP
trainer (/root/reference/LightCTR/distributed_algo_abst.h:105-280: wide LR
over hashed features + per-field embeddings concatenated into a 2-layer MLP)
— rebuilt single-device-first: fused embedding gather (bf16) -> 3-layer MFMA
MLP -> sorted segment-reduce embedding backward + fused sparse optimizer.
Fixed-fields rows (e.g. Criteo 39) required for the concat layout.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops import fm_ref
from ..ops._extension import require_hip_ops, sort_ids
from ..utils.metrics import auc_score
from .mlp import MLP


@dataclass
class WideDeepHyper:
    num_features: int
    num_fields: int = 39
    k: int = 16
    hidden: tuple = (256, 128)
    optimizer: str = "adagrad"  # sparse side
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


class WideDeepModel:
    def __init__(self, hyper: WideDeepHyper, device: str = "cpu",
                 max_batch_nnz: int = 1 << 22):
        self.h = hyper
        self.device = torch.device(device)
        F, nf, K = hyper.num_features, hyper.num_fields, hyper.k
        assert K in (4, 8, 16, 32, 64)
        g = torch.Generator().manual_seed(hyper.seed)
        self.W = torch.zeros(F, device=self.device)  # wide LR weights
        self.E = (torch.randn(F, K, generator=g) * hyper.init_sigma).to(
            self.device)  # embedding table
        self.gradW = torch.zeros_like(self.W)
        self.gradE = torch.zeros_like(self.E)
        self.nW = torch.zeros_like(self.W)
        self.nE = torch.zeros_like(self.E)
        if hyper.optimizer == "ftrl":
            self.zW = torch.zeros_like(self.W)
            self.zE = torch.zeros_like(self.E)
        dims = [nf * K, *hyper.hidden, 1]
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

    def _gather_cpu(self, row_ptr, fids, vals):
        B = row_ptr.numel() - 1
        nf, K = self.h.num_fields, self.h.k
        out = torch.zeros(B, nf * K)
        rp = row_ptr.long()
        counts = (rp[1:] - rp[:-1])
        row_idx = torch.repeat_interleave(torch.arange(B), counts)
        pos = torch.arange(fids.numel()) - rp[:-1][row_idx]
        keep = pos < nf
        emb = self.E[fids.long()] * vals.unsqueeze(1)
        flat_idx = (row_idx * nf + pos).clamp(max=B * nf - 1)
        out.view(B * nf, K).index_add_(0, flat_idx[keep], emb[keep])
        return out

    def _forward_gpu(self, row_ptr, fids, vals, train=True):
        ops = require_hip_ops()
        wide = ops.wide_forward(row_ptr, fids, vals, self.W)
        deep_in = ops.embed_gather(row_ptr, fids, vals, self.E,
                                   self.h.num_fields)
        deep = self.mlp.forward(deep_in, train=train)
        return wide + deep[:, 0], wide

    def _forward_cpu(self, row_ptr, fids, vals, train=True):
        B = row_ptr.numel() - 1
        rp = row_ptr.long()
        row_idx = torch.repeat_interleave(torch.arange(B), rp[1:] - rp[:-1])
        wide = torch.zeros(B)
        wide.index_add_(0, row_idx, self.W[fids.long()] * vals)
        deep_in = self._gather_cpu(row_ptr, fids, vals)
        deep = self.mlp.forward(deep_in, train=train)
        return wide + deep[:, 0], wide

    def predict_proba(self, row_ptr, fids, vals):
        fwd = self._forward_gpu if self._gpu else self._forward_cpu
        pred, _ = fwd(row_ptr, fids, vals, train=False)
        return torch.sigmoid(torch.clamp(pred, -16, 16))

    def train_step(self, row_ptr, fids, vals, labels):
        B = row_ptr.numel() - 1
        scale = 1.0 / B
        if self._gpu:
            ops = require_hip_ops()
            pred, wide = self._forward_gpu(row_ptr, fids, vals)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            dDeep = self.mlp.backward(dpred.unsqueeze(1))  # [B, nf*K]
            gw, gv = ops.embed_backward_emit(row_ptr, vals,
                                             dDeep.contiguous(), dpred,
                                             self.h.num_fields, self.h.k)
            sorted_fids, perm = sort_ids(fids, self.h.num_features)
            ops.fm_sorted_apply(sorted_fids, perm, gw, gv, self.gradW,
                                self.gradE, self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            live = self.uniq[: min(self.uniq.numel(), fids.numel())]
            if self.h.optimizer == "ftrl":
                ops.fm_ftrl_apply(live, self.count, self.W, self.E,
                                  self.zW, self.nW, self.zE, self.nE,
                                  self.gradW, self.gradE, self.h.ftrl_alpha,
                                  self.h.ftrl_beta, self.h.ftrl_l1,
                                  self.h.ftrl_l2,
                                  1 if self.h.ftrl_v == "adagrad" else 0,
                                  self.h.lr, self.h.eps, self.h.l2)
            else:
                ops.fm_adagrad_apply(live, self.count, self.W, self.E,
                                     self.nW, self.nE, self.gradW, self.gradE,
                                     self.h.lr, self.h.eps, self.h.l2)
            self.mlp.apply_grads()
            return loss
        # CPU oracle
        pred, wide = self._forward_cpu(row_ptr, fids, vals)
        loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
        dDeep = self.mlp.backward(dpred.unsqueeze(1))
        rp = row_ptr.long()
        row_idx = torch.repeat_interleave(
            torch.arange(B), rp[1:] - rp[:-1])
        pos = torch.arange(fids.numel()) - rp[:-1][row_idx]
        f = fids.long()
        x = vals
        self.gradW.index_add_(0, f, dpred[row_idx] * x)
        nf, K = self.h.num_fields, self.h.k
        dE = dDeep.view(B, nf, K)[row_idx, pos.clamp(max=nf - 1)] \
            * x.unsqueeze(1)
        dE = torch.where((pos < nf).unsqueeze(1), dE, torch.zeros_like(dE))
        self.gradE.index_add_(0, f, dE)
        uniq = torch.unique(f).int()
        if self.h.optimizer == "ftrl":
            fm_ref.ftrl_apply_ref(uniq, self.W, self.E, self.zW, self.nW,
                                  self.zE, self.nE, self.gradW, self.gradE,
                                  self.h.ftrl_alpha, self.h.ftrl_beta,
                                  self.h.ftrl_l1, self.h.ftrl_l2,
                                  v_adagrad=self.h.ftrl_v == "adagrad",
                                  v_lr=self.h.lr, v_eps=self.h.eps,
                                  v_l2=self.h.l2)
        else:
            fm_ref.adagrad_apply_ref(uniq, self.W, self.E, self.nW, self.nE,
                                     self.gradW, self.gradE, self.h.lr,
                                     self.h.eps, self.h.l2)
        self.mlp.apply_grads()
        return loss

    def state_dict(self):
        return {"W": self.W, "E": self.E, "nW": self.nW, "nE": self.nE,
                "mlp": self.mlp.state_dict(), "hyper": self.h.__dict__}

    def save(self, path):
        torch.save(self.state_dict(), path)

    def load(self, path):
        d = torch.load(path, map_location=self.device, weights_only=True)
        self.W.copy_(d["W"]); self.E.copy_(d["E"])
        self.nW.copy_(d["nW"]); self.nE.copy_(d["nE"])
        self.mlp.load_state_dict(d["mlp"])


class WideDeepTrainer:
    def __init__(self, dataset, hyper: WideDeepHyper, device="cpu",
                 batch_size=256, epochs=5):
        self.ds = dataset
        self.model = WideDeepModel(hyper, device=device)
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
