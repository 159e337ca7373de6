"""Factorization Machine — MI355X-native trainer.

Capability parity with the reference Train_FM_Algo
(/root/reference/LightCTR/train/train_fm_algo.{h,cpp}: ctor(data, epochs, k)
-> Train() -> saveModel(), O(nk) sumVX forward, fused grad accumulation,
Adagrad apply) — rebuilt GPU-first:

  * parameters W [F] and V [F,K] are flat fp32 slabs resident in HBM3E
  * the train step is 5 HIP kernel launches (forward, loss, backward-scatter,
    bitmap-compact, fused sparse optimizer) with no device->host syncs
  * CPU path uses the fp32 torch reference ops (same math; the test oracle)
  * optimizers: adagrad (reference default) and FTRL-proximal (BASELINE
    config #2: "FM k=16 ... FTRL fused update"); both are applied only to
    the features touched by the batch, exactly like the reference's sparse
    updaters.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops import fm_ref
from ..ops._extension import require_hip_ops, sort_ids
from ..utils.checks import validate_csr_batch
from ..utils.metrics import auc_score


@dataclass
class FMHyper:
    num_features: int
    k: int = 16
    optimizer: str = "adagrad"  # adagrad | ftrl
    lr: float = 0.1
    eps: float = 1e-8
    l2: float = 1e-5
    ftrl_alpha: float = 0.15
    ftrl_beta: float = 1.0
    ftrl_l1: float = 1e-4
    ftrl_l2: float = 1e-4
    # latent-factor updater under optimizer="ftrl": "adagrad" (default —
    # FTRL's L1 starves interactions, measured 0.66 vs 0.79 AUC) | "ftrl"
    ftrl_v: str = "adagrad"
    init_sigma: float = 0.01
    seed: int = 1234


class FMModel:
    """Owns parameters + optimizer state + scratch slabs on one device."""

    def __init__(self, hyper: FMHyper, device: str = "cpu",
                 max_batch_nnz: int = 1 << 22):
        self.h = hyper
        self.device = torch.device(device)
        F, K = hyper.num_features, hyper.k
        assert K in (4, 8, 16, 32, 64), "K must divide the 64-lane wavefront"
        g = torch.Generator().manual_seed(hyper.seed)
        self.W = torch.zeros(F, device=self.device)
        self.V = (
            torch.randn(F, K, generator=g) * hyper.init_sigma
        ).to(self.device)
        self.gradW = torch.zeros_like(self.W)
        self.gradV = torch.zeros_like(self.V)
        # optimizer state
        self.nW = torch.zeros_like(self.W)
        self.nV = torch.zeros_like(self.V)
        if hyper.optimizer == "ftrl":
            self.zW = torch.zeros_like(self.W)
            self.zV = torch.zeros_like(self.V)
        # touched bitmap + compaction buffers
        nwords = (F + 63) // 64
        self.touched = torch.zeros(nwords, dtype=torch.int64, device=self.device)
        cap = min(F, max_batch_nnz)
        self.uniq = torch.zeros(cap, dtype=torch.int32, device=self.device)
        self.count = torch.zeros(1, dtype=torch.int32, device=self.device)
        self._use_hip = self.device.type == "cuda"
        self.backward_mode = "sorted"  # "sorted" (default) | "atomic"
        # fused-apply: interior feature segments get their optimizer update
        # inside the segment-reduce kernel. Measured NET-NEGATIVE in both
        # rounds (round 1: optimizer RMWs serialize the segment walk,
        # apply 185->385us vs separate pass 132->37us; round 2, mode 3
        # ftrlW+adagradV: 721 vs 693 us/step — the boundary-fid bitmap
        # pass still runs and the heavier flush costs more than the
        # separate ~110 us sparse apply saves;
        # profiles/r2_06_fm_step_fused_ab.txt). Two-phase stays default;
        # kept selectable + parity tested.
        self.fused_apply = False
        if self._use_hip:
            require_hip_ops()  # fail loudly if extension missing on GPU

    # -- forward/inference --------------------------------------------------
    def forward(self, row_ptr, fids, vals):
        if self._use_hip:
            ops = require_hip_ops()
            pred, sumVX = ops.fm_forward(row_ptr, fids, vals, self.W, self.V)
        else:
            pred, sumVX = fm_ref.fm_forward_ref(row_ptr, fids, vals, self.W, self.V)
        return pred, sumVX

    def predict_proba(self, row_ptr, fids, vals):
        pred, _ = self.forward(row_ptr, fids, vals)
        return torch.sigmoid(torch.clamp(pred, -16, 16))

    # -- one fused training step -------------------------------------------
    def train_step(self, row_ptr, fids, vals, labels) -> torch.Tensor:
        """Runs fwd+loss+bwd+optimizer. Returns per-row loss tensor (device);
        caller reduces/syncs only when it wants the number."""
        B = row_ptr.numel() - 1
        scale = 1.0 / B
        validate_csr_batch(row_ptr, fids, vals, self.h.num_features)
        if self._use_hip:
            ops = require_hip_ops()
            pred, sumVX = ops.fm_forward(row_ptr, fids, vals, self.W, self.V)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            if self.backward_mode == "sorted":
                # contention-free backward: emit per-entry grads
                # (coalesced), bit-range radix-sort the fids, then
                # segment-reduce into the slabs with interior-store
                # flushes (see profiles/ and fm_kernels.hip for the
                # measured evolution incl. the reverted scatter-emit and
                # segmented-scan variants)
                sorted_fids, perm = sort_ids(fids, self.h.num_features)
                gw, gv = ops.fm_backward_emit(row_ptr, fids, vals, self.V,
                                              sumVX, dpred)
                if self.fused_apply:
                    if (self.h.optimizer == "ftrl"
                            and self.h.ftrl_v == "adagrad"):
                        # round 2: mode 3 = FTRL on W + Adagrad on V (the
                        # default pairing) fused into the segment reduce;
                        # only boundary-spanning fids take the slab +
                        # bitmap + sparse-apply pass
                        ops.fm_sorted_apply_fused(
                            sorted_fids, perm, gw, gv, self.gradW,
                            self.gradV, self.touched, self.W, self.V,
                            self.nW, self.nV, self.zW, None, 3,
                            self.h.ftrl_alpha, self.h.ftrl_beta,
                            self.h.ftrl_l1, self.h.ftrl_l2,
                            v_lr=self.h.lr, v_eps=self.h.eps,
                            v_l2=self.h.l2)
                    elif self.h.optimizer == "ftrl":
                        ops.fm_sorted_apply_fused(
                            sorted_fids, perm, gw, gv, self.gradW,
                            self.gradV, self.touched, self.W, self.V,
                            self.nW, self.nV, self.zW, self.zV, 2,
                            self.h.ftrl_alpha, self.h.ftrl_beta,
                            self.h.ftrl_l1, self.h.ftrl_l2)
                    else:
                        ops.fm_sorted_apply_fused(
                            sorted_fids, perm, gw, gv, self.gradW,
                            self.gradV, self.touched, self.W, self.V,
                            self.nW, self.nV, None, None, 1, self.h.lr,
                            self.h.eps, self.h.l2, 0.0)
                else:
                    ops.fm_sorted_apply(sorted_fids, perm, gw, gv,
                                        self.gradW, self.gradV,
                                        self.touched)
            else:
                ops.fm_backward(row_ptr, fids, vals, self.V, sumVX, dpred,
                                self.gradW, self.gradV, self.touched)
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
            return loss
        # CPU reference path
        pred, sumVX = fm_ref.fm_forward_ref(row_ptr, fids, vals, self.W, self.V)
        loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
        gW, gV = fm_ref.fm_backward_ref(row_ptr, fids, vals, self.V, sumVX, dpred)
        self.gradW += gW
        self.gradV += gV
        uniq = torch.unique(fids.long()).int()
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
        return loss

    # -- checkpoint (reference saveModel/loadModel, fm_algo_abst.h:109-135) --
    def state_dict(self) -> dict:
        d = {"W": self.W, "V": self.V, "nW": self.nW, "nV": self.nV,
             "hyper": self.h.__dict__}
        if self.h.optimizer == "ftrl":
            d["zW"], d["zV"] = self.zW, self.zV
        return d

    def save(self, path: str) -> None:
        torch.save(self.state_dict(), path)

    def load(self, path: str) -> None:
        d = torch.load(path, map_location=self.device, weights_only=True)
        self.W.copy_(d["W"])
        self.V.copy_(d["V"])
        self.nW.copy_(d["nW"])
        self.nV.copy_(d["nV"])
        if self.h.optimizer == "ftrl" and "zW" in d:
            self.zW.copy_(d["zW"])
            self.zV.copy_(d["zV"])


class FMTrainer:
    """Reference-shaped API: ctor(dataset, hyper) -> Train() -> save()."""

    def __init__(self, dataset, hyper: FMHyper, device: str = "cpu",
                 batch_size: int = 256, epochs: int = 5):
        self.ds = dataset
        self.model = FMModel(hyper, device=device)
        self.batch_size = batch_size
        self.epochs = epochs
        self.device = torch.device(device)

    def train(self, log=print):
        ds = self.ds.to(self.device)
        N = ds.num_rows
        for ep in range(self.epochs):
            total_loss = 0.0
            nb = 0
            for s in range(0, N, self.batch_size):
                e = min(s + self.batch_size, N)
                b = ds.slice_rows(s, e)
                loss = self.model.train_step(b.row_ptr, b.fids, b.vals, b.labels)
                total_loss += float(loss.mean())
                nb += 1
            if log:
                log(f"epoch {ep}: loss={total_loss / max(nb, 1):.5f}")
        return self

    def evaluate(self, dataset=None) -> dict:
        ds = (dataset or self.ds).to(self.device)
        p = self.model.predict_proba(ds.row_ptr, ds.fids, ds.vals)
        loss = torch.nn.functional.binary_cross_entropy(
            p.clamp(1e-7, 1 - 1e-7), ds.labels
        )
        return {
            "auc": auc_score(p.cpu(), ds.labels.cpu()),
            "logloss": float(loss),
            "accuracy": float(((p > 0.5) == (ds.labels > 0.5)).float().mean()),
        }
