"""Field-aware Factorization Machine — MI355X-native trainer.

Capability parity with the reference Train_FFM_Algo
(/root/reference/LightCTR/train/train_ffm_algo.{h,cpp}) — per-(feature,field)
latent vectors V[F, nfields, K], explicit pairwise interaction. GPU path uses
the fused CDNA4 pairwise kernels (ops/csrc/ffm_kernels.hip) and the generic
sparse fused optimizers; CPU path uses the torch reference ops.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops import ffm_ref, fm_ref
from ..ops._extension import require_hip_ops, sort_ids
from ..utils.checks import validate_csr_batch
from ..utils.metrics import auc_score


@dataclass
class FFMHyper:
    num_features: int
    num_fields: int
    k: int = 8
    optimizer: str = "adagrad"  # adagrad | ftrl
    lr: float = 0.1
    eps: float = 1e-8
    l2: float = 1e-5
    ftrl_alpha: float = 0.15
    ftrl_beta: float = 1.0
    ftrl_l1: float = 1e-4
    ftrl_l2: float = 1e-4
    ftrl_v: str = "adagrad"  # latent updater under ftrl (see models/fm.py)
    init_sigma: float = 0.01
    seed: int = 1234
    # "bf16" = BASELINE config #3 precision: fp32 master weights +
    # optimizer state, bf16 compute mirror of V read by the forward /
    # backward kernels (refreshed in the fused optimizer for exactly the
    # touched features). Exceeds the reference's fp16-wire-only contract
    # (/root/reference/LightCTR/common/float16.h:65-154).
    dtype: str = "fp32"  # fp32 | bf16


class FFMModel:
    def __init__(self, hyper: FFMHyper, device: str = "cpu",
                 max_batch_nnz: int = 1 << 22):
        self.h = hyper
        self.device = torch.device(device)
        F, nf, K = hyper.num_features, hyper.num_fields, hyper.k
        assert K in (4, 8, 16, 32, 64)
        g = torch.Generator().manual_seed(hyper.seed)
        self.W = torch.zeros(F, device=self.device)
        self.V = (torch.randn(F, nf, K, generator=g) * hyper.init_sigma).to(
            self.device)
        self.gradW = torch.zeros_like(self.W)
        self.gradV = torch.zeros_like(self.V)
        self.nW = torch.zeros_like(self.W)
        self.nV = torch.zeros_like(self.V)
        if hyper.optimizer == "ftrl":
            self.zW = torch.zeros_like(self.W)
            self.zV = torch.zeros_like(self.V)
        nwords = (F + 63) // 64
        self.touched = torch.zeros(nwords, dtype=torch.int64,
                                   device=self.device)
        cap = min(F, max_batch_nnz)
        self.uniq = torch.zeros(cap, dtype=torch.int32, device=self.device)
        self.count = torch.zeros(1, dtype=torch.int32, device=self.device)
        self._use_hip = self.device.type == "cuda"
        assert hyper.dtype in ("fp32", "bf16")
        # bf16 compute mirror (fp32 master stays in self.V)
        self.Vh = self.V.to(torch.bfloat16) if hyper.dtype == "bf16" \
            else None
        # backward variants, all parity-tested (tests/test_ffm.py):
        #   rowemit (default since round 2) — per-row staged block emit
        #            with field-map direct quad stores + interior-store
        #            segment reduce: 1.26 + 1.56 ms = 2.82 ms backward
        #            (gpurun_out/ab_ffm_r2.txt) after the float4/half2
        #            staging + dwordx2 store vectorization; the round-1
        #            scalar emit measured 4.7-5.2 ms
        #   sorted — per-run LDS-block recompute, 3.91 ms (random-line
        #            bandwidth bound per the round-1 PMC/ISA analysis)
        #   blocks — wave/entry fp32 emit + block reduce, 5.6 ms
        #   atomic — naive scatter (hot-feature serialization)
        # rowemit needs the staged-emit LDS footprint to fit (mirrors
        # ffm_staged_eligible in ffm_kernels.hip, maxn=40 / 8 waves) and
        # the apply's register-accumulator cap (D <= 1024); larger
        # field*factor products fall back to the sorted walk
        maxn, D = 40, nf * K
        lds_bytes = (8 * nf * (K + 1) * 4 + maxn * D * 2 + maxn * 4
                     + nf * 4 + 4)
        self._rowemit_ok = lds_bytes <= (64 << 10) and D <= 1024
        self.backward_mode = "rowemit" if self._rowemit_ok else "sorted"
        # fuse the sparse optimizer into the apply's interior-run flush
        # (features whose sorted segment is wholly owned by one chunk
        # skip the gradV slab + bitmap + separate optimizer kernel);
        # spanning features keep the two-phase path. Measured 3.65 vs
        # 4.15 ms/step (18.0 vs 15.8M ex/s, bf16 B=65536) — default ON
        # for the supported pairings (adagrad, ftrlW+adagradV); other
        # pairings fall back to two-phase automatically.
        self.fused_apply = True
        self._fields_checked = False
        if self._use_hip:
            require_hip_ops()

    @property
    def _Vc(self):
        """compute-side V: the bf16 mirror in bf16 mode, fp32 master
        otherwise"""
        return self.Vh if self.Vh is not None else self.V

    def forward(self, row_ptr, fields, fids, vals):
        if self._use_hip:
            ops = require_hip_ops()
            return ops.ffm_forward(row_ptr, fields, fids, vals, self.W,
                                   self._Vc)
        if self.Vh is not None:
            return ffm_ref.ffm_forward_ref(row_ptr, fields, fids, vals,
                                           self.W, self.Vh.float())
        return ffm_ref.ffm_forward_ref(row_ptr, fields, fids, vals, self.W,
                                       self.V)

    def predict_proba(self, row_ptr, fields, fids, vals):
        pred = self.forward(row_ptr, fields, fids, vals)
        return torch.sigmoid(torch.clamp(pred, -16, 16))

    def _apply_optimizer_hip(self, ops, live):
        if self.h.optimizer == "ftrl":
            ops.sparse_ftrl_apply(live, self.count, self.W, self.V,
                                  self.zW, self.nW, self.zV, self.nV,
                                  self.gradW, self.gradV, self.h.ftrl_alpha,
                                  self.h.ftrl_beta, self.h.ftrl_l1,
                                  self.h.ftrl_l2,
                                  1 if self.h.ftrl_v == "adagrad" else 0,
                                  self.h.lr, self.h.eps, self.h.l2,
                                  Vh=self.Vh)
        else:
            ops.sparse_adagrad_apply(live, self.count, self.W, self.V,
                                     self.nW, self.nV, self.gradW, self.gradV,
                                     self.h.lr, self.h.eps, self.h.l2,
                                     Vh=self.Vh)

    def train_step(self, row_ptr, fields, fids, vals, labels) -> torch.Tensor:
        B = row_ptr.numel() - 1
        scale = 1.0 / B
        validate_csr_batch(row_ptr, fids, vals, self.h.num_features,
                           fields, self.h.num_fields)
        if not self._fields_checked:
            # one-time guard: an out-of-range field silently corrupts the
            # [F, nfields, K] indexing on device (costs one sync, once)
            fmax = int(fields.max()) if fields.numel() else 0
            assert fmax < self.h.num_fields, \
                f"field id {fmax} >= num_fields {self.h.num_fields}"
            self._fields_checked = True
        if self._use_hip:
            ops = require_hip_ops()
            pred = ops.ffm_forward(row_ptr, fields, fids, vals, self.W,
                                   self._Vc)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            if self.backward_mode == "rowemit":
                # power-of-two block scale keeps 1/B-scaled logloss grads
                # (~1e-8 at B=65536) inside fp16 normal range; the apply
                # divides it back out exactly
                bscale = float(1 << min(24, max(0, B.bit_length() - 1)))
                gw, gblocks = ops.ffm_row_emit(row_ptr, fields, fids, vals,
                                               self._Vc, dpred,
                                               scale=bscale)
                sorted_fids, perm = sort_ids(fids, self.h.num_features)
                fuse = self.fused_apply and (
                    self.h.optimizer == "adagrad"
                    or (self.h.optimizer == "ftrl"
                        and self.h.ftrl_v == "adagrad"))
                if fuse:
                    mode = 1 if self.h.optimizer == "adagrad" else 3
                    wp = ((self.h.lr, self.h.eps, self.h.l2, 0.0)
                          if mode == 1 else
                          (self.h.ftrl_alpha, self.h.ftrl_beta,
                           self.h.ftrl_l1, self.h.ftrl_l2))
                    ops.ffm_blocks_apply_f16(
                        sorted_fids, perm, gblocks, gw, self.gradW,
                        self.gradV.view(self.h.num_features, -1),
                        self.touched, inv_scale=1.0 / bscale,
                        opt_mode=mode,
                        V=self.V.view(self.h.num_features, -1),
                        W=self.W, nW=self.nW,
                        zW=self.zW if mode == 3 else None,
                        nV=self.nV.view(self.h.num_features, -1),
                        Vh=(self.Vh.view(self.h.num_features, -1)
                            if self.Vh is not None else None),
                        p0=wp[0], p1=wp[1], p2=wp[2], p3=wp[3],
                        q0=self.h.lr, q1=self.h.eps, q2=self.h.l2)
                else:
                    ops.ffm_blocks_apply_f16(sorted_fids, perm, gblocks, gw,
                                             self.gradW,
                                             self.gradV.view(
                                                 self.h.num_features, -1),
                                             self.touched,
                                             inv_scale=1.0 / bscale)
            elif self.backward_mode == "blocks":
                row_of_entry = ops.row_index(row_ptr, fids.numel())
                gw, gblocks = ops.ffm_block_emit(row_of_entry, row_ptr,
                                                 fields, fids, vals, self.V,
                                                 dpred)
                sorted_fids, perm = sort_ids(fids, self.h.num_features)
                ops.ffm_blocks_apply(sorted_fids, perm, gblocks, gw,
                                     self.gradW,
                                     self.gradV.view(self.h.num_features,
                                                     -1),
                                     self.touched)
            elif self.backward_mode == "sorted":
                sorted_fids, perm = sort_ids(fids, self.h.num_features)
                row_of_entry = ops.row_index(row_ptr, fids.numel())
                ops.ffm_sorted_backward(sorted_fids, perm, row_of_entry,
                                        row_ptr, fields, fids, vals, self.V,
                                        dpred, self.gradW, self.gradV,
                                        self.touched)
            else:
                ops.ffm_backward(row_ptr, fields, fids, vals, self.V, dpred,
                                 self.gradW, self.gradV, self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            live = self.uniq[: min(self.uniq.numel(), fids.numel())]
            self._apply_optimizer_hip(ops, live)
            return loss
        pred = ffm_ref.ffm_forward_ref(row_ptr, fields, fids, vals, self.W,
                                       self.V)
        loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
        gW, gV = ffm_ref.ffm_backward_ref(row_ptr, fields, fids, vals, self.V,
                                          dpred)
        self.gradW += gW
        self.gradV += gV
        uniq = torch.unique(fids.long()).int()
        F = self.h.num_features
        Vf = self.V.view(F, -1)
        gVf = self.gradV.view(F, -1)
        nVf = self.nV.view(F, -1)
        if self.h.optimizer == "ftrl":
            fm_ref.ftrl_apply_ref(uniq, self.W, Vf, self.zW, self.nW,
                                  self.zV.view(F, -1), nVf, self.gradW, gVf,
                                  self.h.ftrl_alpha, self.h.ftrl_beta,
                                  self.h.ftrl_l1, self.h.ftrl_l2,
                                  v_adagrad=self.h.ftrl_v == "adagrad",
                                  v_lr=self.h.lr, v_eps=self.h.eps,
                                  v_l2=self.h.l2)
        else:
            fm_ref.adagrad_apply_ref(uniq, self.W, Vf, self.nW, nVf,
                                     self.gradW, gVf, self.h.lr, self.h.eps,
                                     self.h.l2)
        return loss

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
        for name in ("W", "V", "nW", "nV"):
            getattr(self, name).copy_(d[name])
        if self.h.optimizer == "ftrl" and "zW" in d:
            self.zW.copy_(d["zW"])
            self.zV.copy_(d["zV"])
        if self.Vh is not None:
            self.Vh.copy_(self.V.to(torch.bfloat16))


class FFMTrainer:
    """Reference-shaped API: ctor(dataset, hyper) -> train() -> save()."""

    def __init__(self, dataset, hyper: FFMHyper, device: str = "cpu",
                 batch_size: int = 256, epochs: int = 5):
        self.ds = dataset
        self.model = FFMModel(hyper, device=device)
        self.batch_size = batch_size
        self.epochs = epochs
        self.device = torch.device(device)

    def train(self, log=print):
        ds = self.ds.to(self.device)
        N = ds.num_rows
        for ep in range(self.epochs):
            tot, nb = 0.0, 0
            for s in range(0, N, self.batch_size):
                b = ds.slice_rows(s, min(s + self.batch_size, N))
                loss = self.model.train_step(b.row_ptr, b.fields, b.fids,
                                             b.vals, b.labels)
                tot += float(loss.mean())
                nb += 1
            if log:
                log(f"epoch {ep}: loss={tot / max(nb, 1):.5f}")
        return self

    def evaluate(self, dataset=None) -> dict:
        ds = (dataset or self.ds).to(self.device)
        p = self.model.predict_proba(ds.row_ptr, ds.fields, ds.fids, ds.vals)
        loss = torch.nn.functional.binary_cross_entropy(
            p.clamp(1e-7, 1 - 1e-7), ds.labels)
        return {
            "auc": auc_score(p.cpu(), ds.labels.cpu()),
            "logloss": float(loss),
            "accuracy": float(((p > 0.5) == (ds.labels > 0.5)).float().mean()),
        }
