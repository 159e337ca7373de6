"""Batch inference + evaluation report.

Parity with FM_Predict / GBM_Predict
(/root/reference/LightCTR/predict/{fm_predict,gbm_predict}.{h,cpp}: batch
predict + logloss/accuracy/AUC report, optional score dump). Works with
any model exposing predict_proba over CSR batches or dense tensors.
"""

from __future__ import annotations

import torch

from ..utils.metrics import auc_score, precision_recall_f1


class BatchPredictor:
    def __init__(self, model, batch_size: int = 8192):
        self.model = model
        self.batch_size = batch_size

    def predict_csr(self, ds) -> torch.Tensor:
        """ds: LibffmDataset on the model's device."""
        out = []
        N = ds.num_rows
        for s in range(0, N, self.batch_size):
            b = ds.slice_rows(s, min(s + self.batch_size, N))
            try:
                p = self.model.predict_proba(b.row_ptr, b.fids, b.vals)
            except TypeError:  # field-aware models take fields too
                p = self.model.predict_proba(b.row_ptr, b.fields, b.fids,
                                             b.vals)
            out.append(p)
        return torch.cat(out)

    def report(self, pred: torch.Tensor, labels: torch.Tensor,
               dump_path: str | None = None) -> dict:
        p = pred.clamp(1e-7, 1 - 1e-7)
        loss = torch.nn.functional.binary_cross_entropy(p, labels)
        prec, rec, f1 = precision_recall_f1(pred, labels)
        out = {
            "auc": auc_score(pred.cpu(), labels.cpu()),
            "logloss": float(loss),
            "accuracy": float(((pred > 0.5) == (labels > 0.5))
                              .float().mean()),
            "precision": prec, "recall": rec, "f1": f1,
        }
        if dump_path:  # score dump (reference fm_predict.cpp option)
            with open(dump_path, "w") as f:
                for v in pred.cpu().tolist():
                    f.write(f"{v:.6f}\n")
        return out
