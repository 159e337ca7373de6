"""Evaluation metrics.

HistAUC re-implements the reference's histogram AUC evaluator semantics
(/root/reference/LightCTR/util/evaluator.h:51-103: bucketized pos/neg
histograms + trapezoid integration) with a configurable bucket count,
device-friendly (torch.bincount on GPU or CPU). auc_score is the exact
rank-based AUC used in tests.
"""

from __future__ import annotations

import torch


class HistAUC:
    """Streaming histogram AUC: add(pred, label) batches, then compute().

    On GPU the histogram accumulate and the scan run as in-tree HIP
    kernels (ops/csrc/misc_kernels.hip auc_hist_add/auc_scan — the §2.4
    "atomic-histogram kernel + device scan" row for evaluator.h:61-94);
    the CPU path keeps the torch.bincount reference implementation.
    """

    def __init__(self, buckets: int = 1 << 20, device: str = "cpu"):
        self.buckets = buckets
        self.device = torch.device(device)
        self._gpu = self.device.type == "cuda"
        if self._gpu:
            from ..ops._extension import require_hip_ops

            self._ops = require_hip_ops()
            # [2, buckets] uint32 as int32 storage: [0]=neg, [1]=pos
            self.hist = torch.zeros(2 * buckets, dtype=torch.int32,
                                    device=self.device)
        else:
            self.pos = torch.zeros(buckets, dtype=torch.float64,
                                   device=device)
            self.neg = torch.zeros(buckets, dtype=torch.float64,
                                   device=device)

    def add(self, pred: torch.Tensor, label: torch.Tensor) -> None:
        """pred in [0,1] (probabilities), label in {0,1}."""
        if self._gpu:
            self._ops.auc_hist_add(pred.float().contiguous(),
                                   label.float().contiguous(), self.hist)
            return
        idx = (pred.clamp(0, 1) * (self.buckets - 1)).long()
        lb = label.bool()
        if lb.any():
            self.pos += torch.bincount(
                idx[lb], minlength=self.buckets
            ).to(self.pos.dtype)
        if (~lb).any():
            self.neg += torch.bincount(
                idx[~lb], minlength=self.buckets
            ).to(self.neg.dtype)

    def compute(self) -> float:
        if self._gpu:
            out = self._ops.auc_scan(self.hist)
            correct, npos, nneg = (float(x) for x in out.cpu())
            if npos == 0 or nneg == 0:
                return 0.5
            return correct / (npos * nneg)
        npos = self.pos.sum()
        nneg = self.neg.sum()
        if npos == 0 or nneg == 0:
            return 0.5
        # P(pos_score > neg_score) + 0.5 P(tie): for each negative in bucket b,
        # count positives in buckets above b, plus half the same-bucket ties.
        cum_pos_above = self.pos.sum() - torch.cumsum(self.pos, 0)
        correct = (self.neg * cum_pos_above).sum() + 0.5 * (self.neg * self.pos).sum()
        return float(correct / (npos * nneg))


def auc_score(pred: torch.Tensor, label: torch.Tensor) -> float:
    """Exact AUC (rank statistic with tie handling)."""
    pred = pred.detach().double().flatten()
    label = label.detach().double().flatten()
    order = torch.argsort(pred)
    sorted_pred = pred[order]
    ranks = torch.empty_like(sorted_pred)
    # average ranks over ties
    uniq, inv, counts = torch.unique_consecutive(
        sorted_pred, return_inverse=True, return_counts=True
    )
    cum = torch.cumsum(counts, 0)
    start = cum - counts
    avg_rank = (start + cum - 1).double() / 2.0 + 1.0
    ranks = avg_rank[inv]
    r = torch.empty_like(ranks)
    r[order] = ranks
    npos = label.sum()
    nneg = label.numel() - npos
    if npos == 0 or nneg == 0:
        return 0.5
    return float((r[label > 0.5].sum() - npos * (npos + 1) / 2) / (npos * nneg))


def precision_recall_f1(pred, label, threshold: float = 0.5):
    """Reference evaluator.h:27-49 semantics."""
    p = (pred >= threshold).double()
    y = label.double()
    tp = (p * y).sum()
    fp = (p * (1 - y)).sum()
    fn = ((1 - p) * y).sum()
    precision = float(tp / (tp + fp)) if (tp + fp) > 0 else 0.0
    recall = float(tp / (tp + fn)) if (tp + fn) > 0 else 0.0
    f1 = (
        2 * precision * recall / (precision + recall)
        if precision + recall > 0
        else 0.0
    )
    return precision, recall, f1
