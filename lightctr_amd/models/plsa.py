"""PLSA topic model trained by EM — MI355X-native.

Capability parity with the reference topic model
(/root/reference/LightCTR/train/train_tm_algo.{h,cpp}: EM over the
doc-word-topic tensor with cached marginal sums). Rebuilt as batched
sparse-tensor EM: documents are a CSR (doc, word, count) matrix; the
E-step posterior p(z|d,w) and both M-step aggregations are computed with
segment reductions over the nonzeros, GPU-friendly.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class PLSAHyper:
    n_topics: int = 10
    max_iters: int = 50
    tol: float = 1e-5
    seed: int = 1234


class PLSAModel:
    def __init__(self, hyper: PLSAHyper, device: str = "cpu"):
        self.h = hyper
        self.device = torch.device(device)
        self.p_w_z = None  # [T, V] p(word | topic)
        self.p_z_d = None  # [D, T] p(topic | doc)

    def fit(self, doc_ids: torch.Tensor, word_ids: torch.Tensor,
            counts: torch.Tensor, n_docs: int, n_words: int, log=None):
        """Nonzero triplets (doc, word, count)."""
        T = self.h.n_topics
        dev = self.device
        doc_ids = doc_ids.long().to(dev)
        word_ids = word_ids.long().to(dev)
        counts = counts.float().to(dev)
        g = torch.Generator().manual_seed(self.h.seed)
        self.p_w_z = torch.rand(T, n_words, generator=g).to(dev)
        self.p_w_z /= self.p_w_z.sum(dim=1, keepdim=True)
        self.p_z_d = torch.rand(n_docs, T, generator=g).to(dev)
        self.p_z_d /= self.p_z_d.sum(dim=1, keepdim=True)
        prev = -float("inf")
        for it in range(self.h.max_iters):
            # E-step: p(z|d,w) over nonzeros  [nnz, T]
            joint = self.p_w_z.t()[word_ids] * self.p_z_d[doc_ids]
            denom = joint.sum(dim=1, keepdim=True).clamp(min=1e-30)
            post = joint / denom
            ll = float((counts * torch.log(denom.squeeze(1))).sum()
                       / counts.sum())
            # M-step: n(d,w) p(z|d,w) aggregated both ways
            weighted = post * counts.unsqueeze(1)
            new_wz = torch.zeros_like(self.p_w_z.t())  # [V, T]
            new_wz.index_add_(0, word_ids, weighted)
            self.p_w_z = (new_wz.t()
                          / new_wz.sum(dim=0).unsqueeze(1).clamp(min=1e-30))
            new_zd = torch.zeros_like(self.p_z_d)
            new_zd.index_add_(0, doc_ids, weighted)
            self.p_z_d = new_zd / new_zd.sum(dim=1, keepdim=True).clamp(
                min=1e-30)
            if log:
                log(f"iter {it}: loglik={ll:.6f}")
            if abs(ll - prev) < self.h.tol:
                break
            prev = ll
        return self

    def top_words(self, topic: int, k: int = 10):
        return self.p_w_z[topic].topk(k).indices.tolist()

    def doc_topics(self) -> torch.Tensor:
        return self.p_z_d

    def save(self, path):
        torch.save({"p_w_z": self.p_w_z, "p_z_d": self.p_z_d,
                    "hyper": self.h.__dict__}, path)

    def load(self, path):
        d = torch.load(path, map_location=self.device, weights_only=True)
        self.p_w_z, self.p_z_d = d["p_w_z"], d["p_z_d"]
