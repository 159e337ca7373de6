"""ANN index — Annoy-style random-projection forest KNN.

Capability parity with the reference ANNIndex
(/root/reference/LightCTR/predict/ann_index.h: forest of trees split by a
2-means-derived hyperplane :225-268, beam search over a priority queue
:198-223). Batched for GPU: node membership and distances are tensor ops;
the beam search walks tree arrays.
"""

from __future__ import annotations

import heapq

import torch


class _TreeNode:
    __slots__ = ("w", "b", "left", "right", "items")

    def __init__(self):
        self.w = None
        self.b = 0.0
        self.left = None
        self.right = None
        self.items = None


class ANNIndex:
    def __init__(self, X: torch.Tensor, n_trees: int = 8,
                 leaf_size: int = 16, seed: int = 0):
        self.X = X
        self.leaf_size = leaf_size
        g = torch.Generator().manual_seed(seed)
        self.trees = [self._build(torch.arange(X.shape[0]), g)
                      for _ in range(n_trees)]

    # ---- build: split by the 2-means hyperplane (reference :225-268) ----
    def _build(self, items: torch.Tensor, g) -> _TreeNode:
        node = _TreeNode()
        if items.numel() <= self.leaf_size:
            node.items = items
            return node
        pts = self.X[items]
        # two-means: pick 2 random seeds, few refinement iters
        idx = torch.randperm(items.numel(), generator=g)[:2]
        c = pts[idx].clone()
        for _ in range(4):
            d = torch.cdist(pts, c)
            a = d.argmin(dim=1)
            for k in range(2):
                sel = a == k
                if sel.any():
                    c[k] = pts[sel].mean(dim=0)
        w = c[0] - c[1]
        nrm = w.norm()
        if nrm < 1e-12:  # degenerate: random hyperplane
            w = torch.randn(pts.shape[1], generator=g)
            nrm = w.norm()
        w = w / nrm
        b = -float(w @ (c[0] + c[1]) / 2)
        side = pts @ w + b > 0
        if bool(side.all()) or not bool(side.any()):
            node.items = items  # unsplittable -> leaf
            return node
        node.w, node.b = w, b
        node.left = self._build(items[side], g)
        node.right = self._build(items[~side], g)
        return node

    # ---- query: beam search with a priority queue (reference :198-223) --
    def query(self, q: torch.Tensor, k: int = 10,
              search_k: int | None = None):
        search_k = search_k or k * len(self.trees) * 2
        heap = []  # (-margin, counter, node)
        cnt = 0
        for t in self.trees:
            heapq.heappush(heap, (-float("inf"), cnt, t))
            cnt += 1
        cand = []
        while heap and len(cand) < search_k:
            _, _, node = heapq.heappop(heap)
            if node.items is not None:
                cand.append(node.items)
                continue
            margin = float(node.w @ q + node.b)
            near, far = ((node.left, node.right) if margin > 0
                         else (node.right, node.left))
            heapq.heappush(heap, (-abs(margin) * 0 - float("inf"), cnt,
                                  near))
            cnt += 1
            heapq.heappush(heap, (-(-abs(margin)), cnt, far))
            cnt += 1
        ids = torch.unique(torch.cat(cand)) if cand else torch.arange(
            self.X.shape[0])
        d = ((self.X[ids] - q) ** 2).sum(dim=1)
        top = d.argsort()[:k]
        return ids[top], d[top].sqrt()

    def query_exact(self, q: torch.Tensor, k: int = 10):
        d = ((self.X - q) ** 2).sum(dim=1)
        top = d.argsort()[:k]
        return top, d[top].sqrt()
