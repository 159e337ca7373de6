"""PCA — Generalized Hebbian (Sanger's rule) + remove-top-PC.

Parity with the reference /root/reference/LightCTR/util/pca.h (GHA
streaming PCA and the common-component-removal post-process for
embeddings). Batched tensor updates; runs on GPU as-is.
"""

from __future__ import annotations

import torch


class SangerPCA:
    def __init__(self, dim: int, n_components: int, lr: float = 0.01,
                 seed: int = 0, device: str = "cpu"):
        g = torch.Generator().manual_seed(seed)
        self.W = torch.randn(n_components, dim, generator=g).to(device)
        self.W /= self.W.norm(dim=1, keepdim=True)
        self.lr = lr

    def partial_fit(self, X: torch.Tensor, iters: int = 1):
        """Sanger update: dW = lr (y x^T - LT(y y^T) W), y = W x."""
        for _ in range(iters):
            Y = X @ self.W.t()  # [N, C]
            yyT = Y.t() @ Y / X.shape[0]
            lower = torch.tril(yyT)
            self.W += self.lr * (Y.t() @ X / X.shape[0] - lower @ self.W)
        return self

    def fit(self, X: torch.Tensor, iters: int = 200):
        mean = X.mean(dim=0, keepdim=True)
        Xc = X - mean
        self.mean = mean
        for _ in range(iters):
            self.partial_fit(Xc)
        return self

    def components(self) -> torch.Tensor:
        return self.W / self.W.norm(dim=1, keepdim=True)

    def transform(self, X: torch.Tensor) -> torch.Tensor:
        return (X - getattr(self, "mean", 0)) @ self.components().t()


def remove_top_pc(X: torch.Tensor, n: int = 1, iters: int = 100
                  ) -> torch.Tensor:
    """Remove the top-n principal components (the embedding post-process
    the reference pairs with PCA)."""
    pca = SangerPCA(X.shape[1], n, lr=0.05, seed=0, device=str(X.device))
    pca.fit(X, iters=iters)
    comp = pca.components()
    proj = (X - pca.mean) @ comp.t() @ comp
    return X - proj
