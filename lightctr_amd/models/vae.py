"""VAE — FC encoder -> reparameterization sample layer -> FC decoder.

Capability parity with the reference Train_VAE_Algo + Sample_Layer
(/root/reference/LightCTR/train/train_vae_algo.h:42-99,
train/layer/sampleLayer.h:84-101: out = mu + exp(0.5*logvar)*eps with the
KL gradient added directly in backward). Built on the framework's manual
DenseLayer engine (bf16 MFMA GEMMs on GPU, fp32 oracle on CPU).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from .mlp import MLP


@dataclass
class VAEHyper:
    in_dim: int = 784
    hidden: int = 128
    z_dim: int = 16
    lr: float = 1e-3
    kl_weight: float = 1.0
    seed: int = 1234


class VAEModel:
    def __init__(self, hyper: VAEHyper, device: str = "cpu"):
        self.h = hyper
        self.device = torch.device(device)
        h = hyper
        self.enc = MLP([h.in_dim, h.hidden, 2 * h.z_dim],
                       acts=["relu", "none"], optimizer="adam", lr=h.lr,
                       seed=h.seed, device=device)
        self.dec = MLP([h.z_dim, h.hidden, h.in_dim],
                       acts=["relu", "sigmoid"], optimizer="adam", lr=h.lr,
                       seed=h.seed + 100, device=device)
        self._gpu = self.device.type == "cuda"
        self._gen = torch.Generator().manual_seed(h.seed + 7)

    def _prep(self, x):
        if self._gpu:
            from ..ops._extension import require_hip_ops

            return require_hip_ops().to_bf16(x.contiguous())
        return x

    def forward(self, x, train=True, sample=True):
        z_par = self.enc.forward(self._prep(x), train=train)  # [B, 2z]
        zd = self.h.z_dim
        mu, logvar = z_par[:, :zd], z_par[:, zd:]
        logvar = logvar.clamp(-8, 8)
        if sample:
            eps = torch.randn(mu.shape, generator=self._gen).to(mu.device)
        else:
            eps = torch.zeros_like(mu)
        z = mu + torch.exp(0.5 * logvar) * eps
        xhat = self.dec.forward(self._prep(z.contiguous()), train=train)
        return xhat, mu, logvar, eps, z

    def train_step(self, x) -> dict:
        B = x.shape[0]
        xhat, mu, logvar, eps, z = self.forward(x, train=True)
        rec = 0.5 * ((xhat - x) ** 2).sum() / B
        kl = -0.5 * (1 + logvar - mu * mu - logvar.exp()).sum() / B
        # backward: reconstruction path
        dxhat = (xhat - x) / B
        dz = self.dec.backward(dxhat)  # [B, z]
        # sample layer backward (+ the KL grads added directly, like the
        # reference Sample_Layer)
        w = self.h.kl_weight
        dmu = dz + w * mu / B
        dlogvar = dz * eps * 0.5 * torch.exp(0.5 * logvar) \
            + w * 0.5 * (logvar.exp() - 1) / B
        self.enc.backward(torch.cat([dmu, dlogvar], dim=1).contiguous())
        self.dec.apply_grads()
        self.enc.apply_grads()
        return {"loss": float(rec + w * kl), "rec": float(rec),
                "kl": float(kl)}

    def reconstruct(self, x):
        xhat, *_ = self.forward(x, train=False, sample=False)
        return xhat

    def generate(self, n: int):
        z = torch.randn(n, self.h.z_dim, generator=self._gen).to(self.device)
        return self.dec.forward(self._prep(z), train=False)

    def save(self, path):
        torch.save({"enc": self.enc.state_dict(),
                    "dec": self.dec.state_dict(),
                    "hyper": self.h.__dict__}, path)

    def load(self, path):
        d = torch.load(path, map_location=self.device, weights_only=True)
        self.enc.load_state_dict(d["enc"])
        self.dec.load_state_dict(d["dec"])
