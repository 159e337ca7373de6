"""Pure-PyTorch fp32 reference implementations of the FM ops.

These are the numerics oracles for the HIP kernels (tests compare the
gfx950 kernels against these to ~1e-4) and the CPU execution path of the
framework (BASELINE config #1: "FM k=8, CPU plumbing path").

Math follows the reference trainer semantics
(/root/reference/LightCTR/train/train_fm_algo.cpp:63-127):
  pred = sum_j w[fid_j] x_j + 1/2 (||sum_j v_j x_j||^2 - sum_j ||v_j x_j||^2)
  dW[fid]   += d * x
  dV[fid,k] += d * (sumVX[k] - v[fid,k] x) x
"""

from __future__ import annotations

import torch


def fm_forward_ref(row_ptr, fids, vals, W, V):
    """Returns (pred [B], sumVX [B,K]). Works on any device, fp32."""
    B = row_ptr.numel() - 1
    K = V.shape[1]
    rp = row_ptr.long()
    counts = rp[1:] - rp[:-1]
    row_idx = torch.repeat_interleave(
        torch.arange(B, device=fids.device), counts.to(fids.device)
    )
    f = fids.long()
    x = vals
    vx = V[f] * x.unsqueeze(1)  # [nnz, K]
    sumVX = torch.zeros(B, K, dtype=V.dtype, device=V.device)
    sumVX.index_add_(0, row_idx, vx)
    sumV2X2 = torch.zeros(B, dtype=V.dtype, device=V.device)
    sumV2X2.index_add_(0, row_idx, (vx * vx).sum(dim=1))
    lin = torch.zeros(B, dtype=W.dtype, device=W.device)
    lin.index_add_(0, row_idx, W[f] * x)
    pred = lin + 0.5 * ((sumVX * sumVX).sum(dim=1) - sumV2X2)
    return pred, sumVX


def logloss_grad_ref(pred, label, scale: float):
    """Stable logloss per-row + dpred = (sigmoid(clamp(pred,±16)) - y)*scale."""
    z = pred
    loss = torch.clamp(z, min=0) - z * label + torch.log1p(torch.exp(-z.abs()))
    sig = torch.sigmoid(torch.clamp(z, -16.0, 16.0))
    return loss, (sig - label) * scale


def fm_backward_ref(row_ptr, fids, vals, V, sumVX, dpred):
    """Returns dense (gradW [F], gradV [F,K]) accumulated over the batch."""
    B = row_ptr.numel() - 1
    F, K = V.shape
    rp = row_ptr.long()
    counts = rp[1:] - rp[:-1]
    row_idx = torch.repeat_interleave(
        torch.arange(B, device=fids.device), counts.to(fids.device)
    )
    f = fids.long()
    x = vals
    d = dpred[row_idx]
    gradW = torch.zeros(F, dtype=V.dtype, device=V.device)
    gradW.index_add_(0, f, d * x)
    gv = (sumVX[row_idx] - V[f] * x.unsqueeze(1)) * (d * x).unsqueeze(1)
    gradV = torch.zeros(F, K, dtype=V.dtype, device=V.device)
    gradV.index_add_(0, f, gv)
    return gradW, gradV


def adagrad_apply_ref(uniq, W, V, nW, nV, gradW, gradV, lr, eps, l2):
    """Sparse Adagrad over unique fids; zeroes applied grad slots."""
    u = uniq.long()
    gv = gradV[u] + l2 * V[u]
    nV[u] = nV[u] + gv * gv
    V[u] = V[u] - lr * gv / torch.sqrt(nV[u] + eps)
    gradV[u] = 0
    gw = gradW[u] + l2 * W[u]
    nW[u] = nW[u] + gw * gw
    W[u] = W[u] - lr * gw / torch.sqrt(nW[u] + eps)
    gradW[u] = 0


def ftrl_apply_ref(uniq, W, V, zW, nW, zV, nV, gradW, gradV,
                   alpha, beta, l1, l2, v_adagrad=False, v_lr=0.05,
                   v_eps=1e-8, v_l2=1e-5):
    """Sparse FTRL-proximal over unique fids (per-coordinate; reference
    gradientUpdater.h:235-278 semantics). Zeroes applied grad slots.
    v_adagrad: FTRL on W only, Adagrad on the latent V block (the classic
    CTR split; see fm_kernels.hip)."""
    u = uniq.long()

    def upd(w, z, n, g):
        g2 = g * g
        sigma = (torch.sqrt(n + g2) - torch.sqrt(n)) / alpha
        z_new = z + g - sigma * w
        n_new = n + g2
        w_new = torch.where(
            z_new.abs() <= l1,
            torch.zeros_like(w),
            -(z_new - torch.sign(z_new) * l1)
            / ((beta + torch.sqrt(n_new)) / alpha + l2),
        )
        return w_new, z_new, n_new

    if v_adagrad:
        gv = gradV[u] + v_l2 * V[u]
        nV[u] = nV[u] + gv * gv
        V[u] = V[u] - v_lr * gv / torch.sqrt(nV[u] + v_eps)
    else:
        V[u], zV[u], nV[u] = upd(V[u], zV[u], nV[u], gradV[u])
    gradV[u] = 0
    W[u], zW[u], nW[u] = upd(W[u], zW[u], nW[u], gradW[u])
    gradW[u] = 0
