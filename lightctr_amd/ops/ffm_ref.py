"""Pure-PyTorch fp32 reference for the FFM ops (test oracle + CPU path).

Math follows the reference trainer semantics
(/root/reference/LightCTR/train/train_ffm_algo.cpp:51-118):
  pred = sum_j w[fid_j] x_j
       + sum_{i<j} dot(V[fid_i, field_j], V[fid_j, field_i]) x_i x_j
  dV[fid_i, field_j] += d x_i x_j V[fid_j, field_i]   (and symmetric)
"""

from __future__ import annotations

import torch


def _row_pairs(n: int, device):
    return torch.triu_indices(n, n, offset=1, device=device)


def ffm_forward_ref(row_ptr, fields, fids, vals, W, V):
    """V: [F, nfields, K]. Returns pred [B]."""
    B = row_ptr.numel() - 1
    pred = torch.zeros(B, dtype=V.dtype, device=V.device)
    for r in range(B):
        lo, hi = int(row_ptr[r]), int(row_ptr[r + 1])
        f = fids[lo:hi].long()
        fl = fields[lo:hi].long()
        x = vals[lo:hi]
        lin = (W[f] * x).sum()
        n = f.numel()
        if n < 2:
            pred[r] = lin
            continue
        ii, jj = _row_pairs(n, V.device)
        a = V[f[ii], fl[jj]]  # [P, K]
        b = V[f[jj], fl[ii]]
        inter = ((a * b).sum(dim=1) * x[ii] * x[jj]).sum()
        pred[r] = lin + inter
    return pred


def ffm_backward_ref(row_ptr, fields, fids, vals, V, dpred):
    """Returns dense (gradW [F], gradV [F,nfields,K])."""
    F, nf, K = V.shape
    gradW = torch.zeros(F, dtype=V.dtype, device=V.device)
    gradV = torch.zeros_like(V)
    B = row_ptr.numel() - 1
    for r in range(B):
        lo, hi = int(row_ptr[r]), int(row_ptr[r + 1])
        f = fids[lo:hi].long()
        fl = fields[lo:hi].long()
        x = vals[lo:hi]
        d = dpred[r]
        gradW.index_add_(0, f, d * x)
        n = f.numel()
        if n < 2:
            continue
        ii, jj = _row_pairs(n, V.device)
        s = (d * x[ii] * x[jj]).unsqueeze(1)  # [P,1]
        a = V[f[ii], fl[jj]]
        b = V[f[jj], fl[ii]]
        gradV.index_put_((f[ii], fl[jj]), s * b, accumulate=True)
        gradV.index_put_((f[jj], fl[ii]), s * a, accumulate=True)
    return gradW, gradV
