"""Hash-sharded NFM across N GPUs (completes the sharded family:
FM / FFM / Wide&Deep / NFM — BASELINE config #4's "Wide&Deep / NFM on
8xMI355X").

Same hybrid layout as parallel/sharded_widedeep.py: the wide weights W
and latent table V are sharded by `fid % world` (deduplicated RCCL
all-to-all pull/push per step, the reference's PS/DHT replacement), and
the bi-interaction MLP is replicated with SUM-all-reduced gradients, so
training is exactly minibatch SGD on the union batch. Equivalence pinned
by tests/test_parallel.py::test_sharded_nfm_matches_single.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..models.mlp import MLP
from ..models.nfm import NFMHyper
from ..ops import fm_ref
from ..ops._extension import require_hip_ops, sort_ids
from .sharded_widedeep import _allreduce_sum


class ShardedNFMModel:
    def __init__(self, hyper: NFMHyper, device: str = "cpu", group=None):
        self.h = hyper
        self.device = torch.device(device)
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        F, K = hyper.num_features, hyper.k
        self.F_local = (F + self.world - 1) // self.world
        g = torch.Generator().manual_seed(hyper.seed + 17 * self.rank)
        dev = self.device
        self.W = torch.zeros(self.F_local, device=dev)
        self.V = (torch.randn(self.F_local, K, generator=g)
                  * hyper.init_sigma).to(dev)
        self.gradW = torch.zeros_like(self.W)
        self.gradV = torch.zeros_like(self.V)
        self.nW = torch.zeros_like(self.W)
        self.nV = torch.zeros_like(self.V)
        if hyper.optimizer == "ftrl":
            self.zW = torch.zeros_like(self.W)
            self.zV = torch.zeros_like(self.V)
        nwords = (self.F_local + 63) // 64
        self.touched = torch.zeros(nwords, dtype=torch.int64, device=dev)
        self.uniq = torch.zeros(self.F_local, dtype=torch.int32, device=dev)
        self.count = torch.zeros(1, dtype=torch.int32, device=dev)
        dims = [K, *hyper.hidden, 1]
        self.mlp = MLP(dims, optimizer=hyper.mlp_optimizer, lr=hyper.mlp_lr,
                       dropout=0.0, seed=hyper.seed, device=device)
        self._use_hip = dev.type == "cuda"
        if self._use_hip:
            require_hip_ops()

    def _exchange(self, send, send_counts, recv_counts):
        out = torch.empty((sum(recv_counts),) + tuple(send.shape[1:]),
                          dtype=send.dtype, device=send.device)
        dist.all_to_all_single(out, send.contiguous(),
                               output_split_sizes=recv_counts,
                               input_split_sizes=send_counts,
                               group=self.group)
        return out

    def train_step(self, row_ptr, fids, vals, labels) -> torch.Tensor:
        h = self.h
        B = row_ptr.numel() - 1
        world = self.world
        scale = 1.0 / (B * world)

        uniq, inverse = torch.unique(fids, return_inverse=True)
        U = uniq.numel()
        owner = uniq.long() % world
        order = torch.argsort(owner, stable=True)
        uniq_o = uniq[order]
        cnt_t = torch.bincount(owner, minlength=world).to(self.device)
        send_counts = cnt_t.cpu().tolist()
        recv_cnt_t = torch.empty(world, dtype=cnt_t.dtype,
                                 device=self.device)
        dist.all_to_all_single(recv_cnt_t, cnt_t, group=self.group)
        recv_counts = recv_cnt_t.cpu().tolist()

        req = self._exchange(uniq_o, send_counts, recv_counts)
        lidx = req.long() // world
        Wl_o = self._exchange(self.W[lidx], recv_counts, send_counts)
        Vl_o = self._exchange(self.V[lidx], recv_counts, send_counts)
        Wl = torch.empty_like(Wl_o)
        Vl = torch.empty_like(Vl_o)
        Wl[order] = Wl_o
        Vl[order] = Vl_o

        fids_local = inverse.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            wide, sumVX, vec, vec_bf = ops.nfm_forward(row_ptr, fids_local,
                                                       vals, Wl, Vl)
            deep = self.mlp.forward(vec_bf, train=True)
            pred = wide + deep[:, 0]
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            ddeep = self.mlp.backward(dpred.unsqueeze(1))
            gw, gv = ops.nfm_backward_emit(row_ptr, fids_local, vals, Vl,
                                           sumVX, ddeep.contiguous(), dpred)
            sorted_l, perm = sort_ids(fids_local, U)
            gWl = torch.zeros(U, device=self.device)
            gVl = torch.zeros(U, h.k, device=self.device)
            scratch = torch.zeros((U + 63) // 64, dtype=torch.int64,
                                  device=self.device)
            ops.fm_sorted_apply(sorted_l, perm, gw, gv, gWl, gVl, scratch)
        else:
            rp = row_ptr.long()
            counts = rp[1:] - rp[:-1]
            row_idx = torch.repeat_interleave(torch.arange(B), counts)
            f = inverse
            x = vals
            vx = Vl[f] * x.unsqueeze(1)
            sumVX = torch.zeros(B, h.k)
            sumVX.index_add_(0, row_idx, vx)
            sumV2X2k = torch.zeros(B, h.k)
            sumV2X2k.index_add_(0, row_idx, vx * vx)
            vec = 0.5 * (sumVX * sumVX - sumV2X2k)
            wide = torch.zeros(B)
            wide.index_add_(0, row_idx, Wl[f] * x)
            deep = self.mlp.forward(vec, train=True)
            pred = wide + deep[:, 0]
            loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
            dvec = self.mlp.backward(dpred.unsqueeze(1))
            gWl = torch.zeros(U)
            gWl.index_add_(0, f, dpred[row_idx] * x)
            gv = (sumVX[row_idx] - Vl[f] * x.unsqueeze(1)) \
                * x.unsqueeze(1) * dvec[row_idx]
            gVl = torch.zeros(U, h.k)
            gVl.index_add_(0, f, gv)

        grads = []
        for la in self.mlp.layers:
            grads += [la._dW, la._db]
        _allreduce_sum(grads, group=self.group)
        self.mlp.apply_grads()

        gW_recv = self._exchange(gWl[order], send_counts, recv_counts)
        gV_recv = self._exchange(gVl[order], send_counts, recv_counts)
        lidx32 = lidx.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            sorted_own, perm_own = sort_ids(lidx32, self.F_local)
            ops.fm_sorted_apply(sorted_own, perm_own, gW_recv.contiguous(),
                                gV_recv.contiguous(), self.gradW, self.gradV,
                                self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            live = self.uniq[: min(self.uniq.numel(), int(lidx32.numel()))]
            if h.optimizer == "ftrl":
                ops.fm_ftrl_apply(live, self.count, self.W, self.V, self.zW,
                                  self.nW, self.zV, self.nV, self.gradW,
                                  self.gradV, h.ftrl_alpha, h.ftrl_beta,
                                  h.ftrl_l1, h.ftrl_l2,
                                  1 if h.ftrl_v == "adagrad" else 0, h.lr,
                                  h.eps, h.l2)
            else:
                ops.fm_adagrad_apply(live, self.count, self.W, self.V,
                                     self.nW, self.nV, self.gradW,
                                     self.gradV, h.lr, h.eps, h.l2)
        else:
            self.gradW.index_add_(0, lidx, gW_recv)
            self.gradV.index_add_(0, lidx, gV_recv)
            own = torch.unique(lidx).int()
            if h.optimizer == "ftrl":
                fm_ref.ftrl_apply_ref(own, self.W, self.V, self.zW, self.nW,
                                      self.zV, self.nV, self.gradW,
                                      self.gradV, h.ftrl_alpha, h.ftrl_beta,
                                      h.ftrl_l1, h.ftrl_l2,
                                      v_adagrad=h.ftrl_v == "adagrad",
                                      v_lr=h.lr, v_eps=h.eps, v_l2=h.l2)
            else:
                fm_ref.adagrad_apply_ref(own, self.W, self.V, self.nW,
                                         self.nV, self.gradW, self.gradV,
                                         h.lr, h.eps, h.l2)
        return loss

    def save(self, path_prefix: str) -> None:
        torch.save({"W": self.W, "V": self.V, "nW": self.nW, "nV": self.nV,
                    "mlp": self.mlp.state_dict(), "rank": self.rank,
                    "world": self.world},
                   f"{path_prefix}.shard{self.rank}of{self.world}.pt")
