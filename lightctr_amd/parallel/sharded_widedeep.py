"""Hash-sharded Wide&Deep across N GPUs (BASELINE config #4 at 8×MI355X).

Hybrid parallelism, MI355X-native:
  * the wide weights W and embedding table E are sharded by `fid % world`
    across ranks' HBM and exchanged per step with deduplicated RCCL
    all-to-alls (exactly the ShardedFMModel scheme — the PS/DHT
    replacement, see parallel/sharded_fm.py)
  * the dense MLP is replicated; its gradients are SUM-all-reduced
    (ring path) so every rank applies the identical union-batch update
    (replicas start identical by construction: same seed).

Equivalence to a single-GPU WideDeepModel trained on the union batch is
pinned by tests/test_parallel.py::test_sharded_widedeep_matches_single.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..models.mlp import MLP
from ..models.wide_deep import WideDeepHyper
from ..ops import fm_ref
from ..ops._extension import require_hip_ops, sort_ids
from .ring import _flatten_into, _unflatten_from


def _allreduce_sum(tensors, group=None):
    tensors = [t for t in tensors if t is not None]
    if not tensors:
        return
    total = sum(t.numel() for t in tensors)
    bucket = torch.empty(total, dtype=tensors[0].dtype,
                         device=tensors[0].device)
    _flatten_into(bucket, tensors)
    dist.all_reduce(bucket, op=dist.ReduceOp.SUM, group=group)
    _unflatten_from(bucket, tensors)


class ShardedWideDeepModel:
    def __init__(self, hyper: WideDeepHyper, device: str = "cpu",
                 group=None):
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
        self.E = (torch.randn(self.F_local, K, generator=g)
                  * hyper.init_sigma).to(dev)
        self.gradW = torch.zeros_like(self.W)
        self.gradE = torch.zeros_like(self.E)
        self.nW = torch.zeros_like(self.W)
        self.nE = torch.zeros_like(self.E)
        if hyper.optimizer == "ftrl":
            self.zW = torch.zeros_like(self.W)
            self.zE = torch.zeros_like(self.E)
        nwords = (self.F_local + 63) // 64
        self.touched = torch.zeros(nwords, dtype=torch.int64, device=dev)
        self.uniq = torch.zeros(self.F_local, dtype=torch.int32, device=dev)
        self.count = torch.zeros(1, dtype=torch.int32, device=dev)
        # replicated MLP: identical init on every rank (same seed)
        dims = [hyper.num_fields * K, *hyper.hidden, 1]
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
        El_o = self._exchange(self.E[lidx], recv_counts, send_counts)
        Wl = torch.empty_like(Wl_o)
        El = torch.empty_like(El_o)
        Wl[order] = Wl_o
        El[order] = El_o

        fids_local = inverse.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            wide = ops.wide_forward(row_ptr, fids_local, vals, Wl)
            deep_in = ops.embed_gather(row_ptr, fids_local, vals, El,
                                       h.num_fields)
            deep = self.mlp.forward(deep_in, train=True)
            pred = wide + deep[:, 0]
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            dDeep = self.mlp.backward(dpred.unsqueeze(1))
            gw, gv = ops.embed_backward_emit(row_ptr, vals,
                                             dDeep.contiguous(), dpred,
                                             h.num_fields, h.k)
            sorted_l, perm = sort_ids(fids_local, U)
            gWl = torch.zeros(U, device=self.device)
            gEl = torch.zeros(U, h.k, device=self.device)
            scratch = torch.zeros((U + 63) // 64, dtype=torch.int64,
                                  device=self.device)
            ops.fm_sorted_apply(sorted_l, perm, gw, gv, gWl, gEl, scratch)
        else:
            rp = row_ptr.long()
            row_idx = torch.repeat_interleave(torch.arange(B),
                                              rp[1:] - rp[:-1])
            pos = torch.arange(fids.numel()) - rp[:-1][row_idx]
            wide = torch.zeros(B)
            wide.index_add_(0, row_idx, Wl[inverse] * vals)
            nf, K = h.num_fields, h.k
            deep_in = torch.zeros(B, nf * K)
            keep = pos < nf
            emb = El[inverse] * vals.unsqueeze(1)
            flat = (row_idx * nf + pos).clamp(max=B * nf - 1)
            deep_in.view(B * nf, K).index_add_(0, flat[keep], emb[keep])
            deep = self.mlp.forward(deep_in, train=True)
            pred = wide + deep[:, 0]
            loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
            dDeep = self.mlp.backward(dpred.unsqueeze(1))
            dE = dDeep.view(B, nf, K)[row_idx, pos.clamp(max=nf - 1)] \
                * vals.unsqueeze(1)
            dE = torch.where(keep.unsqueeze(1), dE, torch.zeros_like(dE))
            gWl = torch.zeros(U)
            gWl.index_add_(0, inverse, dpred[row_idx] * vals)
            gEl = torch.zeros(U, K)
            gEl.index_add_(0, inverse, dE)

        # replicated MLP: union-batch gradient = SUM over ranks
        grads = []
        for la in self.mlp.layers:
            grads += [la._dW, la._db]
        _allreduce_sum(grads, group=self.group)
        self.mlp.apply_grads()

        # sparse side: route per-unique grads to owners, owners apply
        gW_recv = self._exchange(gWl[order], send_counts, recv_counts)
        gE_recv = self._exchange(gEl[order], send_counts, recv_counts)
        lidx32 = lidx.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            sorted_own, perm_own = sort_ids(lidx32, self.F_local)
            ops.fm_sorted_apply(sorted_own, perm_own, gW_recv.contiguous(),
                                gE_recv.contiguous(), self.gradW, self.gradE,
                                self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            live = self.uniq[: min(self.uniq.numel(), int(lidx32.numel()))]
            if h.optimizer == "ftrl":
                ops.fm_ftrl_apply(live, self.count, self.W, self.E, self.zW,
                                  self.nW, self.zE, self.nE, self.gradW,
                                  self.gradE, h.ftrl_alpha, h.ftrl_beta,
                                  h.ftrl_l1, h.ftrl_l2)
            else:
                ops.fm_adagrad_apply(live, self.count, self.W, self.E,
                                     self.nW, self.nE, self.gradW,
                                     self.gradE, h.lr, h.eps, h.l2)
        else:
            self.gradW.index_add_(0, lidx, gW_recv)
            self.gradE.index_add_(0, lidx, gE_recv)
            own = torch.unique(lidx).int()
            if h.optimizer == "ftrl":
                fm_ref.ftrl_apply_ref(own, self.W, self.E, self.zW, self.nW,
                                      self.zE, self.nE, self.gradW,
                                      self.gradE, h.ftrl_alpha, h.ftrl_beta,
                                      h.ftrl_l1, h.ftrl_l2)
            else:
                fm_ref.adagrad_apply_ref(own, self.W, self.E, self.nW,
                                         self.nE, self.gradW, self.gradE,
                                         h.lr, h.eps, h.l2)
        return loss

    def save(self, path_prefix: str) -> None:
        torch.save({"W": self.W, "E": self.E, "nW": self.nW, "nE": self.nE,
                    "mlp": self.mlp.state_dict(), "rank": self.rank,
                    "world": self.world},
                   f"{path_prefix}.shard{self.rank}of{self.world}.pt")
