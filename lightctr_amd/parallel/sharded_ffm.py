"""Hash-sharded FFM across N GPUs (BASELINE config #3: FFM, embedding
table hash-sharded, RCCL collectives over xGMI).

Same scheme as ShardedFMModel with per-(feature, field) latent blocks:
V[F, nfields, K] sharded by `fid % world`; per step the deduplicated
(fid -> [nfields, K] block) working set is exchanged with all-to-alls,
the fused local FFM step runs on a mini-table, and per-unique gradient
blocks route back to their owners. Because FFM blocks are nfields*K
floats, the wire supports fp16 payloads (reference's fp16 wire,
float16.h) — `wire="fp16"` halves the exchange bytes; fp32 is exact and
is what the equivalence test pins.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..models.ffm import FFMHyper
from ..ops import ffm_ref, fm_ref
from ..ops._extension import require_hip_ops, sort_ids


class ShardedFFMModel:
    def __init__(self, hyper: FFMHyper, device: str = "cpu", group=None,
                 wire: str = "fp32"):
        self.h = hyper
        self.device = torch.device(device)
        self.group = group
        self.wire = wire
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        F, nf, K = hyper.num_features, hyper.num_fields, hyper.k
        self.F_local = (F + self.world - 1) // self.world
        g = torch.Generator().manual_seed(hyper.seed + 17 * self.rank)
        dev = self.device
        self.W = torch.zeros(self.F_local, device=dev)
        self.V = (torch.randn(self.F_local, nf, K, generator=g)
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
        self._use_hip = dev.type == "cuda"
        # rowemit caps (mirror models/ffm.py): fall back to the sorted
        # walk when the staged emit / register accumulator cannot fit
        maxn, D = 40, nf * K
        lds_bytes = (8 * nf * (K + 1) * 4 + maxn * D * 2 + maxn * 4
                     + nf * 4 + 4)
        self._rowemit_ok = lds_bytes <= (64 << 10) and D <= 1024
        if self._use_hip:
            require_hip_ops()

    def _exchange(self, send, send_counts, recv_counts):
        if self.wire == "fp16" and send.is_floating_point():
            send = send.to(torch.float16)
        out = torch.empty((sum(recv_counts),) + tuple(send.shape[1:]),
                          dtype=send.dtype, device=send.device)
        dist.all_to_all_single(out, send.contiguous(),
                               output_split_sizes=recv_counts,
                               input_split_sizes=send_counts,
                               group=self.group)
        if out.dtype == torch.float16:
            out = out.float()
        return out

    def train_step(self, row_ptr, fields, fids, vals, labels):
        h = self.h
        B = row_ptr.numel() - 1
        world = self.world
        scale = 1.0 / (B * world)
        nf, K = h.num_fields, h.k

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
        Vl_o = self._exchange(self.V[lidx].reshape(-1, nf * K), recv_counts,
                              send_counts)
        Wl = torch.empty_like(Wl_o)
        Vl = torch.empty(U, nf, K, device=self.device)
        Wl[order] = Wl_o
        Vl.view(U, -1)[order] = Vl_o

        fids_local = inverse.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            # bf16 compute (BASELINE config #3): the gathered union table
            # is read in bf16 by the forward/emit kernels; gradients and
            # the owner-side optimizer stay fp32
            bf16 = getattr(h, "dtype", "fp32") == "bf16"
            Vc = Vl.to(torch.bfloat16) if bf16 else Vl
            pred = ops.ffm_forward(row_ptr, fields, fids_local, vals, Wl, Vc)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            gWl = torch.zeros(U, device=self.device)
            gVl = torch.zeros(U, nf, K, device=self.device)
            scratch = torch.zeros((U + 63) // 64, dtype=torch.int64,
                                  device=self.device)
            sorted_l, perm = sort_ids(fids_local, U)
            if self._rowemit_ok:
                # rowemit backward (round-2 default, 2.82 vs 3.91 ms
                # single-GPU); power-of-two scale keeps the fp16 blocks
                # away from denormal flush at large B*world
                bscale = float(1 << min(24, max(0, (B * world).bit_length()
                                               - 1)))
                gw, gblocks = ops.ffm_row_emit(row_ptr, fields, fids_local,
                                               vals, Vc, dpred,
                                               scale=bscale)
                ops.ffm_blocks_apply_f16(sorted_l, perm, gblocks, gw, gWl,
                                         gVl.view(U, -1), scratch,
                                         inv_scale=1.0 / bscale)
            else:
                row_of_entry = ops.row_index(row_ptr, fids.numel())
                ops.ffm_sorted_backward(sorted_l, perm, row_of_entry,
                                        row_ptr, fields, fids_local, vals,
                                        Vl, dpred, gWl, gVl, scratch)
        else:
            pred = ffm_ref.ffm_forward_ref(row_ptr, fields, fids_local,
                                           vals, Wl, Vl)
            loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
            gWl, gVl = ffm_ref.ffm_backward_ref(row_ptr, fields, fids_local,
                                                vals, Vl, dpred)

        gW_recv = self._exchange(gWl[order], send_counts, recv_counts)
        gV_recv = self._exchange(gVl.view(U, -1)[order], send_counts,
                                 recv_counts)
        lidx32 = lidx.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            sorted_own, perm_own = sort_ids(lidx32, self.F_local)
            # runtime-D (nf*K) segment reduce: the FFM block-apply kernel
            # (fm_sorted_apply only dispatches K in {4..64} and aborted
            # on D=312 — caught by the round-2 on-GPU shakeout)
            ops.ffm_blocks_apply(sorted_own, perm_own,
                                 gV_recv.contiguous(),
                                 gW_recv.contiguous(), self.gradW,
                                 self.gradV.view(self.F_local, -1),
                                 self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            live = self.uniq[: min(self.uniq.numel(), int(lidx32.numel()))]
            if h.optimizer == "ftrl":
                ops.sparse_ftrl_apply(live, self.count, self.W, self.V,
                                      self.zW, self.nW, self.zV, self.nV,
                                      self.gradW, self.gradV, h.ftrl_alpha,
                                      h.ftrl_beta, h.ftrl_l1, h.ftrl_l2,
                                      1 if getattr(h, "ftrl_v", "adagrad")
                                      == "adagrad" else 0, h.lr, h.eps,
                                      h.l2)
            else:
                ops.sparse_adagrad_apply(live, self.count, self.W, self.V,
                                         self.nW, self.nV, self.gradW,
                                         self.gradV, h.lr, h.eps, h.l2)
        else:
            self.gradW.index_add_(0, lidx, gW_recv)
            self.gradV.view(self.F_local, -1).index_add_(0, lidx, gV_recv)
            own = torch.unique(lidx).int()
            Fl = self.F_local
            if h.optimizer == "ftrl":
                fm_ref.ftrl_apply_ref(own, self.W, self.V.view(Fl, -1),
                                      self.zW, self.nW,
                                      self.zV.view(Fl, -1),
                                      self.nV.view(Fl, -1), self.gradW,
                                      self.gradV.view(Fl, -1),
                                      h.ftrl_alpha, h.ftrl_beta, h.ftrl_l1,
                                      h.ftrl_l2,
                                      v_adagrad=getattr(h, "ftrl_v",
                                                        "adagrad")
                                      == "adagrad", v_lr=h.lr, v_eps=h.eps,
                                      v_l2=h.l2)
            else:
                fm_ref.adagrad_apply_ref(own, self.W, self.V.view(Fl, -1),
                                         self.nW, self.nV.view(Fl, -1),
                                         self.gradW,
                                         self.gradV.view(Fl, -1), h.lr,
                                         h.eps, h.l2)
        return loss

    def save(self, path_prefix: str) -> None:
        torch.save({"W": self.W, "V": self.V, "nW": self.nW, "nV": self.nV,
                    "rank": self.rank, "world": self.world},
                   f"{path_prefix}.shard{self.rank}of{self.world}.pt")
