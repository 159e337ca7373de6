"""Hash-sharded FM across N GPUs — the MI355X-native replacement of the
reference's Parameter-Server + ConsistentHash DHT
(/root/reference/LightCTR/distribut/{paramserver.h,consistent_hash.h,
push.h,pull.h}): instead of DHT-over-ZeroMQ, the feature table lives
sharded by `fid % world` across the HBM of all ranks, and each train step
exchanges deduplicated (fid -> embedding row) requests and gradients with
two RCCL all-to-alls over xGMI (the "batched parameter request" the
reference README describes, without per-row round trips).

Every rank is simultaneously worker AND parameter shard (the reference's
4 PS + 4 worker split maps onto 8 ranks each playing both roles — same
aggregate layout, no idle GPUs). Synchronous flavor: gradient exchange
completes before the owner applies its fused optimizer, so training is
exactly minibatch SGD/FTRL on the union batch (the async SSP/DCASGD
semantics of the reference live in lightctr_amd.parallel.ps).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..models.fm import FMHyper
from ..ops import fm_ref
from ..ops._extension import require_hip_ops, sort_ids


class ShardedFMModel:
    def __init__(self, hyper: FMHyper, device: str = "cpu", group=None,
                 wire: str = "fp32"):
        """wire="fp16" sends parameter/gradient payloads in half precision
        (the reference's fp16 PS wire, float16.h), halving xGMI bytes;
        fp32 is exact and is what the equivalence tests pin."""
        self.h = hyper
        self.device = torch.device(device)
        self.group = group
        self.wire = wire
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        F, K = hyper.num_features, hyper.k
        self.F_local = (F + self.world - 1) // self.world
        g = torch.Generator().manual_seed(hyper.seed + 17 * self.rank)
        dev = self.device
        # shard tensors: local row i holds global feature i*world + rank
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
        self._use_hip = dev.type == "cuda"
        if self._use_hip:
            require_hip_ops()

    # ---- parameter exchange ----
    def _exchange(self, send: torch.Tensor, send_counts, recv_counts):
        """all_to_all_single with known splits; send/recv along dim 0."""
        if self.wire == "fp16" and send.is_floating_point():
            send = send.to(torch.float16)
        out_shape = (sum(recv_counts),) + tuple(send.shape[1:])
        out = torch.empty(out_shape, dtype=send.dtype, device=send.device)
        dist.all_to_all_single(out, send.contiguous(),
                               output_split_sizes=recv_counts,
                               input_split_sizes=send_counts,
                               group=self.group)
        if out.dtype == torch.float16:
            out = out.float()
        return out

    def train_step(self, row_ptr, fids, vals, labels) -> torch.Tensor:
        """One synchronous sharded step on this rank's local batch."""
        h = self.h
        B = row_ptr.numel() - 1
        world = self.world
        # global mean over the union batch (all ranks' batches)
        scale = 1.0 / (B * world)

        # 1. dedup local batch; express batch in local mini-table ids
        uniq, inverse = torch.unique(fids, return_inverse=True)
        U = uniq.numel()
        owner = (uniq.long() % world)
        order = torch.argsort(owner, stable=True)
        uniq_o = uniq[order]
        cnt_t = torch.bincount(owner, minlength=world).to(self.device)
        send_counts = cnt_t.cpu().tolist()
        recv_cnt_t = torch.empty(world, dtype=cnt_t.dtype,
                                 device=self.device)
        dist.all_to_all_single(recv_cnt_t, cnt_t, group=self.group)
        recv_counts = recv_cnt_t.cpu().tolist()

        # 2. exchange requested fids; owners gather their shard rows
        req = self._exchange(uniq_o, send_counts, recv_counts)
        lidx = (req.long() // world)
        Wv = self.W[lidx]
        Vv = self.V[lidx]

        # 3. send values back (reverse splits) and un-permute to uniq order
        Wl_o = self._exchange(Wv, recv_counts, send_counts)
        Vl_o = self._exchange(Vv, recv_counts, send_counts)
        Wl = torch.empty_like(Wl_o)
        Vl = torch.empty_like(Vl_o)
        Wl[order] = Wl_o
        Vl[order] = Vl_o

        # 4. local fused compute on the mini-table
        fids_local = inverse.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            pred, sumVX = ops.fm_forward(row_ptr, fids_local, vals, Wl, Vl)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            gw, gv = ops.fm_backward_emit(row_ptr, fids_local, vals, Vl,
                                          sumVX, dpred)
            sorted_l, perm = sort_ids(fids_local, U)
            gWl = torch.zeros(U, device=self.device)
            gVl = torch.zeros(U, h.k, device=self.device)
            scratch_bitmap = torch.zeros((U + 63) // 64, dtype=torch.int64,
                                         device=self.device)
            ops.fm_sorted_apply(sorted_l, perm, gw, gv, gWl, gVl,
                                scratch_bitmap)
        else:
            pred, sumVX = fm_ref.fm_forward_ref(row_ptr, fids_local, vals,
                                                Wl, Vl)
            loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
            gWl, gVl = fm_ref.fm_backward_ref(row_ptr, fids_local, vals, Vl,
                                              sumVX, dpred)

        # 5. route gradients to owners (same splits as the request)
        gW_recv = self._exchange(gWl[order], send_counts, recv_counts)
        gV_recv = self._exchange(gVl[order], send_counts, recv_counts)

        # 6. owner accumulates (features may arrive from several ranks) and
        #    applies the fused sparse optimizer on its shard
        lidx32 = lidx.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            sorted_own, perm_own = sort_ids(lidx32, self.F_local)
            ops.fm_sorted_apply(sorted_own, perm_own, gW_recv.contiguous(),
                                gV_recv.contiguous(), self.gradW, self.gradV,
                                self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            if h.optimizer == "ftrl":
                ops.fm_ftrl_apply(self.uniq, self.count, self.W, self.V,
                                  self.zW, self.nW, self.zV, self.nV,
                                  self.gradW, self.gradV, h.ftrl_alpha,
                                  h.ftrl_beta, h.ftrl_l1, h.ftrl_l2,
                                  1 if h.ftrl_v == "adagrad" else 0, h.lr,
                                  h.eps, h.l2)
            else:
                ops.fm_adagrad_apply(self.uniq, self.count, self.W, self.V,
                                     self.nW, self.nV, self.gradW, self.gradV,
                                     h.lr, h.eps, h.l2)
        else:
            self.gradW.index_add_(0, lidx, gW_recv)
            self.gradV.index_add_(0, lidx, gV_recv)
            own_uniq = torch.unique(lidx).int()
            if h.optimizer == "ftrl":
                fm_ref.ftrl_apply_ref(own_uniq, self.W, self.V, self.zW,
                                      self.nW, self.zV, self.nV, self.gradW,
                                      self.gradV, h.ftrl_alpha, h.ftrl_beta,
                                      h.ftrl_l1, h.ftrl_l2,
                                      v_adagrad=h.ftrl_v == "adagrad",
                                      v_lr=h.lr, v_eps=h.eps, v_l2=h.l2)
            else:
                fm_ref.adagrad_apply_ref(own_uniq, self.W, self.V, self.nW,
                                         self.nV, self.gradW, self.gradV,
                                         h.lr, h.eps, h.l2)
        return loss

    # ---- SSP-1 pipelined stepping ------------------------------------
    # The reference's async PS tolerates bounded staleness (SSP); the
    # pipelined flavor prefetches batch t+1's parameter working set while
    # step t is still computing/pushing, so the value all-to-all overlaps
    # compute at the cost of params being one optimizer step stale
    # (staleness 1 << the reference's threshold of 10). Enable by calling
    # prefetch() for the next batch before finishing the current step, or
    # just call train_step_pipelined per batch.

    def _pull_issue(self, fids):
        """Dedup + routing + issue the value exchanges with async_op; the
        returned state is finished by _pull_wait."""
        world = self.world
        uniq, inverse = torch.unique(fids, return_inverse=True)
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
        Wv = self.W[lidx].contiguous()
        Vv = self.V[lidx].contiguous()
        if self.wire == "fp16":
            Wv = Wv.to(torch.float16)
            Vv = Vv.to(torch.float16)
        Wl_o = torch.empty((sum(send_counts),), dtype=Wv.dtype,
                           device=self.device)
        Vl_o = torch.empty((sum(send_counts), self.h.k), dtype=Vv.dtype,
                           device=self.device)
        hW = dist.all_to_all_single(Wl_o, Wv,
                                    output_split_sizes=send_counts,
                                    input_split_sizes=recv_counts,
                                    group=self.group, async_op=True)
        hV = dist.all_to_all_single(Vl_o, Vv,
                                    output_split_sizes=send_counts,
                                    input_split_sizes=recv_counts,
                                    group=self.group, async_op=True)
        return {"uniq": uniq, "inverse": inverse, "order": order,
                "lidx": lidx, "send_counts": send_counts,
                "recv_counts": recv_counts, "Wl_o": Wl_o, "Vl_o": Vl_o,
                "hW": hW, "hV": hV}

    def _pull_wait(self, st):
        st["hW"].wait()
        st["hV"].wait()
        U = st["uniq"].numel()
        Wl = torch.empty(U, device=self.device)
        Vl = torch.empty(U, self.h.k, device=self.device)
        Wl[st["order"]] = st["Wl_o"].float()
        Vl[st["order"]] = st["Vl_o"].float()
        return Wl, Vl

    def prefetch(self, fids):
        """Issue the parameter pull for an upcoming batch (params as of the
        LAST completed optimizer step — SSP staleness 1)."""
        self._prefetched = self._pull_issue(fids)

    def train_step_pipelined(self, row_ptr, fids, vals, labels,
                             next_fids=None):
        """Step with prefetched params; issues next_fids' pull right after
        this step's local compute so the value exchange overlaps the grad
        routing + owner apply."""
        h = self.h
        B = row_ptr.numel() - 1
        st = getattr(self, "_prefetched", None)
        if st is None:
            st = self._pull_issue(fids)
        self._prefetched = None
        Wl, Vl = self._pull_wait(st)
        scale = 1.0 / (B * self.world)
        fids_local = st["inverse"].to(torch.int32)
        U = st["uniq"].numel()
        if self._use_hip:
            ops = require_hip_ops()
            pred, sumVX = ops.fm_forward(row_ptr, fids_local, vals, Wl, Vl)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            gw, gv = ops.fm_backward_emit(row_ptr, fids_local, vals, Vl,
                                          sumVX, dpred)
            sorted_l, perm = sort_ids(fids_local, U)
            gWl = torch.zeros(U, device=self.device)
            gVl = torch.zeros(U, h.k, device=self.device)
            scratch = torch.zeros((U + 63) // 64, dtype=torch.int64,
                                  device=self.device)
            ops.fm_sorted_apply(sorted_l, perm, gw, gv, gWl, gVl, scratch)
        else:
            pred, sumVX = fm_ref.fm_forward_ref(row_ptr, fids_local, vals,
                                                Wl, Vl)
            loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
            gWl, gVl = fm_ref.fm_backward_ref(row_ptr, fids_local, vals, Vl,
                                              sumVX, dpred)
        if next_fids is not None:
            # overlap: next batch's pull (stale by this step's update)
            self.prefetch(next_fids)
        self._route_and_apply(st, gWl, gVl)
        return loss

    def _route_and_apply(self, st, gWl, gVl):
        h = self.h
        gW_recv = self._exchange(gWl[st["order"]], st["send_counts"],
                                 st["recv_counts"])
        gV_recv = self._exchange(gVl[st["order"]], st["send_counts"],
                                 st["recv_counts"])
        lidx = st["lidx"]
        lidx32 = lidx.to(torch.int32)
        if self._use_hip:
            ops = require_hip_ops()
            sorted_own, perm_own = sort_ids(lidx32, self.F_local)
            ops.fm_sorted_apply(sorted_own, perm_own, gW_recv.contiguous(),
                                gV_recv.contiguous(), self.gradW, self.gradV,
                                self.touched)
            self.count.zero_()
            ops.bitmap_compact(self.touched, self.uniq, self.count)
            if h.optimizer == "ftrl":
                ops.fm_ftrl_apply(self.uniq, self.count, self.W, self.V,
                                  self.zW, self.nW, self.zV, self.nV,
                                  self.gradW, self.gradV, h.ftrl_alpha,
                                  h.ftrl_beta, h.ftrl_l1, h.ftrl_l2,
                                  1 if h.ftrl_v == "adagrad" else 0, h.lr,
                                  h.eps, h.l2)
            else:
                ops.fm_adagrad_apply(self.uniq, self.count, self.W, self.V,
                                     self.nW, self.nV, self.gradW,
                                     self.gradV, h.lr, h.eps, h.l2)
        else:
            self.gradW.index_add_(0, lidx, gW_recv)
            self.gradV.index_add_(0, lidx, gV_recv)
            own_uniq = torch.unique(lidx).int()
            if h.optimizer == "ftrl":
                fm_ref.ftrl_apply_ref(own_uniq, self.W, self.V, self.zW,
                                      self.nW, self.zV, self.nV, self.gradW,
                                      self.gradV, h.ftrl_alpha, h.ftrl_beta,
                                      h.ftrl_l1, h.ftrl_l2,
                                      v_adagrad=h.ftrl_v == "adagrad",
                                      v_lr=h.lr, v_eps=h.eps, v_l2=h.l2)
            else:
                fm_ref.adagrad_apply_ref(own_uniq, self.W, self.V, self.nW,
                                         self.nV, self.gradW, self.gradV,
                                         h.lr, h.eps, h.l2)

    # ---- inference (pull-only) ----
    def predict_proba(self, row_ptr, fids, vals):
        world = self.world
        uniq, inverse = torch.unique(fids, return_inverse=True)
        owner = (uniq.long() % world)
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
            pred, _ = ops.fm_forward(row_ptr, fids_local, vals, Wl, Vl)
        else:
            pred, _ = fm_ref.fm_forward_ref(row_ptr, fids_local, vals, Wl, Vl)
        return torch.sigmoid(torch.clamp(pred, -16, 16))

    # ---- sharded checkpoint (one file per rank shard) ----
    def save(self, path_prefix: str) -> None:
        d = {"W": self.W, "V": self.V, "nW": self.nW, "nV": self.nV,
             "rank": self.rank, "world": self.world, "hyper": self.h.__dict__}
        if self.h.optimizer == "ftrl":
            d["zW"], d["zV"] = self.zW, self.zV
        torch.save(d, f"{path_prefix}.shard{self.rank}of{self.world}.pt")

    def load(self, path_prefix: str) -> None:
        d = torch.load(f"{path_prefix}.shard{self.rank}of{self.world}.pt",
                       map_location=self.device, weights_only=True)
        assert d["world"] == self.world, "shard layout mismatch"
        self.W.copy_(d["W"]); self.V.copy_(d["V"])
        self.nW.copy_(d["nW"]); self.nV.copy_(d["nV"])
        if self.h.optimizer == "ftrl" and "zW" in d:
            self.zW.copy_(d["zW"]); self.zV.copy_(d["zV"])
