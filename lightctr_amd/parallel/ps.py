"""Asynchronous Parameter-Server mode (SSP + delay compensation).

Parity with the reference's PS stack
(/root/reference/LightCTR/distribut/{paramserver.h,worker.h,push.h,pull.h},
master.h): ranks [0, ps_shards) are parameter shards, the rest are workers.
Each worker trains on its own shard of data at its OWN pace (true async —
no collective barriers between workers); per batch it PULLs the touched
features' parameters from their owning shards and PUSHes gradients back.
The PS applies per-key updates as requests arrive:

  * SSP gate      — a pull from a worker more than `staleness` epochs ahead
                    of the slowest worker is refused; the worker sleeps and
                    retries (reference paramserver.h:20,127-137, pull.h:50-67)
  * stale-push drop — pushes older than `staleness` epochs vs the shard
                    clock are dropped (reference paramserver.h:189-210)
  * updaters      — sgd | adagrad | dcasgd | dcasgda, the reference's
                    delay-compensated async SGD family
                    (paramserver.h:252-300): DCASGD keeps a per-worker
                    shadow copy of each pulled weight and compensates
                    w -= lr*(g + lambda*g^2*(w - shadow));
                    DCASGDA uses an adaptive lambda/sqrt(eps+EMA(g^2))
  * wire codec    — fp32, fp16 or int8-quantile payloads (reference's
                    varint/fp16/int8 compression, network.h +
                    quantile_compress.h), chosen per PSConfig.

Transport is torch.distributed point-to-point (gloo on CPU tests; on the
GPU node NCCL(=RCCL) send/recv over xGMI). The rendezvous that the
reference's Master performs by hand (handshake, id assignment, topology
broadcast, FIN barrier — master.h:28-200) is torch.distributed init +
explicit FIN messages here.
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass

import torch
import torch.distributed as dist

OP_PULL, OP_PUSH, OP_FIN = 1, 2, 3
FLAG_RETRY = 1


@dataclass
class PSConfig:
    num_features: int
    k: int = 8
    ps_shards: int = 1
    updater: str = "dcasgd"  # sgd | adagrad | dcasgd | dcasgda
    lr: float = 0.05
    dc_lambda: float = 0.1  # reference's DCASGD lambda
    eps: float = 1e-8
    staleness: int = 10  # SSP threshold (reference kStalenessStepThreshold)
    wire: str = "fp32"  # fp32 | fp16 | int8
    init_sigma: float = 0.01
    seed: int = 1234


def _owner(fids: torch.Tensor, ps_shards: int) -> torch.Tensor:
    return fids % ps_shards


def setup_pair_groups(cfg: "PSConfig"):
    """Create one subgroup per worker: [all PS ranks] + [worker w].
    MUST be called collectively by every rank after init_process_group
    (replaces the reference Master's broadcast_topology, master.h:146-190).
    PS serving threads use per-worker groups so blocking recvs for
    different workers proceed concurrently (gloo irecv completion flags
    do not progress without wait(), so polling is not an option)."""
    world = dist.get_world_size()
    groups = {}
    for w in range(cfg.ps_shards, world):
        groups[w] = dist.new_group(list(range(cfg.ps_shards)) + [w])
    return groups


class PSShard:
    """Server loop for one parameter-shard rank."""

    def __init__(self, cfg: PSConfig, device: str = "cpu"):
        self.cfg = cfg
        self.device = torch.device(device)
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.n_workers = self.world - cfg.ps_shards
        F = (cfg.num_features + cfg.ps_shards - 1) // cfg.ps_shards
        self.F_local = F
        g = torch.Generator().manual_seed(cfg.seed + 31 * self.rank)
        K = cfg.k
        self.W = torch.zeros(F, device=self.device)
        self.V = (torch.randn(F, K, generator=g) * cfg.init_sigma).to(
            self.device)
        if cfg.updater in ("adagrad", "dcasgda"):
            self.nW = torch.zeros(F, device=self.device)
            self.nV = torch.zeros(F, K, device=self.device)
        if cfg.updater in ("dcasgd", "dcasgda"):
            self.shadowW = torch.zeros(self.n_workers, F, device=self.device)
            self.shadowV = torch.zeros(self.n_workers, F, K,
                                       device=self.device)
        self.worker_epoch = torch.zeros(self.n_workers, dtype=torch.long)
        self._codec = None
        if cfg.wire == "int8":
            from ..utils.compress import QuantileCodec

            self._codec = QuantileCodec(mode="log", device=device)

    # --- wire helpers (per-peer strictly ordered send/recv) ---
    def _recv(self, shape, dtype, src, group):
        t = torch.empty(shape, dtype=dtype, device=self.device)
        dist.recv(t, src=src, group=group)
        return t

    def serve(self, groups):
        """One serving thread per worker (blocking recv on the worker's
        pair group), like the reference Delivery's recv event-loops +
        handler pool (network.h:385-474). PULLS ARE LOCK-FREE (the
        reference PS is deliberately Hogwild on reads,
        paramserver.h:138-139: a pull may observe a value mid-update —
        async-SGD semantics absorb it); only pushes serialize against
        each other under a lock so concurrent same-key updates do not
        interleave their read-modify-writes. Returns when all workers
        FIN."""
        cfg = self.cfg
        self._lock = threading.Lock()
        workers = list(range(cfg.ps_shards, self.world))
        threads = [threading.Thread(target=self._serve_one,
                                    args=(w, groups[w]), daemon=True)
                   for w in workers]
        for t in threads:
            t.start()
        for t in threads:
            t.join()

    def _serve_one(self, w, group):
        cfg = self.cfg
        wi = w - cfg.ps_shards
        while True:
            hdr = torch.zeros(4, dtype=torch.long, device=self.device)
            dist.recv(hdr, src=w, group=group)
            op, epoch, n, flags = [int(v) for v in hdr.cpu()]
            if op == OP_FIN:
                return
            if op == OP_PULL:
                self._handle_pull(w, wi, epoch, n, group)
            elif op == OP_PUSH:
                self._handle_push(w, wi, epoch, n, flags, group)

    def _handle_pull(self, w, wi, epoch, n, group):
        cfg = self.cfg
        fids = self._recv((n,), torch.long, w, group)
        self.worker_epoch[wi] = epoch
        fastest_allowed = int(self.worker_epoch[
            self.worker_epoch >= 0].min()) + cfg.staleness
        reply = torch.zeros(4, dtype=torch.long, device=self.device)
        if epoch > fastest_allowed:
            reply[3] = FLAG_RETRY  # SSP gate: refuse, worker retries
            dist.send(reply, dst=w, group=group)
            return
        dist.send(reply, dst=w, group=group)
        lidx = fids // cfg.ps_shards
        # Hogwild read — no lock (see serve() docstring). The shadow
        # copies are per-worker slices written only by this worker's
        # serving thread, so they need no lock either.
        Wv = self.W[lidx].clone()
        Vv = self.V[lidx].clone()
        if cfg.updater in ("dcasgd", "dcasgda"):
            self.shadowW[wi, lidx] = Wv
            self.shadowV[wi, lidx] = Vv
        if cfg.wire == "fp16":
            dist.send(Wv.to(torch.float16), dst=w, group=group)
            dist.send(Vv.to(torch.float16), dst=w, group=group)
        else:  # params always fp32/fp16 (int8 is for gradients)
            dist.send(Wv, dst=w, group=group)
            dist.send(Vv, dst=w, group=group)

    def _handle_push(self, w, wi, epoch, n, flags, group):
        cfg = self.cfg
        fids = self._recv((n,), torch.long, w, group)
        K = cfg.k
        if cfg.wire == "int8" and flags == 1:
            scale = self._recv((1,), torch.float32, w, group)
            cw = self._recv((n,), torch.uint8, w, group)
            cv = self._recv((n, K), torch.uint8, w, group)
            gW = self._codec.decode(cw) * scale
            gV = self._codec.decode(cv) * scale
        elif cfg.wire == "fp16":
            gW = self._recv((n,), torch.float16, w, group).float()
            gV = self._recv((n, K), torch.float16, w, group).float()
        else:
            gW = self._recv((n,), torch.float32, w, group)
            gV = self._recv((n, K), torch.float32, w, group)
        # wire validity check (reference paramserver.h:172,244)
        if not bool(torch.isfinite(gW).all() and torch.isfinite(gV).all()):
            return  # drop corrupt push rather than poison the table
        # stale-push drop (reference paramserver.h:199-208)
        if epoch + cfg.staleness < int(self.worker_epoch.max()):
            return
        lidx = fids // cfg.ps_shards
        with self._lock:
            self._apply(wi, lidx, gW, gV)

    def _apply(self, wi, lidx, gW, gV):
        cfg = self.cfg
        lr = cfg.lr
        if self.device.type == "cuda":
            # fused per-shard updater kernel (GPU-resident state; reference
            # paramserver.h:217-306 semantics)
            from ..ops._extension import require_hip_ops

            ops = require_hip_ops()
            upd = {"sgd": 0, "adagrad": 1, "dcasgd": 2, "dcasgda": 3}[
                cfg.updater]
            ops.ps_apply(
                lidx, gW.contiguous(), gV.contiguous(), self.W, self.V,
                getattr(self, "nW", None), getattr(self, "nV", None),
                self.shadowW[wi] if hasattr(self, "shadowW") else None,
                self.shadowV[wi] if hasattr(self, "shadowV") else None,
                upd, lr, cfg.dc_lambda, cfg.eps)
            return
        if cfg.updater == "sgd":
            self.W.index_add_(0, lidx, -lr * gW)
            self.V.index_add_(0, lidx, -lr * gV)
        elif cfg.updater == "adagrad":
            self.nW[lidx] += gW * gW
            self.nV[lidx] += gV * gV
            self.W[lidx] -= lr * gW / (self.nW[lidx] + cfg.eps).sqrt()
            self.V[lidx] -= lr * gV / (self.nV[lidx] + cfg.eps).sqrt()
        elif cfg.updater in ("dcasgd", "dcasgda"):
            lamW = lamV = cfg.dc_lambda
            if cfg.updater == "dcasgda":
                self.nW[lidx] = 0.95 * self.nW[lidx] + 0.05 * gW * gW
                self.nV[lidx] = 0.95 * self.nV[lidx] + 0.05 * gV * gV
                lamW = cfg.dc_lambda / (self.nW[lidx] + cfg.eps).sqrt()
                lamV = cfg.dc_lambda / (self.nV[lidx] + cfg.eps).sqrt()
            Wc = self.W[lidx]
            Vc = self.V[lidx]
            newW = Wc - lr * (gW + lamW * gW * gW
                              * (Wc - self.shadowW[wi, lidx]))
            newV = Vc - lr * (gV + lamV * gV * gV
                              * (Vc - self.shadowV[wi, lidx]))
            self.W[lidx] = newW
            self.V[lidx] = newV
            self.shadowW[wi, lidx] = newW
            self.shadowV[wi, lidx] = newV


class PSWorker:
    """Client ops: pull / push / fin against all shards."""

    def __init__(self, cfg: PSConfig, group, device: str = "cpu"):
        self.cfg = cfg
        self.group = group  # this worker's pair group from setup_pair_groups
        self.device = torch.device(device)
        self.rank = dist.get_rank()
        self._codec = None
        if cfg.wire == "int8":
            from ..utils.compress import QuantileCodec

            self._codec = QuantileCodec(mode="log", device=device)

    def pull(self, fids_uniq: torch.Tensor, epoch: int):
        """fids_uniq: int64 sorted unique. Returns (W[U], V[U,K]) in the
        same order. Retries on SSP refusal (50 ms, like pull.h:63-67)."""
        cfg = self.cfg
        owner = _owner(fids_uniq, cfg.ps_shards)
        Wout = torch.empty(fids_uniq.numel(), device=self.device)
        Vout = torch.empty(fids_uniq.numel(), cfg.k, device=self.device)
        for ps in range(cfg.ps_shards):
            sel = owner == ps
            f = fids_uniq[sel]
            if f.numel() == 0:
                continue
            while True:
                hdr = torch.tensor([OP_PULL, epoch, f.numel(), 0],
                                   dtype=torch.long, device=self.device)
                dist.send(hdr, dst=ps, group=self.group)
                dist.send(f, dst=ps, group=self.group)
                reply = torch.zeros(4, dtype=torch.long, device=self.device)
                dist.recv(reply, src=ps, group=self.group)
                if int(reply[3]) != FLAG_RETRY:
                    break
                time.sleep(0.05)
            if cfg.wire == "fp16":
                Wv = torch.empty(f.numel(), dtype=torch.float16,
                                 device=self.device)
                Vv = torch.empty(f.numel(), cfg.k, dtype=torch.float16,
                                 device=self.device)
                dist.recv(Wv, src=ps, group=self.group)
                dist.recv(Vv, src=ps, group=self.group)
                Wout[sel] = Wv.float()
                Vout[sel] = Vv.float()
            else:
                Wv = torch.empty(f.numel(), device=self.device)
                Vv = torch.empty(f.numel(), cfg.k, device=self.device)
                dist.recv(Wv, src=ps, group=self.group)
                dist.recv(Vv, src=ps, group=self.group)
                Wout[sel] = Wv
                Vout[sel] = Vv
        return Wout, Vout

    def push(self, fids_uniq: torch.Tensor, gW: torch.Tensor,
             gV: torch.Tensor, epoch: int):
        cfg = self.cfg
        # "preferred value" gradient filter (reference push.h:60-71 +
        # distributed_algo_abst.h:76-79): drop features whose grads are all
        # ~0 (nothing to send) or contain non-finite / >15-magnitude values
        mag = torch.maximum(gW.abs(), gV.abs().amax(dim=1))
        ok = torch.isfinite(gW) & torch.isfinite(gV).all(dim=1) \
            & (mag > 1e-12) & (mag < 15.0)
        if not bool(ok.all()):
            fids_uniq, gW, gV = fids_uniq[ok], gW[ok], gV[ok]
        from ..utils.trace import debug_log

        debug_log(f"push: {fids_uniq.numel()} keys epoch={epoch}")
        owner = _owner(fids_uniq, cfg.ps_shards)
        for ps in range(cfg.ps_shards):
            sel = owner == ps
            f = fids_uniq[sel]
            if f.numel() == 0:
                continue
            use_int8 = cfg.wire == "int8"
            hdr = torch.tensor([OP_PUSH, epoch, f.numel(),
                                1 if use_int8 else 0],
                               dtype=torch.long, device=self.device)
            dist.send(hdr, dst=ps, group=self.group)
            dist.send(f, dst=ps, group=self.group)
            if use_int8:
                gw, gv = gW[sel], gV[sel]
                scale = torch.maximum(gw.abs().max(), gv.abs().max()) \
                    .reshape(1).clamp(min=1e-12)
                dist.send(scale, dst=ps, group=self.group)
                dist.send(self._codec.encode(gw / scale), dst=ps,
                          group=self.group)
                dist.send(self._codec.encode(gv / scale), dst=ps,
                          group=self.group)
            elif cfg.wire == "fp16":
                dist.send(gW[sel].to(torch.float16), dst=ps,
                          group=self.group)
                dist.send(gV[sel].to(torch.float16), dst=ps,
                          group=self.group)
            else:
                dist.send(gW[sel].contiguous(), dst=ps, group=self.group)
                dist.send(gV[sel].contiguous(), dst=ps, group=self.group)

    def fin(self):
        for ps in range(self.cfg.ps_shards):
            hdr = torch.tensor([OP_FIN, 0, 0, 0], dtype=torch.long,
                               device=self.device)
            dist.send(hdr, dst=ps, group=self.group)


def ps_train_fm(cfg: PSConfig, group, gen_batch, steps: int,
                batch_size: int, device: str = "cpu",
                epoch_per_step: int = 1, fin: bool = True,
                epoch_base: int = 0):
    """Worker-side FM training loop against the PS (reference
    Distributed_Algo_Abst::Train / batchGradCompute shape,
    distributed_algo_abst.h:130-280). gen_batch(step) -> CSR batch.
    Returns list of per-step mean losses."""
    from ..ops import fm_ref
    from ..ops._extension import require_hip_ops, sort_ids

    worker = PSWorker(cfg, group, device=device)
    losses = []
    use_hip = torch.device(device).type == "cuda"
    for step in range(steps):
        epoch = epoch_base + step * epoch_per_step
        row_ptr, fids, vals, labels = gen_batch(step)
        uniq, inverse = torch.unique(fids.long(), return_inverse=True)
        Wl, Vl = worker.pull(uniq, epoch)
        fids_local = inverse.to(torch.int32)
        scale = 1.0 / (row_ptr.numel() - 1)
        if use_hip:
            ops = require_hip_ops()
            pred, sumVX = ops.fm_forward(row_ptr, fids_local, vals, Wl, Vl)
            loss, dpred = ops.logloss_grad(pred, labels, scale)
            U = uniq.numel()
            gw, gv = ops.fm_backward_emit(row_ptr, fids_local, vals, Vl,
                                          sumVX, dpred)
            sorted_l, perm = sort_ids(fids_local, U)
            gWl = torch.zeros(U, device=device)
            gVl = torch.zeros(U, cfg.k, device=device)
            bitmap = torch.zeros((U + 63) // 64, dtype=torch.int64,
                                 device=device)
            ops.fm_sorted_apply(sorted_l, perm, gw, gv, gWl, gVl, bitmap)
        else:
            pred, sumVX = fm_ref.fm_forward_ref(row_ptr, fids_local, vals,
                                                Wl, Vl)
            loss, dpred = fm_ref.logloss_grad_ref(pred, labels, scale)
            gWl, gVl = fm_ref.fm_backward_ref(row_ptr, fids_local, vals, Vl,
                                              sumVX, dpred)
        worker.push(uniq, gWl, gVl, epoch)
        losses.append(float(loss.mean()))
    if fin:
        worker.fin()
    return losses
