"""Multi-process (gloo, world=2) tests of the distributed layer on CPU.

The same code paths run over RCCL on the GPU node (bench.py N>1); these
tests pin down the collective algebra: sharded all-to-all FM step ==
single-model step on the union batch; ring allreduce/broadcast semantics.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

WORLD = 2


def _init(rank, port):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)
    return dist


def _sharded_fm_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        from lightctr_amd.models.fm import FMHyper, FMModel
        from lightctr_amd.parallel.sharded_fm import ShardedFMModel
        from conftest import make_random_csr

        F, K = 1000, 8
        h = FMHyper(num_features=F, k=K, optimizer="adagrad", seed=99)
        sharded = ShardedFMModel(h, device="cpu")
        # local batch per rank
        row_ptr, fids, vals, labels = make_random_csr(
            B=64, F_total=F, seed=100 + rank, binary_vals=False)
        loss = sharded.train_step(row_ptr, fids, vals, labels)
        assert torch.isfinite(loss).all()

        # reconstruct full updated table from both shards
        W_full = torch.zeros(F)
        V_full = torch.zeros(F, K)
        gathered_W = [torch.zeros_like(sharded.W) for _ in range(WORLD)]
        gathered_V = [torch.zeros_like(sharded.V) for _ in range(WORLD)]
        dist.all_gather(gathered_W, sharded.W)
        dist.all_gather(gathered_V, sharded.V)
        for r in range(WORLD):
            idx = torch.arange(r, F, WORLD)
            W_full[idx] = gathered_W[r][: idx.numel()]
            V_full[idx] = gathered_V[r][: idx.numel()]

        if rank == 0:
            # single-model equivalent: union batch, scale 1/(B*world)
            single = FMModel(h, device="cpu")
            # initialize single model from the SHARD inits
            for r in range(WORLD):
                g = torch.Generator().manual_seed(h.seed + 17 * r)
                Fl = (F + WORLD - 1) // WORLD
                Vr = torch.randn(Fl, K, generator=g) * h.init_sigma
                idx = torch.arange(r, F, WORLD)
                single.V[idx] = Vr[: idx.numel()]
                single.W[idx] = 0.0
            batches = [make_random_csr(B=64, F_total=F, seed=100 + r,
                                       binary_vals=False) for r in range(WORLD)]
            # union batch = concat rows
            import torch as t

            rp0, f0, v0, l0 = batches[0]
            rp1, f1, v1, l1 = batches[1]
            rp = t.cat([rp0[:-1], rp1 + rp0[-1]])
            fids_u = t.cat([f0, f1])
            vals_u = t.cat([v0, v1])
            labels_u = t.cat([l0, l1])
            # train_step uses scale 1/B_union = 1/(64*2): matches sharded
            single.train_step(rp, fids_u, vals_u, labels_u)
            ok_w = torch.allclose(W_full, single.W, atol=1e-5)
            ok_v = torch.allclose(V_full, single.V, atol=1e-5)
            q.put(("result", ok_w, ok_v,
                   float((W_full - single.W).abs().max()),
                   float((V_full - single.V).abs().max())))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def _ring_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        from lightctr_amd.parallel.ring import (allreduce_gradients,
                                                broadcast_params)

        p1 = torch.full((10,), float(rank))
        p2 = torch.full((5,), float(rank * 2))
        broadcast_params([p1, p2], src=0)
        assert p1.sum() == 0 and p2.sum() == 0  # rank0's values

        g1 = torch.full((1000,), float(rank + 1))
        g2 = torch.full((3,), float(10 * (rank + 1)))
        allreduce_gradients([g1, g2], bucket_mb=0.001)  # force multi-bucket
        ok = torch.allclose(g1, torch.full((1000,), 1.5)) and \
            torch.allclose(g2, torch.full((3,), 15.0))
        if rank == 0:
            q.put(("result", bool(ok)))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def _run_spawn(fn, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(fn, args=(port, q), nprocs=WORLD, join=True,
                       start_method="spawn")
    assert not q.empty(), "worker produced no result"
    msg = q.get()
    assert msg[0] == "result", f"worker error: {msg}"
    return msg[1:]


def test_sharded_fm_matches_single_model():
    ok_w, ok_v, dw, dv = _run_spawn(_sharded_fm_worker, 29531)
    assert ok_w, f"W mismatch max={dw}"
    assert ok_v, f"V mismatch max={dv}"


def test_ring_broadcast_allreduce():
    (ok,) = _run_spawn(_ring_worker, 29532)
    assert ok


def _ring_dp_mlp_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        import torch as t
        from lightctr_amd.models.mlp import MLP
        from lightctr_amd.parallel.ring import (RingDataParallel,
                                                mlp_param_grads)

        # identical init (same seed); different local batches
        mlp = MLP([8, 16, 1], optimizer="adagrad", lr=0.05, seed=3,
                  device="cpu")
        rdp = RingDataParallel(mlp_param_grads(mlp))
        rdp.sync_init()
        g = t.Generator().manual_seed(50 + rank)
        X = t.randn(32, 8, generator=g)
        y = t.randn(32, 1, generator=g)
        for _ in range(3):
            out = mlp.forward(X)
            mlp.backward((out - y) / 32)
            rdp.sync_gradients()  # average grads over the ring
            mlp.apply_grads()
        # all ranks must hold identical parameters now
        W0 = mlp.layers[0].W.clone()
        gathered = [t.zeros_like(W0) for _ in range(WORLD)]
        dist.all_gather(gathered, W0)
        same = all(t.allclose(gathered[0], w, atol=1e-7) for w in gathered)
        if rank == 0:
            q.put(("result", bool(same)))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_ring_dp_mlp_sync():
    """Reference ring-CNN mode semantics on the dense engine: broadcast
    init + per-step gradient all-reduce keeps replicas identical."""
    (ok,) = _run_spawn(_ring_dp_mlp_worker, 29533)
    assert ok


def _sharded_fm_w4_worker(rank, port, q):
    try:
        import torch.distributed as dist

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=4)
        from lightctr_amd.models.fm import FMHyper, FMModel
        from lightctr_amd.parallel.sharded_fm import ShardedFMModel
        from conftest import make_random_csr

        F, K = 4000, 8
        h = FMHyper(num_features=F, k=K, optimizer="ftrl", seed=13)
        sharded = ShardedFMModel(h, device="cpu")
        for step in range(3):
            row_ptr, fids, vals, labels = make_random_csr(
                B=48, F_total=F, seed=step * 11 + rank, binary_vals=False)
            loss = sharded.train_step(row_ptr, fids, vals, labels)
            assert torch.isfinite(loss).all()
        # reconstruct + compare against single model over the same batches
        gathered_W = [torch.zeros_like(sharded.W) for _ in range(4)]
        gathered_V = [torch.zeros_like(sharded.V) for _ in range(4)]
        dist.all_gather(gathered_W, sharded.W)
        dist.all_gather(gathered_V, sharded.V)
        if rank == 0:
            W_full = torch.zeros(F)
            V_full = torch.zeros(F, K)
            for r in range(4):
                idx = torch.arange(r, F, 4)
                W_full[idx] = gathered_W[r][: idx.numel()]
                V_full[idx] = gathered_V[r][: idx.numel()]
            single = FMModel(h, device="cpu")
            for r in range(4):
                g = torch.Generator().manual_seed(h.seed + 17 * r)
                Fl = (F + 3) // 4
                Vr = torch.randn(Fl, K, generator=g) * h.init_sigma
                idx = torch.arange(r, F, 4)
                single.V[idx] = Vr[: idx.numel()]
            for step in range(3):
                bs = [make_random_csr(B=48, F_total=F, seed=step * 11 + r,
                                      binary_vals=False) for r in range(4)]
                rp = bs[0][0]
                fids_u, vals_u, labels_u = bs[0][1], bs[0][2], bs[0][3]
                for r in range(1, 4):
                    rp = torch.cat([rp[:-1], bs[r][0] + rp[-1]])
                    fids_u = torch.cat([fids_u, bs[r][1]])
                    vals_u = torch.cat([vals_u, bs[r][2]])
                    labels_u = torch.cat([labels_u, bs[r][3]])
                single.train_step(rp, fids_u, vals_u, labels_u)
            ok = torch.allclose(W_full, single.W, atol=1e-5) and \
                torch.allclose(V_full, single.V, atol=1e-5)
            q.put(("result", bool(ok),
                   float((V_full - single.V).abs().max())))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_sharded_fm_world4_multi_step():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(_sharded_fm_w4_worker, args=(29534, q), nprocs=4,
                       join=True, start_method="spawn")
    assert not q.empty()
    msg = q.get()
    assert msg[0] == "result", f"worker error: {msg}"
    assert msg[1], f"3-step world-4 sharded FTRL diverged: {msg[2]}"


def _ring_cnn_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        import torch as t
        from lightctr_amd.models.cnn import CNNHyper, CNNModel
        from lightctr_amd.parallel.ring import (RingDataParallel,
                                                cnn_param_grads)

        m = CNNModel(CNNHyper(in_shape=(1, 8, 8), n_classes=2, lr=1e-2,
                              seed=9))
        rdp = RingDataParallel(cnn_param_grads(m))
        rdp.sync_init()
        g = t.Generator().manual_seed(70 + rank)
        X = t.rand(16, 1, 8, 8, generator=g)
        y = t.randint(0, 2, (16,), generator=g)
        for _ in range(2):
            # reference ring-CNN semantics: local fwd/bwd, ring all-reduce
            # of the fused grads, then the per-layer optimizer step
            logits = m.forward(X, train=True)
            p = t.softmax(logits, dim=1)
            yk = t.nn.functional.one_hot(y, 2).float()
            dx = m.head.backward((p - yk) / 16)
            for la in reversed(m.layers):
                dx = la.backward(dx)
            rdp.sync_gradients()
            m.head.apply_grads()
            for la in m.layers:
                la.apply_grads()
        W0 = m.layers[0].fc.W.clone()
        gathered = [t.zeros_like(W0) for _ in range(WORLD)]
        dist.all_gather(gathered, W0)
        same = all(t.allclose(gathered[0], w, atol=1e-6) for w in gathered)
        if rank == 0:
            q.put(("result", bool(same)))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_ring_cnn_replicas_stay_synced():
    """Reference WORKER_RING CNN mode (train_cnn_algo.h) on this engine."""
    (ok,) = _run_spawn(_ring_cnn_worker, 29535)
    assert ok


def _sharded_wd_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        import torch as t
        from lightctr_amd.models.wide_deep import (WideDeepHyper,
                                                   WideDeepModel)
        from lightctr_amd.parallel.sharded_widedeep import (
            ShardedWideDeepModel)
        from conftest import make_random_csr

        F, K, nf = 1200, 8, 10
        h = WideDeepHyper(num_features=F, num_fields=nf, k=K,
                          optimizer="adagrad", hidden=(16,), seed=21,
                          mlp_optimizer="adagrad", mlp_lr=0.01)
        sharded = ShardedWideDeepModel(h, device="cpu")
        batches = [make_random_csr(B=32, F_total=F, min_f=nf, max_f=nf,
                                   seed=50 + 3 * s + rank,
                                   binary_vals=False)
                   for s in range(2)]
        for rp, fi, v, lb in batches:
            loss = sharded.train_step(rp, fi, v, lb)
            assert t.isfinite(loss).all()
        gathered_W = [t.zeros_like(sharded.W) for _ in range(WORLD)]
        gathered_E = [t.zeros_like(sharded.E) for _ in range(WORLD)]
        dist.all_gather(gathered_W, sharded.W)
        dist.all_gather(gathered_E, sharded.E)
        if rank == 0:
            W_full = t.zeros(F)
            E_full = t.zeros(F, K)
            for r in range(WORLD):
                idx = t.arange(r, F, WORLD)
                W_full[idx] = gathered_W[r][: idx.numel()]
                E_full[idx] = gathered_E[r][: idx.numel()]
            single = WideDeepModel(h, device="cpu")
            for r in range(WORLD):
                g = t.Generator().manual_seed(h.seed + 17 * r)
                Fl = (F + WORLD - 1) // WORLD
                Er = t.randn(Fl, K, generator=g) * h.init_sigma
                idx = t.arange(r, F, WORLD)
                single.E[idx] = Er[: idx.numel()]
            for s in range(2):
                bs = [make_random_csr(B=32, F_total=F, min_f=nf, max_f=nf,
                                      seed=50 + 3 * s + r,
                                      binary_vals=False)
                      for r in range(WORLD)]
                rp, fi, v, lb = bs[0]
                for r in range(1, WORLD):
                    rp = t.cat([rp[:-1], bs[r][0] + rp[-1]])
                    fi = t.cat([fi, bs[r][1]])
                    v = t.cat([v, bs[r][2]])
                    lb = t.cat([lb, bs[r][3]])
                single.train_step(rp, fi, v, lb)
            ok_e = t.allclose(E_full, single.E, atol=1e-5)
            ok_w = t.allclose(W_full, single.W, atol=1e-5)
            ok_mlp = t.allclose(sharded.mlp.layers[0].W,
                                single.mlp.layers[0].W, atol=1e-5)
            q.put(("result", bool(ok_e and ok_w and ok_mlp),
                   float((E_full - single.E).abs().max()),
                   float((sharded.mlp.layers[0].W
                          - single.mlp.layers[0].W).abs().max())))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_sharded_widedeep_matches_single():
    """Sharded embedding + SUM-allreduced replicated MLP == single model
    on the union batch (2 steps, adagrad both sides)."""
    ok, de, dmlp = _run_spawn(_sharded_wd_worker, 29536)
    assert ok, f"maxdiff E={de} mlp={dmlp}"


def _sharded_ffm_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        import torch as t
        from lightctr_amd.models.ffm import FFMHyper, FFMModel
        from lightctr_amd.parallel.sharded_ffm import ShardedFFMModel
        from conftest import make_random_csr

        F, nf, K = 800, 6, 4
        h = FFMHyper(num_features=F, num_fields=nf, k=K,
                     optimizer="adagrad", seed=33)
        sharded = ShardedFFMModel(h, device="cpu")
        batches = []
        for r in range(WORLD):
            rp, fi, v, lb = make_random_csr(B=24, F_total=F, min_f=3,
                                            max_f=nf, seed=90 + r,
                                            binary_vals=False)
            fl = (fi.long() % nf).int()
            batches.append((rp, fl, fi, v, lb))
        rp, fl, fi, v, lb = batches[rank]
        loss = sharded.train_step(rp, fl, fi, v, lb)
        assert t.isfinite(loss).all()
        gathered_W = [t.zeros_like(sharded.W) for _ in range(WORLD)]
        gathered_V = [t.zeros_like(sharded.V) for _ in range(WORLD)]
        dist.all_gather(gathered_W, sharded.W)
        dist.all_gather(gathered_V, sharded.V)
        if rank == 0:
            W_full = t.zeros(F)
            V_full = t.zeros(F, nf, K)
            for r in range(WORLD):
                idx = t.arange(r, F, WORLD)
                W_full[idx] = gathered_W[r][: idx.numel()]
                V_full[idx] = gathered_V[r][: idx.numel()]
            single = FFMModel(h, device="cpu")
            for r in range(WORLD):
                g = t.Generator().manual_seed(h.seed + 17 * r)
                Fl = (F + WORLD - 1) // WORLD
                Vr = t.randn(Fl, nf, K, generator=g) * h.init_sigma
                idx = t.arange(r, F, WORLD)
                single.V[idx] = Vr[: idx.numel()]
            rp, fl, fi, v, lb = batches[0]
            for r in range(1, WORLD):
                rp2, fl2, fi2, v2, lb2 = batches[r]
                rp = t.cat([rp[:-1], rp2 + rp[-1]])
                fl = t.cat([fl, fl2])
                fi = t.cat([fi, fi2])
                v = t.cat([v, v2])
                lb = t.cat([lb, lb2])
            single.train_step(rp, fl, fi, v, lb)
            ok = t.allclose(W_full, single.W, atol=1e-5) and \
                t.allclose(V_full, single.V, atol=1e-5)
            q.put(("result", bool(ok),
                   float((V_full - single.V).abs().max())))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_sharded_ffm_matches_single():
    ok, dv = _run_spawn(_sharded_ffm_worker, 29537)
    assert ok, f"V maxdiff {dv}"


def _sharded_fm_fp16_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        from lightctr_amd.models.fm import FMHyper
        from lightctr_amd.parallel.sharded_fm import ShardedFMModel
        from conftest import make_random_csr

        h = FMHyper(num_features=1000, k=8, optimizer="adagrad", seed=99)
        m = ShardedFMModel(h, device="cpu", wire="fp16")
        for step in range(3):
            rp, fi, v, lb = make_random_csr(B=64, F_total=1000,
                                            seed=step * 5 + rank,
                                            binary_vals=False)
            loss = m.train_step(rp, fi, v, lb)
            assert torch.isfinite(loss).all()
        if rank == 0:
            q.put(("result", bool(torch.isfinite(m.V).all()
                                  and m.V.abs().sum() > 0)))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_sharded_fm_fp16_wire_trains():
    (ok,) = _run_spawn(_sharded_fm_fp16_worker, 29538)
    assert ok


def _sharded_fm_pipelined_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        from lightctr_amd.models.fm import FMHyper
        from lightctr_amd.parallel.sharded_fm import ShardedFMModel
        from conftest import make_random_csr

        h = FMHyper(num_features=1000, k=8, optimizer="adagrad", seed=99)
        sync = ShardedFMModel(h, device="cpu")
        pipe = ShardedFMModel(h, device="cpu")       # same seed -> same init
        over = ShardedFMModel(h, device="cpu")       # SSP-1 overlap flavor

        batches = [make_random_csr(B=64, F_total=1000, seed=step * 7 + rank,
                                   binary_vals=False) for step in range(3)]

        # pipelined WITHOUT overlap (pull issued at step start) must equal
        # the synchronous step bit-for-bit across multiple steps
        sync_losses, pipe_losses = [], []
        for rp, fi, v, lb in batches:
            sync_losses.append(float(sync.train_step(rp, fi, v, lb).sum()))
        for rp, fi, v, lb in batches:
            pipe_losses.append(
                float(pipe.train_step_pipelined(rp, fi, v, lb).sum()))
        ok_eq = (torch.allclose(sync.W, pipe.W, atol=1e-6)
                 and torch.allclose(sync.V, pipe.V, atol=1e-6))

        # SSP-1 overlap: prefetch batch t+1 before batch t's apply. Step 1
        # trains on identical params -> identical loss; later steps use
        # params stale by one update but must stay finite and train.
        over_losses = []
        for step, (rp, fi, v, lb) in enumerate(batches):
            nxt = batches[step + 1][1] if step + 1 < len(batches) else None
            over_losses.append(
                float(over.train_step_pipelined(rp, fi, v, lb,
                                                next_fids=nxt).sum()))
        ok_first = abs(over_losses[0] - sync_losses[0]) < 1e-6
        ok_finite = all(torch.isfinite(torch.tensor(over_losses)))

        if rank == 0:
            q.put(("result", bool(ok_eq), bool(ok_first), bool(ok_finite),
                   sync_losses, pipe_losses, over_losses))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_sharded_fm_pipelined():
    ok_eq, ok_first, ok_finite, sl, pl, ol = _run_spawn(
        _sharded_fm_pipelined_worker, 29539)
    assert ok_eq, f"pipelined(no overlap) != sync: {sl} vs {pl}"
    assert ok_first, f"SSP-1 first step differs: {sl[0]} vs {ol[0]}"
    assert ok_finite, f"SSP-1 losses not finite: {ol}"


def _sharded_fm_w3_worker(rank, port, q):
    try:
        import torch.distributed as dist

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=3)
        from lightctr_amd.models.fm import FMHyper
        from lightctr_amd.parallel.sharded_fm import ShardedFMModel
        from conftest import make_random_csr

        # odd world + F not divisible by world: exercises shard rounding
        h = FMHyper(num_features=1001, k=8, optimizer="adagrad", seed=7)
        m = ShardedFMModel(h, device="cpu")
        for step in range(2):
            rp, fi, v, lb = make_random_csr(B=48, F_total=1001,
                                            seed=step * 3 + rank,
                                            binary_vals=False)
            loss = m.train_step(rp, fi, v, lb)
            assert torch.isfinite(loss).all()
        # sharded checkpoint round-trip: save, perturb, load, compare
        import tempfile

        d = tempfile.mkdtemp()
        prefix = os.path.join(d, "ck")
        m.save(prefix)
        W0, V0 = m.W.clone(), m.V.clone()
        m.W.add_(1.0)
        m.V.mul_(0.5)
        m.load(prefix)
        ok = (torch.equal(m.W, W0) and torch.equal(m.V, V0))
        # one more step after resume must still work
        rp, fi, v, lb = make_random_csr(B=48, F_total=1001, seed=99 + rank,
                                        binary_vals=False)
        loss = m.train_step(rp, fi, v, lb)
        ok = ok and bool(torch.isfinite(loss).all())
        if rank == 0:
            q.put(("result", ok))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_sharded_fm_world3_and_checkpoint():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(_sharded_fm_w3_worker, args=(29541, q), nprocs=3,
                       join=True, start_method="spawn")
    assert not q.empty()
    msg = q.get()
    assert msg[0] == "result", f"worker error: {msg}"
    assert msg[1], "world-3 sharded train/checkpoint failed"


def _sharded_nfm_worker(rank, port, q):
    try:
        dist = _init(rank, port)
        import torch as t
        from lightctr_amd.models.nfm import NFMHyper, NFMModel
        from lightctr_amd.parallel.sharded_nfm import ShardedNFMModel
        from conftest import make_random_csr

        F, K = 1200, 8
        h = NFMHyper(num_features=F, k=K, optimizer="adagrad", hidden=(16,),
                     seed=23, mlp_optimizer="adagrad", mlp_lr=0.01,
                     dropout=0.0)
        sharded = ShardedNFMModel(h, device="cpu")
        batches = [make_random_csr(B=32, F_total=F, seed=60 + 3 * s + rank,
                                   binary_vals=False) for s in range(2)]
        for rp, fi, v, lb in batches:
            loss = sharded.train_step(rp, fi, v, lb)
            assert t.isfinite(loss).all()
        gathered_W = [t.zeros_like(sharded.W) for _ in range(WORLD)]
        gathered_V = [t.zeros_like(sharded.V) for _ in range(WORLD)]
        dist.all_gather(gathered_W, sharded.W)
        dist.all_gather(gathered_V, sharded.V)
        if rank == 0:
            W_full = t.zeros(F)
            V_full = t.zeros(F, K)
            for r in range(WORLD):
                idx = t.arange(r, F, WORLD)
                W_full[idx] = gathered_W[r][: idx.numel()]
                V_full[idx] = gathered_V[r][: idx.numel()]
            single = NFMModel(h, device="cpu")
            for r in range(WORLD):
                g = t.Generator().manual_seed(h.seed + 17 * r)
                Fl = (F + WORLD - 1) // WORLD
                Vr = t.randn(Fl, K, generator=g) * h.init_sigma
                idx = t.arange(r, F, WORLD)
                single.V[idx] = Vr[: idx.numel()]
                single.W[idx] = 0.0
            for s in range(2):
                bs = [make_random_csr(B=32, F_total=F, seed=60 + 3 * s + r,
                                      binary_vals=False)
                      for r in range(WORLD)]
                rp, fi, v, lb = bs[0]
                for r in range(1, WORLD):
                    rp = t.cat([rp[:-1], bs[r][0] + rp[-1]])
                    fi = t.cat([fi, bs[r][1]])
                    v = t.cat([v, bs[r][2]])
                    lb = t.cat([lb, bs[r][3]])
                single.train_step(rp, fi, v, lb)
            ok_v = t.allclose(V_full, single.V, atol=1e-5)
            ok_w = t.allclose(W_full, single.W, atol=1e-5)
            ok_mlp = t.allclose(sharded.mlp.layers[0].W,
                                single.mlp.layers[0].W, atol=1e-5)
            q.put(("result", bool(ok_v and ok_w and ok_mlp),
                   float((V_full - single.V).abs().max()),
                   float((sharded.mlp.layers[0].W
                          - single.mlp.layers[0].W).abs().max())))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


def test_sharded_nfm_matches_single():
    """Sharded NFM (sharded W/V + SUM-allreduced bi-interaction MLP) ==
    single model on the union batch."""
    ok, dv, dmlp = _run_spawn(_sharded_nfm_worker, 29542)
    assert ok, f"maxdiff V={dv} mlp={dmlp}"
