"""Async PS mode tests (gloo, CPU): 1 PS + 2 workers training FM with
SSP/DCASGD semantics; codecs round-trip."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _ps_worker(rank, port, q, updater, wire):
    try:
        import torch.distributed as dist

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        world = 3
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from lightctr_amd.parallel.ps import (PSConfig, PSShard,
                                              ps_train_fm, setup_pair_groups)
        from conftest import make_random_csr

        cfg = PSConfig(num_features=2000, k=8, ps_shards=1, updater=updater,
                       lr=0.1, wire=wire, staleness=10)
        groups = setup_pair_groups(cfg)
        if rank == 0:
            shard = PSShard(cfg, device="cpu")
            shard.serve(groups)
            ok = bool(torch.isfinite(shard.W).all()
                      and torch.isfinite(shard.V).all()
                      and shard.W.abs().sum() > 0)
            q.put(("result", ok))
        else:
            def gen(step):
                return make_random_csr(B=64, F_total=2000,
                                       seed=step * 7 + rank,
                                       binary_vals=False)

            losses = ps_train_fm(cfg, groups[rank], gen, steps=12,
                                 batch_size=64, device="cpu")
            assert all(l == l for l in losses)  # finite
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("error", rank, traceback.format_exc()))
        raise


@pytest.mark.parametrize("updater,wire", [("dcasgd", "fp32"),
                                          ("adagrad", "fp16"),
                                          ("dcasgda", "int8"),
                                          ("sgd", "fp32")])
def test_ps_mode_trains(updater, wire):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29550 + abs(hash((updater, wire))) % 100
    mp.start_processes(_ps_worker, args=(port, q, updater, wire), nprocs=3,
                       join=True, start_method="spawn")
    assert not q.empty()
    msg = q.get()
    assert msg[0] == "result", f"worker error: {msg}"
    assert msg[1]


def test_quantile_codec_roundtrip_cpu():
    from lightctr_amd.utils.compress import QuantileCodec

    for mode in ("uniform", "log", "normal"):
        c = QuantileCodec(levels=256, mode=mode)
        g = torch.Generator().manual_seed(3)
        x = torch.randn(10000, generator=g).clamp(-1, 1) * 0.5
        code = c.encode(x)
        assert code.dtype == torch.uint8
        y = c.decode(code)
        # decoded value must be the nearest table entry
        d_direct = (x - y).abs()
        dists = (x.unsqueeze(1) - c.table.unsqueeze(0)).abs().min(dim=1).values
        assert torch.allclose(d_direct, dists, atol=1e-6), mode


def test_lowbit_codec_roundtrip_cpu():
    from lightctr_amd.utils.compress import LowBitCodec

    g = torch.Generator().manual_seed(5)
    x = torch.randn(1000, generator=g)
    for bits in (1, 2):
        c = LowBitCodec(bits)
        words, meta, n = c.encode(x)
        y = c.decode(words, meta, n)
        assert y.shape == x.shape
        assert torch.equal(torch.sign(y), torch.sign(x).where(
            torch.sign(x) != 0, torch.ones_like(x) * -1))
        # 2-bit carries magnitude info: correlation should be decent
        if bits == 2:
            corr = torch.corrcoef(torch.stack([x, y]))[0, 1]
            assert corr > 0.7


def test_pq_roundtrip_cpu():
    from lightctr_amd.utils.compress import ProductQuantizer

    g = torch.Generator().manual_seed(7)
    X = torch.randn(500, 16, generator=g)
    pq = ProductQuantizer(dim=16, n_sub=4, n_centroids=32, iters=8)
    pq.fit(X)
    codes = pq.encode(X)
    assert codes.shape == (500, 4) and codes.dtype == torch.uint8
    Y = pq.decode(codes)
    mse = ((X - Y) ** 2).mean()
    base = (X ** 2).mean()
    assert mse < base * 0.7  # quantization reduces energy meaningfully


def test_fp16_codec():
    from lightctr_amd.utils.compress import Fp16Codec

    x = torch.randn(100)
    y = Fp16Codec.decode(Fp16Codec.encode(x))
    assert (x - y).abs().max() < 1e-2


@pytest.mark.gpu
def test_quantile_codec_gpu_parity():
    from lightctr_amd.utils.compress import QuantileCodec

    c_cpu = QuantileCodec(levels=256, mode="log")
    c_gpu = QuantileCodec(levels=256, mode="log", device="cuda:0")
    g = torch.Generator().manual_seed(11)
    x = (torch.randn(100000, generator=g) * 0.3).clamp(-1, 1)
    code_cpu = c_cpu.encode(x)
    code_gpu = c_gpu.encode(x.cuda())
    assert torch.equal(code_gpu.cpu(), code_cpu)
    assert torch.allclose(c_gpu.decode(code_gpu).cpu(),
                          c_cpu.decode(code_cpu), atol=1e-6)


@pytest.mark.gpu
def test_lowbit_codec_gpu_parity():
    from lightctr_amd.utils.compress import LowBitCodec

    g = torch.Generator().manual_seed(13)
    x = torch.randn(4096, generator=g)
    for bits in (1, 2):
        c = LowBitCodec(bits)
        w_cpu, meta_cpu, n = c.encode(x)
        w_gpu, meta_gpu, _ = c.encode(x.cuda())
        assert torch.equal(w_gpu.cpu(), w_cpu)
        y_cpu = c.decode(w_cpu, meta_cpu, n)
        y_gpu = c.decode(w_gpu, meta_gpu, n)
        assert torch.allclose(y_gpu.cpu(), y_cpu, atol=1e-6)


@pytest.mark.gpu
@pytest.mark.parametrize("updater", ["sgd", "adagrad", "dcasgd", "dcasgda"])
def test_ps_apply_kernel_parity(updater):
    """GPU fused PS updater kernel vs the torch reference math."""
    from lightctr_amd.ops import hip_ops

    F, K, n = 500, 8, 120
    g = torch.Generator().manual_seed(17)
    lidx = torch.randint(0, F, (n,), generator=g).unique().long().cuda()
    n = lidx.numel()
    gW = torch.randn(n, generator=g).cuda() * 0.01
    gV = torch.randn(n, K, generator=g).cuda() * 0.01
    W = torch.randn(F, generator=g).cuda()
    V = (torch.randn(F, K, generator=g) * 0.1).cuda()
    nW = torch.rand(F, generator=g).cuda() * 0.1
    nV = torch.rand(F, K, generator=g).cuda() * 0.1
    shW = torch.randn(F, generator=g).cuda() * 0.1
    shV = torch.randn(F, K, generator=g).cuda() * 0.1
    lr, lam, eps = 0.05, 0.1, 1e-8
    # torch reference (mirrors parallel/ps.PSShard._apply CPU math)
    rW, rV = W.clone(), V.clone()
    rnW, rnV, rshW, rshV = nW.clone(), nV.clone(), shW.clone(), shV.clone()
    if updater == "sgd":
        rW.index_add_(0, lidx, -lr * gW)
        rV.index_add_(0, lidx, -lr * gV)
    elif updater == "adagrad":
        rnW[lidx] += gW * gW
        rnV[lidx] += gV * gV
        rW[lidx] -= lr * gW / (rnW[lidx] + eps).sqrt()
        rV[lidx] -= lr * gV / (rnV[lidx] + eps).sqrt()
    else:
        lamW = lamV = lam
        if updater == "dcasgda":
            rnW[lidx] = 0.95 * rnW[lidx] + 0.05 * gW * gW
            rnV[lidx] = 0.95 * rnV[lidx] + 0.05 * gV * gV
            lamW = lam / (rnW[lidx] + eps).sqrt()
            lamV = lam / (rnV[lidx] + eps).sqrt()
        newW = rW[lidx] - lr * (gW + lamW * gW * gW
                                * (rW[lidx] - rshW[lidx]))
        newV = rV[lidx] - lr * (gV + lamV * gV * gV
                                * (rV[lidx] - rshV[lidx]))
        rW[lidx] = newW
        rV[lidx] = newV
        rshW[lidx] = newW
        rshV[lidx] = newV
    upd = {"sgd": 0, "adagrad": 1, "dcasgd": 2, "dcasgda": 3}[updater]
    hip_ops.ps_apply(lidx, gW, gV, W, V,
                     nW if updater in ("adagrad", "dcasgda") else None,
                     nV if updater in ("adagrad", "dcasgda") else None,
                     shW if updater.startswith("dcasgd") else None,
                     shV if updater.startswith("dcasgd") else None,
                     upd, lr, lam, eps)
    assert torch.allclose(W, rW, atol=1e-5, rtol=1e-5), updater
    assert torch.allclose(V, rV, atol=1e-5, rtol=1e-5), updater
    if updater in ("adagrad", "dcasgda"):
        assert torch.allclose(nW, rnW, atol=1e-6)
        assert torch.allclose(nV, rnV, atol=1e-6)
    if updater.startswith("dcasgd"):
        assert torch.allclose(shV, rshV, atol=1e-5)
