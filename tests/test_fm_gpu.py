"""GPU numerics tests: gfx950 HIP kernels vs the fp32 torch reference ops."""

import pytest
import torch

from lightctr_amd.models.fm import FMHyper, FMModel
from lightctr_amd.ops import fm_ref

from conftest import make_random_csr

pytestmark = pytest.mark.gpu


def _csr(B=256, F_total=50_000, seed=0, device="cuda:0"):
    return make_random_csr(B=B, F_total=F_total, seed=seed, device=device,
                           binary_vals=False)


def test_hip_ops_loaded():
    """On a GPU box the native extension must be importable — no fallback."""
    from lightctr_amd.ops._extension import require_hip_ops

    ops = require_hip_ops()
    assert hasattr(ops, "fm_forward")


@pytest.mark.parametrize("K", [4, 8, 16, 32])
def test_fm_forward_parity(K):
    from lightctr_amd.ops import hip_ops

    row_ptr, fids, vals, _ = _csr(seed=K)
    F = 50_000
    g = torch.Generator().manual_seed(K)
    W = torch.randn(F, generator=g).cuda()
    V = (torch.randn(F, K, generator=g) * 0.1).cuda()
    pred, sumVX = hip_ops.fm_forward(row_ptr, fids, vals, W, V)
    pred_ref, sumVX_ref = fm_ref.fm_forward_ref(row_ptr, fids, vals, W, V)
    assert torch.allclose(sumVX, sumVX_ref, atol=1e-4, rtol=1e-4)
    assert torch.allclose(pred, pred_ref, atol=1e-3, rtol=1e-4)


def test_logloss_grad_parity():
    from lightctr_amd.ops import hip_ops

    g = torch.Generator().manual_seed(9)
    pred = (torch.randn(1000, generator=g) * 4).cuda()
    label = torch.randint(0, 2, (1000,), generator=g).float().cuda()
    loss, dpred = hip_ops.logloss_grad(pred, label, 0.01)
    loss_ref, dpred_ref = fm_ref.logloss_grad_ref(pred, label, 0.01)
    assert torch.allclose(loss, loss_ref, atol=1e-5, rtol=1e-5)
    assert torch.allclose(dpred, dpred_ref, atol=1e-6, rtol=1e-5)


@pytest.mark.parametrize("K", [8, 16])
def test_fm_backward_parity(K):
    from lightctr_amd.ops import hip_ops

    row_ptr, fids, vals, labels = _csr(seed=100 + K)
    F = 50_000
    g = torch.Generator().manual_seed(2)
    V = (torch.randn(F, K, generator=g) * 0.1).cuda()
    W = torch.randn(F, generator=g).cuda()
    pred, sumVX = hip_ops.fm_forward(row_ptr, fids, vals, W, V)
    _, dpred = hip_ops.logloss_grad(pred, labels, 1.0 / 256)
    gradW = torch.zeros(F).cuda()
    gradV = torch.zeros(F, K).cuda()
    touched = torch.zeros((F + 63) // 64, dtype=torch.int64).cuda()
    hip_ops.fm_backward(row_ptr, fids, vals, V, sumVX, dpred, gradW, gradV,
                        touched)
    gW_ref, gV_ref = fm_ref.fm_backward_ref(row_ptr, fids, vals, V, sumVX,
                                            dpred)
    assert torch.allclose(gradW, gW_ref, atol=1e-5, rtol=1e-4)
    assert torch.allclose(gradV, gV_ref, atol=1e-5, rtol=1e-4)
    # touched bitmap marks exactly the batch's unique fids
    uniq_ref = torch.unique(fids.long())
    words = touched.cpu().numpy().astype("uint64")
    marked = [w * 64 + b for w in range(len(words)) for b in range(64)
              if (int(words[w]) >> b) & 1]
    assert torch.equal(torch.tensor(marked, dtype=torch.long),
                       uniq_ref.cpu())


def test_bitmap_compact():
    from lightctr_amd.ops import hip_ops

    F = 4096
    bitmap = torch.zeros((F + 63) // 64, dtype=torch.int64).cuda()
    fids = torch.tensor([3, 64, 65, 700, 4095], dtype=torch.long)
    for f in fids:
        w, b = int(f) // 64, int(f) % 64
        bitmap[w] |= 1 << b
    uniq = torch.zeros(F, dtype=torch.int32).cuda()
    count = torch.zeros(1, dtype=torch.int32).cuda()
    hip_ops.bitmap_compact(bitmap, uniq, count)
    n = int(count.item())
    assert n == len(fids)
    got = torch.sort(uniq[:n].long().cpu()).values
    assert torch.equal(got, fids)
    assert int(bitmap.abs().sum()) == 0  # cleared


@pytest.mark.parametrize("opt", ["adagrad", "ftrl"])
def test_fm_optimizer_apply_parity(opt):
    from lightctr_amd.ops import hip_ops

    F, K = 5000, 16
    g = torch.Generator().manual_seed(4)
    uniq = torch.unique(torch.randint(0, F, (800,), generator=g)).int().cuda()
    count = torch.tensor([uniq.numel()], dtype=torch.int32).cuda()
    W = torch.randn(F, generator=g).cuda()
    V = (torch.randn(F, K, generator=g) * 0.1).cuda()
    gradW = torch.randn(F, generator=g).cuda() * 0.01
    gradV = torch.randn(F, K, generator=g).cuda() * 0.01
    nW = torch.rand(F, generator=g).cuda() * 0.1
    nV = torch.rand(F, K, generator=g).cuda() * 0.1
    ref = {t: x.clone() for t, x in
           [("W", W), ("V", V), ("gW", gradW), ("gV", gradV), ("nW", nW),
            ("nV", nV)]}
    if opt == "adagrad":
        hip_ops.fm_adagrad_apply(uniq, count, W, V, nW, nV, gradW, gradV,
                                 0.05, 1e-8, 1e-4)
        fm_ref.adagrad_apply_ref(uniq, ref["W"], ref["V"], ref["nW"],
                                 ref["nV"], ref["gW"], ref["gV"], 0.05, 1e-8,
                                 1e-4)
    else:
        zW = torch.zeros(F).cuda()
        zV = torch.zeros(F, K).cuda()
        ref["zW"], ref["zV"] = zW.clone(), zV.clone()
        hip_ops.fm_ftrl_apply(uniq, count, W, V, zW, nW, zV, nV, gradW,
                              gradV, 0.05, 1.0, 1e-4, 1e-4)
        fm_ref.ftrl_apply_ref(uniq, ref["W"], ref["V"], ref["zW"], ref["nW"],
                              ref["zV"], ref["nV"], ref["gW"], ref["gV"],
                              0.05, 1.0, 1e-4, 1e-4)
        assert torch.allclose(zW, ref["zW"], atol=1e-5)
        assert torch.allclose(zV, ref["zV"], atol=1e-5)
    assert torch.allclose(W, ref["W"], atol=1e-5, rtol=1e-5)
    assert torch.allclose(V, ref["V"], atol=1e-5, rtol=1e-5)
    assert torch.allclose(nW, ref["nW"], atol=1e-6)
    assert int(gradW[uniq.long()].abs().sum()) == 0  # zeroed
    assert int(gradV[uniq.long()].abs().sum()) == 0


def test_fm_gpu_train_convergence():
    """Full fused GPU step on synthetic Criteo slice: loss decreases."""
    from lightctr_amd.data.synthetic import SyntheticCriteo

    gen = SyntheticCriteo(num_features=1 << 16, seed=21, device="cuda:0")
    hyper = FMHyper(num_features=1 << 16, k=16, optimizer="ftrl")
    model = FMModel(hyper, device="cuda:0")
    losses = []
    for step in range(30):
        row_ptr, fields, fids, vals, labels = gen.batch(4096)
        loss = model.train_step(row_ptr, fids, vals, labels)
        losses.append(float(loss.mean()))
    assert losses[-1] < losses[0] * 0.98, losses[:3] + losses[-3:]


def test_fm_gpu_vs_cpu_one_step():
    """One identical train step on GPU vs CPU reference path: params match."""
    F, K = 20_000, 16
    row_ptr, fids, vals, labels = make_random_csr(B=128, F_total=F, seed=42,
                                                  binary_vals=False)
    h = FMHyper(num_features=F, k=K, optimizer="adagrad", seed=7)
    cpu = FMModel(h, device="cpu")
    gpu = FMModel(h, device="cuda:0")
    gpu.W.copy_(cpu.W)
    gpu.V.copy_(cpu.V)
    cpu.train_step(row_ptr, fids, vals, labels)
    gpu.train_step(row_ptr.cuda(), fids.cuda(), vals.cuda(), labels.cuda())
    assert torch.allclose(gpu.W.cpu(), cpu.W, atol=1e-4, rtol=1e-4)
    assert torch.allclose(gpu.V.cpu(), cpu.V, atol=1e-4, rtol=1e-4)


def test_fused_apply_matches_two_phase():
    """fused optimizer-in-apply == slab+compact+apply two-phase, both
    optimizers, on skewed data (long runs exercise interior fast path)."""
    from lightctr_amd.data.synthetic import SyntheticCriteo

    # ftrl_v covers pure FTRL (mode 2) and the default FTRL-W/Adagrad-V
    # pairing (round-2 fused mode 3)
    for opt, ftrl_v in (("adagrad", "ftrl"), ("ftrl", "ftrl"),
                        ("ftrl", "adagrad")):
        h = FMHyper(num_features=1 << 15, k=16, optimizer=opt, seed=31,
                    ftrl_v=ftrl_v)
        a = FMModel(h, device="cuda:0")
        a.fused_apply = True
        b = FMModel(h, device="cuda:0")
        b.W.copy_(a.W)
        b.V.copy_(a.V)
        b.fused_apply = False
        gen = SyntheticCriteo(num_features=1 << 15, seed=77, device="cuda:0")
        for _ in range(3):
            row_ptr, fields, fids, vals, labels = gen.batch(4096)
            a.train_step(row_ptr, fids, vals, labels)
            b.train_step(row_ptr, fids, vals, labels)
        assert torch.allclose(a.W, b.W, atol=1e-5), \
            (a.W - b.W).abs().max()
        assert torch.allclose(a.V, b.V, atol=1e-5), \
            (a.V - b.V).abs().max()
        assert torch.allclose(a.nV, b.nV, atol=1e-5)


def test_sorted_apply_single_hot_feature():
    """Extreme skew: every entry is the SAME fid (one segment spanning all
    chunks/subgroups) — partial flushes must sum exactly."""
    from lightctr_amd.ops import hip_ops

    B, F, K = 1024, 100, 16
    row_ptr = torch.arange(0, (B + 1) * 4, 4, dtype=torch.int32).cuda()
    fids = torch.full((B * 4,), 7, dtype=torch.int32).cuda()
    g = torch.Generator().manual_seed(3)
    vals = torch.rand(B * 4, generator=g).cuda()
    labels = torch.randint(0, 2, (B,), generator=g).float().cuda()
    W = torch.randn(F, generator=g).cuda()
    V = (torch.randn(F, K, generator=g) * 0.1).cuda()
    pred, sumVX = hip_ops.fm_forward(row_ptr, fids, vals, W, V)
    _, dpred = hip_ops.logloss_grad(pred, labels, 1.0 / B)
    gw, gv = hip_ops.fm_backward_emit(row_ptr, fids, vals, V, sumVX, dpred)
    sorted_fids, perm = torch.sort(fids)
    gradW = torch.zeros(F).cuda()
    gradV = torch.zeros(F, K).cuda()
    touched = torch.zeros((F + 63) // 64, dtype=torch.int64).cuda()
    hip_ops.fm_sorted_apply(sorted_fids, perm, gw, gv, gradW, gradV, touched)
    gW_ref, gV_ref = fm_ref.fm_backward_ref(row_ptr, fids, vals, V, sumVX,
                                            dpred)
    assert torch.allclose(gradW, gW_ref, atol=1e-3, rtol=1e-4)
    assert torch.allclose(gradV, gV_ref, atol=1e-3, rtol=1e-4)
    assert int(touched.cpu()[0]) == 1 << 7  # only fid 7 marked


def test_fm_forward_k64():
    from lightctr_amd.ops import hip_ops

    row_ptr, fids, vals, _ = _csr(B=64, seed=64)
    F, K = 50_000, 64
    g = torch.Generator().manual_seed(64)
    W = torch.randn(F, generator=g).cuda()
    V = (torch.randn(F, K, generator=g) * 0.1).cuda()
    pred, sumVX = hip_ops.fm_forward(row_ptr, fids, vals, W, V)
    pred_ref, sumVX_ref = fm_ref.fm_forward_ref(row_ptr, fids, vals, W, V)
    assert torch.allclose(sumVX, sumVX_ref, atol=1e-4, rtol=1e-4)
    assert torch.allclose(pred, pred_ref, atol=2e-3, rtol=1e-4)


@pytest.mark.parametrize("seed", [101, 202, 303, 404])
def test_fm_step_fuzz_parity(seed):
    """Randomized CSR shapes/values: one full GPU step == CPU oracle step."""
    g = torch.Generator().manual_seed(seed)
    F = int(torch.randint(500, 60_000, (1,), generator=g))
    K = [4, 8, 16, 32][seed % 4]
    B = int(torch.randint(16, 700, (1,), generator=g))
    min_f = int(torch.randint(1, 5, (1,), generator=g))
    max_f = min_f + int(torch.randint(1, 60, (1,), generator=g))
    row_ptr, fids, vals, labels = make_random_csr(
        B=B, F_total=F, min_f=min_f, max_f=max_f, seed=seed,
        binary_vals=bool(seed % 2))
    h = FMHyper(num_features=F, k=K,
                optimizer="ftrl" if seed % 2 else "adagrad", seed=seed)
    cpu = FMModel(h, device="cpu")
    gpu = FMModel(h, device="cuda:0")
    gpu.W.copy_(cpu.W)
    gpu.V.copy_(cpu.V)
    cpu.train_step(row_ptr, fids, vals, labels)
    gpu.train_step(row_ptr.cuda(), fids.cuda(), vals.cuda(), labels.cuda())
    assert torch.allclose(gpu.W.cpu(), cpu.W, atol=2e-4, rtol=1e-3), \
        (F, K, B, (gpu.W.cpu() - cpu.W).abs().max())
    assert torch.allclose(gpu.V.cpu(), cpu.V, atol=2e-4, rtol=1e-3)


def test_radix_sort_index_parity():
    """Bit-range rocPRIM sort == torch.sort on the id distributions the
    backward actually sees (power-law fids, localized ids, tiny arrays)."""
    from lightctr_amd.ops._extension import sort_ids

    torch.manual_seed(7)
    for upper, n in [(1 << 24, 2_500_000), (1000, 4096), (7, 64), (1, 1)]:
        fids = torch.randint(0, upper, (n,), dtype=torch.int32,
                             device="cuda")
        s, p = sort_ids(fids, upper)
        ref_s, ref_p = torch.sort(fids)
        assert torch.equal(s, ref_s)
        assert p.dtype == torch.int32  # half the radix payload of int64
        # perm must be a valid permutation mapping fids -> sorted order
        assert torch.equal(fids[p.long()], s)
        assert torch.equal(torch.sort(p.long()).values,
                           torch.arange(n, device="cuda"))


def test_scatter_emit_matches_gather_apply():
    """Scatter-emit (pos=inverse perm, apply with perm=None) produces the
    same slab gradients as the default emit+gather pipeline."""
    from lightctr_amd.ops._extension import require_hip_ops, sort_ids

    ops = require_hip_ops()
    torch.manual_seed(3)
    F, K, B = 5000, 16, 512
    row_ptr, fids, vals, labels = make_random_csr(B=B, F_total=F,
                                                  binary_vals=False, seed=3)
    row_ptr, fids = row_ptr.cuda(), fids.cuda()
    vals, labels = vals.cuda(), labels.cuda()
    V = torch.randn(F, K, device="cuda") * 0.05
    W = torch.randn(F, device="cuda") * 0.05
    pred, sumVX = ops.fm_forward(row_ptr, fids, vals, W, V)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / B)
    sorted_fids, perm = sort_ids(fids, F)
    nnz = fids.numel()

    def run(scatter):
        gradW = torch.zeros(F, device="cuda")
        gradV = torch.zeros(F, K, device="cuda")
        touched = torch.zeros((F + 63) // 64, dtype=torch.int64,
                              device="cuda")
        if scatter:
            pos = ops.inv_perm_i32(perm)
            gw, gv = ops.fm_backward_emit(row_ptr, fids, vals, V, sumVX,
                                          dpred, pos)
            ops.fm_sorted_apply(sorted_fids, None, gw, gv, gradW, gradV,
                                touched)
        else:
            gw, gv = ops.fm_backward_emit(row_ptr, fids, vals, V, sumVX,
                                          dpred)
            ops.fm_sorted_apply(sorted_fids, perm, gw, gv, gradW, gradV,
                                touched)
        return gradW, gradV, touched

    gW1, gV1, t1 = run(False)
    gW2, gV2, t2 = run(True)
    assert torch.allclose(gW1, gW2, atol=1e-5)
    assert torch.allclose(gV1, gV2, atol=1e-5)
    assert torch.equal(t1, t2)


def test_segscan_apply_matches_walk():
    """chunk=-1 selects the segmented-scan apply; slabs must match the
    default walk kernel bit-for-bit up to fp32 reduction order."""
    from lightctr_amd.ops._extension import require_hip_ops, sort_ids

    ops = require_hip_ops()
    torch.manual_seed(11)
    F, K, B = 3000, 16, 777
    row_ptr, fids, vals, labels = make_random_csr(B=B, F_total=F,
                                                  binary_vals=False, seed=11)
    row_ptr, fids = row_ptr.cuda(), fids.cuda()
    vals, labels = vals.cuda(), labels.cuda()
    V = torch.randn(F, K, device="cuda") * 0.05
    W = torch.randn(F, device="cuda") * 0.05
    pred, sumVX = ops.fm_forward(row_ptr, fids, vals, W, V)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / B)
    gw, gv = ops.fm_backward_emit(row_ptr, fids, vals, V, sumVX, dpred)
    sorted_fids, perm = sort_ids(fids, F)

    out = []
    # 0 = walk, -1 = lane-per-entry segscan, -2 = quad-per-entry segscan
    for chunk in (0, -1, -2):
        gradW = torch.zeros(F, device="cuda")
        gradV = torch.zeros(F, K, device="cuda")
        touched = torch.zeros((F + 63) // 64, dtype=torch.int64,
                              device="cuda")
        ops.fm_sorted_apply(sorted_fids, perm, gw, gv, gradW, gradV,
                            touched, chunk)
        out.append((gradW, gradV, touched))
    for i in (1, 2):
        assert torch.allclose(out[0][0], out[i][0], atol=1e-5), i
        assert torch.allclose(out[0][1], out[i][1], atol=1e-5), i
        assert torch.equal(out[0][2], out[i][2]), i
