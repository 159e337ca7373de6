"""Dense engine tests: MFMA GEMM parity vs torch, MLP training, NFM / W&D."""

import pytest
import torch

from lightctr_amd.data import LibffmDataset
from lightctr_amd.data.synthetic import SyntheticCriteo
from lightctr_amd.models.mlp import MLP, DenseLayer
from lightctr_amd.models.nfm import NFMHyper, NFMModel, NFMTrainer
from lightctr_amd.models.wide_deep import (WideDeepHyper, WideDeepModel,
                                           WideDeepTrainer)


def test_mlp_cpu_learns_xor_like():
    g = torch.Generator().manual_seed(0)
    X = torch.randn(512, 8, generator=g)
    y = ((X[:, 0] * X[:, 1]) > 0).float().unsqueeze(1)
    mlp = MLP([8, 32, 1], optimizer="adam", lr=1e-2, seed=1, device="cpu")
    for _ in range(300):
        out = mlp.forward(X)
        p = torch.sigmoid(out)
        dloss = (p - y) / 512
        mlp.backward(dloss)
        mlp.apply_grads()
    acc = ((torch.sigmoid(mlp.forward(X, train=False)) > 0.5) == (y > 0.5))
    assert acc.float().mean() > 0.9


def test_dense_layer_backward_matches_autograd():
    g = torch.Generator().manual_seed(3)
    x = torch.randn(32, 16, generator=g)
    layer = DenseLayer(16, 8, act="relu", optimizer="adam", device="cpu",
                       seed=5)
    W = layer.W.clone().requires_grad_(True)
    b = layer.b.clone().requires_grad_(True)
    y_ref = torch.relu(x @ W.t() + b)
    dy = torch.randn(32, 8, generator=g)
    y_ref.backward(dy)
    y, _ = layer.forward(x)
    dx = layer.backward(dy)
    assert torch.allclose(y, y_ref.detach(), atol=1e-6)
    assert torch.allclose(layer._dW, W.grad, atol=1e-5)
    assert torch.allclose(layer._db, b.grad, atol=1e-5)
    xg = x.clone().requires_grad_(True)
    torch.relu(xg @ W.detach().t() + b.detach()).backward(dy)
    assert torch.allclose(dx, xg.grad, atol=1e-5)


def test_nfm_cpu_convergence():
    gen = SyntheticCriteo(num_features=1 << 13, seed=31)
    row_ptr, fields, fids, vals, labels = gen.batch(1024)
    ds = LibffmDataset(row_ptr, fields, fids, vals, labels)
    h = NFMHyper(num_features=1 << 13, k=8, hidden=(32,))
    tr = NFMTrainer(ds, h, device="cpu", batch_size=256, epochs=3)
    m0 = tr.evaluate()
    tr.train(log=None)
    m1 = tr.evaluate()
    assert m1["logloss"] < m0["logloss"]
    assert m1["auc"] > 0.55


def test_widedeep_cpu_convergence():
    gen = SyntheticCriteo(num_features=1 << 13, seed=37)
    row_ptr, fields, fids, vals, labels = gen.batch(1024)
    ds = LibffmDataset(row_ptr, fields, fids, vals, labels)
    h = WideDeepHyper(num_features=1 << 13, num_fields=39, k=8,
                      hidden=(64, 32))
    tr = WideDeepTrainer(ds, h, device="cpu", batch_size=256, epochs=3)
    m0 = tr.evaluate()
    tr.train(log=None)
    m1 = tr.evaluate()
    assert m1["logloss"] < m0["logloss"]
    assert m1["auc"] > 0.55


# ---------------- GPU ----------------

@pytest.mark.gpu
@pytest.mark.parametrize("shape", [(128, 128, 64), (256, 384, 156),
                                   (1000, 250, 100), (64, 1, 256),
                                   (513, 129, 65), (512, 256, 128),
                                   (384, 128, 624), (256, 256, 64),
                                   (512, 512, 192),
                                   # 256x128-tile path (N%256!=0, K%64==0)
                                   (512, 384, 128), (256, 624, 256),
                                   (512, 128, 256),
                                   # 256^2 8-phase path: NT=1/2/tail tiles
                                   (256, 256, 88), (512, 256, 624),
                                   (256, 512, 40)])
def test_gemm_bf16_parity(shape):
    """MFMA GEMM vs torch bf16 matmul (fp32 accumulate) incl. odd tails.

    Asymmetric random operands (transpose-detecting, guide §5.4 rule 16)."""
    from lightctr_amd.ops import hip_ops

    M, N, K = shape
    g = torch.Generator().manual_seed(M + N + K)
    A = (torch.randn(M, K, generator=g) * 0.5).to(torch.bfloat16).cuda()
    Bst = (torch.randn(N, K, generator=g) * 0.5).to(torch.bfloat16).cuda()
    bias = torch.randn(N, generator=g).cuda()
    C = hip_ops.gemm_bf16(A, Bst, bias, M, N, K, 0, 0, 0, False)
    ref = (A.float() @ Bst.float().t()) + bias
    assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5, rtol=1e-2), \
        (C - ref).abs().max()


@pytest.mark.gpu
def test_gemm_bf16_transA_parity():
    from lightctr_amd.ops import hip_ops

    M, N, K = 200, 96, 320  # A stored [K, M]
    g = torch.Generator().manual_seed(9)
    Ast = (torch.randn(K, M, generator=g) * 0.5).to(torch.bfloat16).cuda()
    Bst = (torch.randn(N, K, generator=g) * 0.5).to(torch.bfloat16).cuda()
    C = hip_ops.gemm_bf16(Ast, Bst, None, M, N, K, 1, 0, 0, False)
    ref = Ast.float().t() @ Bst.float().t()
    assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5, rtol=1e-2)


@pytest.mark.gpu
def test_gemm_bf16_transB_parity():
    from lightctr_amd.ops import hip_ops

    M, N, K = 160, 200, 192  # B stored [K, N] (wgrad operand layout)
    g = torch.Generator().manual_seed(10)
    A = (torch.randn(M, K, generator=g) * 0.5).to(torch.bfloat16).cuda()
    B = (torch.randn(K, N, generator=g) * 0.5).to(torch.bfloat16).cuda()
    C = hip_ops.gemm_bf16(A, B, None, M, N, K, 0, 1, 0, False)
    ref = A.float() @ B.float()
    assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5, rtol=1e-2)


@pytest.mark.gpu
def test_gemm_relu_bias_epilogue():
    from lightctr_amd.ops import hip_ops

    M, N, K = 256, 64, 128
    g = torch.Generator().manual_seed(11)
    A = torch.randn(M, K, generator=g).to(torch.bfloat16).cuda()
    Bst = torch.randn(N, K, generator=g).to(torch.bfloat16).cuda()
    bias = torch.randn(N, generator=g).cuda()
    C, Cbf = hip_ops.gemm_bf16_full(A, Bst, bias, M, N, K, 0, 0, 1)
    ref = torch.relu(A.float() @ Bst.float().t() + bias)
    assert torch.allclose(C, ref, atol=0.2, rtol=1e-2)
    assert torch.allclose(Cbf.float(), C, atol=0.1, rtol=1e-2)


@pytest.mark.gpu
def test_mlp_gpu_matches_cpu():
    """One fwd/bwd/step of the bf16 MFMA MLP vs the fp32 CPU oracle."""
    g = torch.Generator().manual_seed(4)
    x = torch.randn(256, 64, generator=g)
    dy = torch.randn(256, 1, generator=g) * 0.1
    cpu = MLP([64, 32, 1], optimizer="adam", lr=1e-3, seed=7, device="cpu")
    gpu = MLP([64, 32, 1], optimizer="adam", lr=1e-3, seed=7, device="cuda:0")
    for lc, lg in zip(cpu.layers, gpu.layers):
        lg.W.copy_(lc.W)
        lg.b.copy_(lc.b)
        lg.Wbf.copy_(lg.W.to(torch.bfloat16))
        lg.Wtbf.copy_(lg.W.t().contiguous().to(torch.bfloat16))
    y_cpu = cpu.forward(x)
    y_gpu = gpu.forward(x.cuda().to(torch.bfloat16))
    assert torch.allclose(y_gpu.cpu(), y_cpu, atol=0.08, rtol=0.05), \
        (y_gpu.cpu() - y_cpu).abs().max()
    dx_cpu = cpu.backward(dy)
    dx_gpu = gpu.backward(dy.cuda())
    assert torch.allclose(dx_gpu.cpu(), dx_cpu, atol=0.02, rtol=0.05), \
        (dx_gpu.cpu() - dx_cpu).abs().max()
    cpu.apply_grads()
    gpu.apply_grads()
    for lc, lg in zip(cpu.layers, gpu.layers):
        assert torch.allclose(lg.W.cpu(), lc.W, atol=5e-3), \
            (lg.W.cpu() - lc.W).abs().max()


@pytest.mark.gpu
def test_nfm_gpu_convergence():
    gen = SyntheticCriteo(num_features=1 << 15, seed=41, device="cuda:0")
    h = NFMHyper(num_features=1 << 15, k=16, hidden=(64,))
    model = NFMModel(h, device="cuda:0")
    losses = []
    for _ in range(25):
        row_ptr, fields, fids, vals, labels = gen.batch(4096)
        loss = model.train_step(row_ptr, fids, vals, labels)
        losses.append(float(loss.mean()))
    assert losses[-1] < losses[0] * 0.99, losses[:3] + losses[-3:]


@pytest.mark.gpu
def test_widedeep_gpu_convergence():
    gen = SyntheticCriteo(num_features=1 << 15, seed=43, device="cuda:0")
    h = WideDeepHyper(num_features=1 << 15, num_fields=39, k=16,
                      hidden=(256, 128))
    model = WideDeepModel(h, device="cuda:0")
    losses = []
    for _ in range(25):
        row_ptr, fields, fids, vals, labels = gen.batch(4096)
        loss = model.train_step(row_ptr, fids, vals, labels)
        losses.append(float(loss.mean()))
    assert losses[-1] < losses[0] * 0.99, losses[:3] + losses[-3:]


@pytest.mark.gpu
def test_nfm_gpu_forward_parity():
    """GPU fused bi-interaction vs CPU reference pieces."""
    from lightctr_amd.ops import hip_ops

    gen = SyntheticCriteo(num_features=1 << 12, seed=47, device="cuda:0")
    row_ptr, fields, fids, vals, labels = gen.batch(64)
    F, K = 1 << 12, 8
    g = torch.Generator().manual_seed(5)
    W = torch.randn(F, generator=g).cuda()
    V = (torch.randn(F, K, generator=g) * 0.1).cuda()
    wide, sumVX, vec, vec_bf = hip_ops.nfm_forward(row_ptr, fids, vals, W, V)
    from lightctr_amd.ops import fm_ref

    pred_ref, sumVX_ref = fm_ref.fm_forward_ref(row_ptr, fids, vals, W, V)
    assert torch.allclose(sumVX, sumVX_ref, atol=1e-4)
    # wide + vec.sum() must equal the FM prediction
    assert torch.allclose(wide + vec.sum(dim=1), pred_ref, atol=1e-3)


@pytest.mark.gpu
def test_gemm_sigmoid_epilogue_and_256path_bf16_emit():
    from lightctr_amd.ops import hip_ops

    M, N, K = 512, 256, 128  # routes through the 256^2 kernel
    g = torch.Generator().manual_seed(21)
    A = (torch.randn(M, K, generator=g) * 0.3).to(torch.bfloat16).cuda()
    Bst = (torch.randn(N, K, generator=g) * 0.3).to(torch.bfloat16).cuda()
    bias = torch.randn(N, generator=g).cuda()
    C, Cbf = hip_ops.gemm_bf16_full(A, Bst, bias, M, N, K, 0, 0, 2)
    ref = torch.sigmoid((A.float() @ Bst.float().t() + bias)
                        .clamp(-16, 16))
    assert torch.allclose(C, ref, atol=0.02, rtol=1e-2)
    assert torch.allclose(Cbf.float(), C, atol=0.01, rtol=1e-2)


@pytest.mark.gpu
def test_gemm_v2_race_screen():
    """Sync-structure kernels need multi-run race screens (guide two-lane
    discipline): 10 repeated runs of a v2-routed shape must all match."""
    from lightctr_amd.ops import hip_ops

    M, N, K = 2048, 2048, 2048
    g = torch.Generator().manual_seed(33)
    A = (torch.randn(M, K, generator=g) * 0.4).to(torch.bfloat16).cuda()
    Bst = (torch.randn(N, K, generator=g) * 0.4).to(torch.bfloat16).cuda()
    ref = A.float() @ Bst.float().t()
    for it in range(10):
        C = hip_ops.gemm_bf16(A, Bst, None, M, N, K, 0, 0, 0, False)
        assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5, rtol=1e-2), \
            (it, (C - ref).abs().max())


@pytest.mark.gpu
def test_gemm_wgrad128_parity_and_race():
    """K-major x K-major wgrad kernel (blocked glds + ds_read_b64_tr_b16
    fragment reads): parity vs fp32 reference across full/partial tiles
    and K-tails, plus a 10-run screen of the split-K atomics."""
    from lightctr_amd.ops import hip_ops

    for (M, N, K) in [(256, 624, 4096), (128, 256, 2048), (256, 256, 1056),
                      (144, 112, 1024)]:
        g = torch.Generator().manual_seed(M + K)
        At = (torch.randn(K, M, generator=g) * 0.5).to(torch.bfloat16).cuda()
        Bt = (torch.randn(K, N, generator=g) * 0.5).to(torch.bfloat16).cuda()
        C = hip_ops.gemm_wgrad_bf16(At, Bt, M, N, K)
        ref = At.float().t() @ Bt.float()
        assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5, rtol=1e-2), \
            (M, N, K, (C - ref).abs().max())
    M, N, K = 256, 624, 8192
    g = torch.Generator().manual_seed(3)
    At = (torch.randn(K, M, generator=g) * 0.5).to(torch.bfloat16).cuda()
    Bt = (torch.randn(K, N, generator=g) * 0.5).to(torch.bfloat16).cuda()
    ref = At.float().t() @ Bt.float()
    for it in range(10):
        C = hip_ops.gemm_wgrad_bf16(At, Bt, M, N, K)
        assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5, rtol=1e-2), it


@pytest.mark.gpu
def test_gemm_p8_race_screen():
    """The 8-phase counted-vmcnt schedule is a new sync structure (guide
    two-lane discipline): repeated runs at a multi-tile shape and a
    tail shape must all match the fp32 reference."""
    from lightctr_amd.ops import hip_ops

    for (M, N, K) in [(1024, 1024, 1024), (512, 256, 624)]:
        g = torch.Generator().manual_seed(41 + K)
        A = (torch.randn(M, K, generator=g) * 0.4).to(torch.bfloat16).cuda()
        Bst = (torch.randn(N, K, generator=g) * 0.4).to(torch.bfloat16).cuda()
        ref = A.float() @ Bst.float().t()
        for it in range(10):
            C = hip_ops.gemm_bf16(A, Bst, None, M, N, K, 0, 0, 0, False)
            assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5, rtol=1e-2), \
                (it, M, N, K, (C - ref).abs().max())


@pytest.mark.gpu
def test_gemm_apipe_schedule_variants():
    """The gemm256 per-phase A-read pipeline (LCTR_GEMM_APIPE, default 2)
    and its env fallbacks all match the fp32 reference; 5-run screen per
    variant since the pipelined tile body is a new schedule."""
    import os

    from lightctr_amd.ops import hip_ops

    M, N, K = 1024, 1024, 1024
    g = torch.Generator().manual_seed(7)
    A = (torch.randn(M, K, generator=g) * 0.4).to(torch.bfloat16).cuda()
    Bst = (torch.randn(N, K, generator=g) * 0.4).to(torch.bfloat16).cuda()
    ref = A.float() @ Bst.float().t()
    old = os.environ.get("LCTR_GEMM_APIPE")
    try:
        for v in ("0", "1", "2"):
            os.environ["LCTR_GEMM_APIPE"] = v
            for it in range(5):
                C = hip_ops.gemm_bf16(A, Bst, None, M, N, K, 0, 0, 0,
                                      False)
                assert torch.allclose(C, ref, atol=2e-2 * K ** 0.5,
                                      rtol=1e-2), (v, it)
    finally:
        if old is None:
            os.environ.pop("LCTR_GEMM_APIPE", None)
        else:
            os.environ["LCTR_GEMM_APIPE"] = old


@pytest.mark.gpu
def test_gemm_degenerate_shapes():
    """The 1-wide-layer fast paths (gemv N=1, weighted rowsum M=1,
    outer product K=1) match the fp32 reference in the exact call
    patterns the MLP uses."""
    from lightctr_amd.ops import hip_ops

    g = torch.Generator().manual_seed(99)
    B, In = 4096, 128
    # forward: C[B,1] = X[B,In] @ w[1,In]^T + bias, sigmoid epilogue
    X = (torch.randn(B, In, generator=g) * 0.5).to(torch.bfloat16).cuda()
    w = (torch.randn(1, In, generator=g) * 0.5).to(torch.bfloat16).cuda()
    bias = torch.randn(1, generator=g).cuda()
    C, Cbf = hip_ops.gemm_bf16_full(X, w, bias, B, 1, In, 0, 0, 2)
    z = X.float() @ w.float().t() + bias
    ref = torch.sigmoid(torch.clamp(z, -16, 16))
    assert torch.allclose(C, ref, atol=3e-2, rtol=1e-2), (C - ref).abs().max()
    assert torch.allclose(Cbf.float(), ref, atol=5e-2), "bf16 mirror"

    # wgrad: dW[1,In] = dZ[B,1]^T @ X[B,In]  (transA=1, transB=1, no bias)
    dZ = (torch.randn(B, 1, generator=g) * 0.1).to(torch.bfloat16).cuda()
    dW = hip_ops.gemm_bf16(dZ, X, None, 1, In, B, 1, 1, 0, False)
    ref = dZ.float().t() @ X.float()
    assert torch.allclose(dW, ref, atol=2e-1, rtol=1e-2), \
        (dW - ref).abs().max()

    # small-output wgrad: dW[64,16] = dZ[B,64]^T @ X[B,16], deep K
    B2 = 16384
    dZ2 = (torch.randn(B2, 64, generator=g) * 0.1).to(torch.bfloat16).cuda()
    X2 = (torch.randn(B2, 16, generator=g) * 0.5).to(torch.bfloat16).cuda()
    dW2 = hip_ops.gemm_bf16(dZ2, X2, None, 64, 16, B2, 1, 1, 0, False)
    ref2 = dZ2.float().t() @ X2.float()
    assert torch.allclose(dW2, ref2, atol=5e-1, rtol=1e-2),         (dW2 - ref2).abs().max()

    # dgrad: dX[B,In] = dZ[B,1] @ Wt[In,1]^T  (K=1 outer product)
    Wt = (torch.randn(In, 1, generator=g) * 0.5).to(torch.bfloat16).cuda()
    dX = hip_ops.gemm_bf16(dZ, Wt, None, B, In, 1, 0, 0, 0, False)
    ref = dZ.float() @ Wt.float().t()
    assert torch.allclose(dX, ref, atol=1e-3, rtol=1e-2), \
        (dX - ref).abs().max()


def test_gradient_clipping():
    """DenseLayer clips backprop deltas elementwise at +-clip before
    forming weight grads (the reference FC error_clip_threshold
    semantics); matches a manual clamp oracle."""
    from lightctr_amd.models.mlp import DenseLayer

    torch.manual_seed(0)
    la = DenseLayer(8, 4, act="none", optimizer="adagrad", lr=0.1,
                    clip=1.0, device="cpu")
    x = torch.randn(16, 8)
    la.forward(x, train=True)
    dy = torch.randn(16, 4) * 100.0  # exploding deltas
    la.backward(dy)
    ref_dW = torch.clamp(dy, -1.0, 1.0).t() @ x
    assert torch.allclose(la._dW, ref_dW, atol=1e-5)
    assert la._dW.abs().max() <= 16 * 8  # bounded by clip * batch

    # clip=0 disables
    la2 = DenseLayer(8, 4, act="none", optimizer="adagrad", lr=0.1,
                     device="cpu")
    la2.forward(x, train=True)
    la2.backward(dy)
    assert torch.allclose(la2._dW, dy.t() @ x, atol=1e-4)
