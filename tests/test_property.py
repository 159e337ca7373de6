"""Property-based tests (hypothesis, derandomized for CI determinism)
for the compression codecs and updater math — the reference pins these
with hand-picked examples (quantile_compress.h, gradientUpdater.h);
properties cover the input space more densely.
"""
import hypothesis.strategies as st
import numpy as np
import torch
from hypothesis import given, settings

from lightctr_amd.utils.compress import (Fp16Codec, LowBitCodec,
                                         ProductQuantizer, QuantileCodec)
from lightctr_amd.utils.updaters import FTRL

SET = dict(derandomize=True, max_examples=40, deadline=None)

floats = st.lists(
    st.floats(min_value=-1.0, max_value=1.0, allow_nan=False,
              allow_infinity=False, width=32),
    min_size=1, max_size=200)


@settings(**SET)
@given(floats)
def test_quantile_uniform_roundtrip_error_bound(xs):
    """decode(encode(x)) is within half a bucket of x, and encoding is
    monotone (order-preserving up to ties) — reference
    quantile_compress.h:137-148 binary-search semantics."""
    x = torch.tensor(xs, dtype=torch.float32)
    c = QuantileCodec(levels=256, mode="uniform", lo=-1.0, hi=1.0)
    code = c.encode(x)
    y = c.decode(code)
    # uniform bucket width = 2/256; nearest-level error <= half width
    assert (y - x).abs().max().item() <= (2.0 / 256) / 2 + 1e-6
    # monotone: sorting inputs sorts codes
    order = torch.argsort(x)
    sc = code[order].to(torch.int16)
    assert bool((sc[1:] >= sc[:-1]).all())


@settings(**SET)
@given(floats)
def test_quantile_log_idempotent(xs):
    """Quantization is idempotent: decode∘encode is a projection."""
    x = torch.tensor(xs, dtype=torch.float32)
    c = QuantileCodec(levels=256, mode="log", lo=-1.0, hi=1.0)
    y = c.decode(c.encode(x))
    z = c.decode(c.encode(y))
    assert torch.equal(y, z)


@settings(**SET)
@given(floats)
def test_fp16_codec_eps(xs):
    x = torch.tensor(xs, dtype=torch.float32)
    y = Fp16Codec.decode(Fp16Codec.encode(x))
    # fp16 relative error 2^-11 in [-1,1], absolute <= ~5e-4
    assert (y - x).abs().max().item() <= 1e-3


@settings(**SET)
@given(floats)
def test_lowbit_sign_preserved(xs):
    """1-bit codec preserves sign of every nonzero element and the
    decoded magnitude is the mean |x| (reference lowbit_quantize
    product_quantizer.h:24-45 semantics)."""
    x = torch.tensor(xs, dtype=torch.float32)
    codec = LowBitCodec(bits=1)
    words, meta, n = codec.encode(x)
    y = codec.decode(words, meta, n)
    nz = x != 0
    assert bool((torch.sign(y[nz]) == torch.sign(x[nz])).all())


@settings(**SET)
@given(st.integers(min_value=0, max_value=2**31 - 1))
def test_pq_roundtrip_projection(seed):
    """PQ decode(encode(.)) is a projection onto centroid products:
    applying it twice equals applying it once."""
    g = torch.Generator().manual_seed(seed)
    X = torch.randn(64, 8, generator=g)
    pq = ProductQuantizer(dim=8, n_sub=2, n_centroids=8, iters=3,
                          seed=seed & 0xFFFF)
    pq.fit(X)
    Y = pq.decode(pq.encode(X))
    Z = pq.decode(pq.encode(Y))
    assert torch.allclose(Y, Z, atol=1e-5)


@settings(**SET)
@given(st.floats(min_value=-0.5, max_value=0.5, width=32),
       st.floats(min_value=0.625, max_value=5.0, width=32))
def test_ftrl_l1_dead_zone(z_small, l1):
    """FTRL-proximal: a weight whose accumulated z stays inside the L1
    ball is exactly zero (reference gradientUpdater.h:235-278)."""
    u = FTRL(alpha=0.1, beta=1.0, l1=l1, l2=0.0)
    w = torch.zeros(1)
    # one tiny gradient keeps |z| = |g| <= 0.5 < l1 (w starts at 0)
    u.update(w, torch.tensor([z_small]))
    assert w.item() == 0.0
