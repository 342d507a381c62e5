"""Compression codec numerics against NumPy golden models (the reference's
test pattern: re-implement each compressor in NumPy and require the native
implementation to match element-for-element — tests/test_onebit.py:32-116,
tests/utils.py:31-52)."""

import numpy as np
import pytest
import torch

from byteps_amd import compression as comp
from byteps_amd import ops as K


def _rand(n, seed=0):
    rng = np.random.default_rng(seed)
    return torch.from_numpy(rng.standard_normal(n).astype(np.float32))


# -- golden models ----------------------------------------------------------

def golden_onebit(x: np.ndarray):
    scale = np.abs(x).sum() / x.size
    signs = np.where(x >= 0, 1.0, -1.0).astype(np.float32)
    return signs * scale


def golden_dithering_linear_bounds(x: np.ndarray, s: int):
    norm = np.abs(x).max()
    return norm / s  # stochastic rounding error bound per element


# -- tests ------------------------------------------------------------------

@pytest.mark.parametrize("n", [64, 100, 1000, 65536 + 7])
def test_onebit_matches_golden(n):
    x = _rand(n)
    c = comp.OnebitCompressor()
    out = c.decompress(comp.BaseCompressor._payload_cat(c.compress(x)), n)
    expect = golden_onebit(x.numpy())
    np.testing.assert_allclose(out.numpy(), expect, rtol=1e-5, atol=1e-6)


def test_onebit_error_feedback_converges():
    """EF property: err = x - Q(x); compressing (x + err) repeatedly keeps
    the accumulated estimate unbiased (reference vanilla_ef semantics)."""
    n = 512
    x = _rand(n, seed=1)
    c = comp.ErrorFeedback(comp.OnebitCompressor())
    total = torch.zeros(n)
    for _ in range(200):
        g = x.clone()
        out = c.decompress(comp.BaseCompressor._payload_cat(c.compress(g)), n)
        total += out
    mean_est = total / 200
    # with EF the long-run average approaches x
    assert (mean_est - x).abs().mean() < 0.1 * x.abs().mean()


@pytest.mark.parametrize("k", [1, 16, 100])
def test_topk(k):
    n = 300
    x = _rand(n, seed=2)
    c = comp.TopkCompressor(k)
    cp = c.compress(x)
    out = c.decompress(comp.BaseCompressor._payload_cat(cp), n, cp.aux)
    # golden: zero except k largest |x|
    xa = x.numpy()
    keep = np.argsort(-np.abs(xa), kind="stable")[:k]
    expect = np.zeros(n, dtype=np.float32)
    expect[keep] = xa[keep]
    # ties can pick either index — compare magnitudes of nonzeros
    got_nz = np.sort(np.abs(out.numpy()[out.numpy() != 0]))
    exp_nz = np.sort(np.abs(expect[expect != 0]))
    np.testing.assert_allclose(got_nz, exp_nz, rtol=1e-6)


def test_randomk_deterministic_indices():
    n, k = 1000, 64
    x = _rand(n, seed=3)
    c1 = comp.RandomkCompressor(k, seed=7)
    c2 = comp.RandomkCompressor(k, seed=7)
    cp1, cp2 = c1.compress(x), c2.compress(x)
    idx1 = cp1.parts[0].view(torch.int32)
    idx2 = cp2.parts[0].view(torch.int32)
    assert torch.equal(idx1, idx2), "same seed/round must give same draws"
    # values are the gathered originals
    val1 = cp1.parts[1].view(torch.float32)
    assert torch.allclose(val1, x[idx1.long()])
    # next round differs
    cp3 = c1.compress(x)
    assert not torch.equal(cp3.parts[0].view(torch.int32), idx1)


@pytest.mark.parametrize("natural", [False, True])
def test_dithering_bounds_and_unbiasedness(natural):
    n, s = 4096, 64
    x = _rand(n, seed=4)
    c = comp.DitheringCompressor(s, natural=natural, seed=9)
    cp = c.compress(x)
    out = c.decompress(comp.BaseCompressor._payload_cat(cp), n)
    xa = x.numpy()
    if not natural:
        bound = golden_dithering_linear_bounds(xa, s) + 1e-6
        assert np.abs(out.numpy() - xa).max() <= bound
    else:
        # natural: relative error ≤ 1 ulp of the power-of-2 grid (factor 2)
        nz = np.abs(xa) > np.abs(xa).max() * 1e-6
        ratio = out.numpy()[nz] / xa[nz]
        assert (ratio > 0).all()
        assert ratio.max() <= 2.0 + 1e-5 and ratio.min() >= 0.5 - 1e-5
    # stochastic rounding is unbiased: average of many rounds ≈ x
    acc = np.zeros(n, dtype=np.float64)
    rounds = 60
    cc = comp.DitheringCompressor(s, natural=natural, seed=10)
    for _ in range(rounds):
        cpi = cc.compress(x)
        acc += cc.decompress(comp.BaseCompressor._payload_cat(cpi), n).numpy()
    est = acc / rounds
    tol = (np.abs(xa).max() / s if not natural else np.abs(est).max()) * 0.2
    assert np.abs(est - xa).mean() < max(tol, 0.05)


def test_registry_chain():
    c = comp.create({"compressor_type": "onebit", "ef_type": "vanilla",
                     "momentum_type": "nesterov"})
    assert isinstance(c, comp.NesterovMomentum)
    assert isinstance(c.inner, comp.ErrorFeedback)
    assert isinstance(c.inner.inner, comp.OnebitCompressor)
    assert comp.create({}) is None


def test_nesterov_momentum_math():
    n = 64
    g = _rand(n, seed=5)
    m = torch.zeros(n)
    g2, m2 = g.clone(), m.clone()
    K.nesterov_(g, m, 0.9)
    # reference math (impl/nesterov_momentum.cc:39-49)
    m2.mul_(0.9).add_(g2)
    g2.add_(m2, alpha=0.9)
    assert torch.allclose(g, g2) and torch.allclose(m, m2)


def test_fp8_roundtrip_and_bounds():
    x = _rand(10000, seed=6) * 3
    c = comp.Fp8Compressor()
    cp = c.compress(x)
    assert cp.nbytes == 10000 + 4     # 4x compression + amax
    out = c.decompress(comp.BaseCompressor._payload_cat(cp), 10000)
    amax = x.abs().max()
    rel = (out - x).abs() / (x.abs() + 1e-9)
    normal = x.abs() > float(amax) / 448 * 2 ** -6
    assert rel[normal].max() < 1 / 16 + 1e-3          # e4m3 mantissa bound
    assert (out - x).abs()[~normal].max() <= float(amax) / 448 * 2 ** -6
    # deterministic (no RNG)
    cp2 = comp.Fp8Compressor().compress(x)
    assert torch.equal(cp.parts[1], cp2.parts[1])


def test_fp8_registry():
    c = comp.create({"compressor_type": "fp8", "ef_type": "vanilla"})
    assert isinstance(c, comp.ErrorFeedback)
    assert c.codec == comp.FP8
