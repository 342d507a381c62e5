"""HIP kernel numerics on MI355X: every gfx950 kernel against a plain
PyTorch fp32 (or CPU-codec) reference."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from byteps_amd import compression as comp  # noqa: E402
from byteps_amd import ops as K  # noqa: E402


def _cuda(n, seed=0, dtype=torch.float32):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return torch.randn(n, generator=g).to("cuda", dtype)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("n", [64, 4096, 1 << 20, (1 << 20) + 13])
def test_scale(dtype, n):
    x = _cuda(n, 1, dtype)
    ref = x.float() * 0.125
    K.scale_(x, 0.125)
    torch.cuda.synchronize()
    tol = 1e-6 if dtype == torch.float32 else 1e-2
    assert torch.allclose(x.float(), ref, rtol=tol, atol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("n", [64, 4096, (1 << 20) + 5])
def test_axpy(dtype, n):
    y = _cuda(n, 2, dtype)
    x = _cuda(n, 3, dtype)
    ref = y.float() + 0.5 * x.float()
    K.axpy_(y, x, 0.5)
    torch.cuda.synchronize()
    tol = 1e-6 if dtype == torch.float32 else 2e-2
    assert torch.allclose(y.float(), ref, rtol=tol, atol=tol)


@pytest.mark.parametrize("mode", ["l1", "l2", "max"])
@pytest.mark.parametrize("n", [64, 100000, (1 << 21) + 9])
def test_norm(mode, n):
    x = _cuda(n, 4)
    got = K.norm(x, mode)
    torch.cuda.synchronize()
    ref = {"l1": x.abs().sum(), "l2": x.norm(), "max": x.abs().max()}[mode]
    assert torch.allclose(got.squeeze(), ref, rtol=1e-4, atol=1e-4)


def test_nesterov():
    g = _cuda(100000, 5)
    m = _cuda(100000, 6)
    g_ref, m_ref = g.clone(), m.clone()
    K.nesterov_(g, m, 0.9)
    torch.cuda.synchronize()
    m_ref.mul_(0.9).add_(g_ref)
    g_ref.add_(m_ref, alpha=0.9)
    assert torch.allclose(g, g_ref, rtol=1e-6, atol=1e-6)
    assert torch.allclose(m, m_ref, rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("n", [64, 1000, (1 << 20) + 17])
def test_onebit_gpu_matches_cpu(n):
    xg = _cuda(n, 7)
    xc = xg.cpu()
    bits_g, sc_g = K.onebit_compress(xg)
    torch.cuda.synchronize()
    bits_c, sc_c = K.onebit_compress(xc)
    assert torch.equal(bits_g.cpu(), bits_c), "bit-exact sign packing"
    assert torch.allclose(sc_g.cpu(), sc_c, rtol=1e-4)
    out_g = K.onebit_decompress(bits_g, sc_g, n)
    torch.cuda.synchronize()
    scale = xg.abs().sum() / n
    assert torch.allclose(out_g.abs(), scale.expand(n), rtol=1e-4)
    assert torch.equal(out_g.sign(), torch.where(xg >= 0, 1.0, -1.0))


def test_onebit_error_fused():
    n = 12345
    x = _cuda(n, 8)
    bits, sc = K.onebit_compress(x)
    err = torch.empty_like(x)
    K.onebit_error(x, bits, sc, err)
    torch.cuda.synchronize()
    dec = K.onebit_decompress(bits, sc, n)
    assert torch.allclose(err, x - dec, rtol=1e-5, atol=1e-6)


def test_randomk_gpu_matches_cpu():
    n, k = 100000, 512
    xg = _cuda(n, 9)
    idx_g, val_g = K.randomk_compress(xg, k, seed=42)
    torch.cuda.synchronize()
    idx_c, val_c = K.randomk_compress(xg.cpu(), k, seed=42)
    assert torch.equal(idx_g.cpu(), idx_c), "counter-mode RNG must agree"
    assert torch.allclose(val_g.cpu(), val_c)
    out = K.sparse_decompress(idx_g, val_g, n)
    torch.cuda.synchronize()
    assert torch.allclose(out[idx_g.long()], xg[idx_g.long()])
    mask = torch.ones(n, dtype=torch.bool, device="cuda")
    mask[idx_g.long()] = False
    assert (out[mask] == 0).all()


def test_topk_gpu():
    n, k = 65536, 256
    x = _cuda(n, 10)
    idx, val = K.topk_compress(x, k)
    torch.cuda.synchronize()
    ref_val, ref_idx = torch.topk(x.abs(), k)
    assert torch.allclose(val.abs().sort(descending=True).values,
                          ref_val.sort(descending=True).values)
    assert torch.allclose(val, x[idx.long()])


@pytest.mark.parametrize("natural", [False, True])
def test_dithering_gpu_matches_cpu(natural):
    n, s = 100000, 64
    xg = _cuda(n, 11)
    code_g, norm_g = K.dithering_compress(xg, s, seed=77, natural=natural)
    torch.cuda.synchronize()
    code_c, norm_c = K.dithering_compress(xg.cpu(), s, seed=77,
                                          natural=natural)
    assert torch.allclose(norm_g.cpu(), norm_c, rtol=1e-4)
    # same seed + same norm ⇒ same stochastic decisions (counter-mode RNG);
    # norm may differ in last ulp between GPU/CPU reductions, so compare
    # with the same norm forced
    code_g2, _ = K.dithering_compress(xg, s, seed=77, natural=natural,
                                      norm_t=norm_c.to("cuda"))
    torch.cuda.synchronize()
    agree = (code_g2.cpu() == code_c).float().mean().item()
    assert agree > 0.9999, agree
    out = K.dithering_decompress(code_g, norm_g, s, natural)
    torch.cuda.synchronize()
    if not natural:
        bound = (norm_g.item() / s) * 1.001 + 1e-7
        assert (out - xg).abs().max().item() <= bound


def test_compressor_classes_on_gpu():
    n = 1 << 18
    x = _cuda(n, 12)
    for params in (
        {"compressor_type": "onebit", "ef_type": "vanilla"},
        {"compressor_type": "topk", "compressor_k": 1024},
        {"compressor_type": "randomk", "compressor_k": 1024},
        {"compressor_type": "dithering", "compressor_k": 64},
        {"compressor_type": "onebit", "ef_type": "vanilla",
         "momentum_type": "nesterov"},
    ):
        c = comp.create(dict(params))
        g = x.clone()
        cp = c.compress(g)
        payload = comp.BaseCompressor._payload_cat(cp)
        out = c.decompress(payload, n, cp.aux)
        assert out.shape[0] == n and out.is_cuda
        torch.cuda.synchronize()


def test_fp8_gpu_matches_cpu():
    n = 100000
    xg = _cuda(n, 13) * 5
    code_g, amax_g = K.fp8_compress(xg)
    torch.cuda.synchronize()
    code_c, amax_c = K.fp8_compress(xg.cpu(), amax_t=amax_g.cpu())
    assert torch.equal(code_g.cpu(), code_c), "fp8 bytes must be bit-identical"
    out = K.fp8_decompress(code_g, amax_g)
    torch.cuda.synchronize()
    out_c = K.fp8_decompress(code_c, amax_c)
    assert torch.allclose(out.cpu(), out_c)
    rel = (out - xg).abs() / (xg.abs() + 1e-9)
    normal = xg.abs() > amax_g / 448 * 2 ** -6
    assert rel[normal].max().item() < 1 / 16 + 1e-3


# -- fused cast+scale (hot-path averaging epilogue) --------------------------

@pytest.mark.parametrize("src_dt,dst_dt", [
    (torch.float32, torch.float32),
    (torch.bfloat16, torch.float32),
    (torch.float16, torch.float32),
    (torch.bfloat16, torch.bfloat16),
])
def test_cast_scale_single(src_dt, dst_dt):
    n = (1 << 18) + 64
    src = _cuda(n, 7, src_dt)
    dst = torch.empty(n, dtype=dst_dt, device="cuda")
    ref = (src.float() * 0.125).to(dst_dt).float()
    K.cast_scale_(dst, src, 0.125)
    torch.cuda.synchronize()
    assert torch.equal(dst.float(), ref)


@pytest.mark.parametrize("src_dt,dst_dt", [
    (torch.float32, torch.float32),
    (torch.bfloat16, torch.float32),
    (torch.bfloat16, torch.bfloat16),
])
def test_cast_scale_many(src_dt, dst_dt):
    """Multi-bucket descriptor kernel — the exact engine epilogue shape
    (segment lengths are multiples of 64, as bucket alignment guarantees)."""
    sizes = [64, 4096, 1 << 16, 192, (1 << 18)]
    srcs = [_cuda(n, 11 + i, src_dt) for i, n in enumerate(sizes)]
    dsts = [torch.empty(n, dtype=dst_dt, device="cuda") for n in sizes]
    refs = [(s.float() * 0.25).to(dst_dt).float() for s in srcs]
    desc, total_vec, _vec = K.build_cast_scale_desc(dsts, srcs)
    K.cast_scale_many_(desc, len(sizes), total_vec, 0.25, src_dt, dst_dt,
                       dsts[0].device)
    torch.cuda.synchronize()
    for d, r in zip(dsts, refs):
        assert torch.equal(d.float(), r)


def test_engine_epilogue_uses_fused_kernel():
    """GradEngine at world=1 never averages, so drive the epilogue
    directly: buffers divided in-place by the descriptor kernel must match
    _foreach_div_."""
    import byteps_amd.torch as bps
    from byteps_amd.torch.engine import GradEngine
    bps.init()
    m = torch.nn.Sequential(torch.nn.Linear(128, 256),
                            torch.nn.Linear(256, 32)).cuda()
    eng = GradEngine(list(m.named_parameters()))
    for b in eng.buckets:
        b.buffer.normal_()
    refs = [b.buffer.float() / 4.0 for b in eng.buckets]
    eng.world = 4          # pretend world for the divide
    eng._apply_wire_and_average(needs_wire=False, divide=True)
    torch.cuda.synchronize()
    for b, r in zip(eng.buckets, refs):
        assert torch.allclose(b.buffer.float(), r, rtol=1e-6, atol=1e-7)
    bps.shutdown()
