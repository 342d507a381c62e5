"""Python wrappers over the native core (`byteps_amd.ops._core`).

GPU tensors run the hand-written gfx950 HIP kernels; CPU tensors fall back
to the OpenMP reducer/codecs in the same .so.  On a machine **with** a GPU
the HIP extension is mandatory — a missing .so raises instead of silently
falling back to eager PyTorch.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

try:
    from . import _core  # type: ignore
    _core_err: Optional[Exception] = None
except Exception as e:                                    # pragma: no cover
    _core = None
    _core_err = e


def core():
    if _core is None:
        if torch.cuda.is_available():
            raise RuntimeError(
                "byteps_amd native core (_core.so) is missing on a GPU "
                "machine — build it with `python -m byteps_amd.ops.build` "
                "(original import error: %r)" % (_core_err,))
        raise RuntimeError(
            "byteps_amd native core not built: %r" % (_core_err,))
    return _core


def have_core() -> bool:
    return _core is not None


_DTYPE_CODE = {torch.float32: 0, torch.float16: 1, torch.bfloat16: 2,
               torch.float64: 3}


def _code(t: torch.Tensor) -> int:
    try:
        return _DTYPE_CODE[t.dtype]
    except KeyError:
        raise TypeError("unsupported dtype %s" % t.dtype)


def _stream(t: torch.Tensor) -> int:
    if t.is_cuda:
        return torch.cuda.current_stream(t.device).cuda_stream
    return 0


def _check_gpu(t: torch.Tensor) -> None:
    assert t.is_contiguous(), "kernel requires contiguous tensor"
    assert t.data_ptr() % 16 == 0, "kernel requires 16-byte alignment"


# -- elementwise ------------------------------------------------------------

def scale_(t: torch.Tensor, alpha: float) -> torch.Tensor:
    """In-place t *= alpha via the HIP kernel (GPU) or OpenMP (CPU)."""
    if t.is_cuda:
        _check_gpu(t)
        core().scale(t.data_ptr(), t.numel(), alpha, _code(t), _stream(t))
    else:
        t.mul_(alpha)
    return t


def cast_scale_(dst: torch.Tensor, src: torch.Tensor,
                alpha: float) -> torch.Tensor:
    """dst = float(src) * alpha in one fused pass — the averaging divide
    folded into the reduced-precision wire cast-back."""
    if dst.is_cuda:
        _check_gpu(dst), _check_gpu(src)
        assert dst.numel() == src.numel()
        core().cast_scale(dst.data_ptr(), src.data_ptr(), dst.numel(), alpha,
                          _code(src), _code(dst), _stream(dst))
    else:
        torch.mul(src.reshape(-1).to(dst.dtype), alpha,
                  out=dst.reshape(-1))
    return dst


def build_cast_scale_desc(dsts, srcs):
    """Pack (dst_ptr, src_ptr, vec_prefix) descriptors for
    :func:`cast_scale_many` into one device int64 tensor.  Every segment's
    numel must divide the vector width (engine buckets are lcm(64, world)
    aligned).  Returns (desc_tensor, total_vec, vec)."""
    n = len(dsts)
    assert n > 0 and len(srcs) == n
    vec = 8 if (dsts[0].element_size() == 2 and srcs[0].element_size() == 2) \
        else 4
    arr = torch.empty(3 * n + 1, dtype=torch.int64)
    pref = 0
    for i, (d, s) in enumerate(zip(dsts, srcs)):
        assert d.numel() == s.numel() and d.numel() % vec == 0
        arr[i] = d.data_ptr()
        arr[n + i] = s.data_ptr()
        arr[2 * n + i] = pref
        pref += d.numel() // vec
    arr[3 * n] = pref
    return arr.to(dsts[0].device), pref, vec


def cast_scale_many_(desc: torch.Tensor, nseg: int, total_vec: int,
                     alpha: float, src_dtype: torch.dtype,
                     dst_dtype: torch.dtype, device) -> None:
    """One launch over all buckets: dst_i = float(src_i) * alpha."""
    core().cast_scale_many(
        desc.data_ptr(), nseg, total_vec, alpha,
        _DTYPE_CODE[src_dtype], _DTYPE_CODE[dst_dtype],
        torch.cuda.current_stream(device).cuda_stream)


def axpy_(y: torch.Tensor, x: torch.Tensor, alpha: float = 1.0) -> torch.Tensor:
    """In-place y += alpha*x (fp32 accumulate for bf16)."""
    if y.is_cuda:
        _check_gpu(y), _check_gpu(x)
        assert y.dtype == x.dtype and y.numel() == x.numel()
        core().axpy(y.data_ptr(), x.data_ptr(), y.numel(), alpha, _code(y),
                    _stream(y))
    else:
        y.add_(x, alpha=alpha)
    return y


def nesterov_(grad: torch.Tensor, mom: torch.Tensor, mu: float) -> None:
    """Fused m = mu*m + g ; g += mu*m (reference
    impl/nesterov_momentum.cc:39-49)."""
    if grad.is_cuda:
        _check_gpu(grad), _check_gpu(mom)
        core().nesterov(grad.data_ptr(), mom.data_ptr(), grad.numel(), mu,
                        _code(grad), _stream(grad))
    else:
        mom.mul_(mu).add_(grad)
        grad.add_(mom, alpha=mu)


def norm(t: torch.Tensor, mode: str = "l1") -> torch.Tensor:
    """Returns a 1-element tensor (same device): l1 → Σ|x|, l2 → sqrt(Σx²),
    max → max|x|.  GPU: single fused reduction kernel."""
    m = {"l1": 0, "l2": 1, "max": 2}[mode]
    if t.is_cuda:
        _check_gpu(t)
        out = torch.zeros(1, dtype=torch.float32, device=t.device)
        core().norm(t.data_ptr(), t.numel(), m, out.data_ptr(), _code(t),
                    _stream(t))
        return out.sqrt() if mode == "l2" else out
    x = t.float()
    if mode == "l1":
        return x.abs().sum().reshape(1)
    if mode == "l2":
        return x.norm().reshape(1)
    return x.abs().max().reshape(1)


# -- onebit -----------------------------------------------------------------

def onebit_compress(x: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """→ (bits uint64 [ceil(n/64)], scale_sum float[1]); sign-pack with
    L1/n scale (reference impl/onebit.cc:34-71)."""
    n = x.numel()
    nwords = (n + 63) // 64
    if x.is_cuda:
        _check_gpu(x)
        assert x.dtype == torch.float32
        bits = torch.empty(nwords, dtype=torch.int64, device=x.device)
        sc = torch.zeros(1, dtype=torch.float32, device=x.device)
        core().onebit_compress(x.data_ptr(), n, bits.data_ptr(), sc.data_ptr(),
                               _stream(x))
        return bits, sc
    xf = x.float().contiguous()
    bits = torch.zeros(nwords, dtype=torch.int64)
    sc_val = core().cpu_onebit_compress(xf.data_ptr(), n, bits.data_ptr()) \
        if have_core() else None
    if sc_val is None:                     # pure-torch golden fallback
        raise RuntimeError("native core required")
    return bits, torch.tensor([sc_val], dtype=torch.float32)


def onebit_decompress(bits: torch.Tensor, scale_sum: torch.Tensor,
                      n: int, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if out is None:
        out = torch.empty(n, dtype=torch.float32, device=bits.device)
    if bits.is_cuda:
        core().onebit_decompress(bits.data_ptr(), scale_sum.data_ptr(), n,
                                 out.data_ptr(), _stream(bits))
    else:
        core().cpu_onebit_decompress(bits.data_ptr(),
                                     float(scale_sum.item()), n,
                                     out.data_ptr())
    return out


def onebit_error(x: torch.Tensor, bits: torch.Tensor,
                 scale_sum: torch.Tensor,
                 err: torch.Tensor) -> torch.Tensor:
    """err = x - decompress(bits) — fused error-feedback update
    (reference impl/onebit.cc:113-140)."""
    n = x.numel()
    if x.is_cuda:
        core().onebit_error(x.data_ptr(), bits.data_ptr(),
                            scale_sum.data_ptr(), n, err.data_ptr(),
                            _stream(x))
    else:
        dec = onebit_decompress(bits, scale_sum, n)
        torch.sub(x.reshape(-1), dec, out=err.reshape(-1))
    return err


# -- sparse (randomk / topk) ------------------------------------------------

def randomk_compress(x: torch.Tensor, k: int,
                     seed: int) -> Tuple[torch.Tensor, torch.Tensor]:
    n = x.numel()
    if x.is_cuda:
        _check_gpu(x)
        idx = torch.empty(k, dtype=torch.int32, device=x.device)
        val = torch.empty(k, dtype=torch.float32, device=x.device)
        core().randomk_compress(x.data_ptr(), n, k, seed, idx.data_ptr(),
                                val.data_ptr(), _stream(x))
        return idx, val
    idx = torch.empty(k, dtype=torch.int32)
    core().cpu_randomk_indices(n, k, seed, idx.data_ptr())
    val = x.reshape(-1).float()[idx.long()].contiguous()
    return idx, val


def topk_compress(x: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """k largest-|x| as (idx int32, val fp32).  Selection via torch.topk
    (rocPRIM — a library call, like GEMMs via hipBLASLt); gather fused by
    the HIP kernel on GPU (reference impl/topk.cc:43-78 used a CPU heap)."""
    flat = x.reshape(-1)
    _, idx64 = torch.topk(flat.abs(), k, sorted=False)
    idx = idx64.to(torch.int32).contiguous()
    if x.is_cuda:
        val = torch.empty(k, dtype=torch.float32, device=x.device)
        core().sparse_gather(flat.data_ptr(), idx.data_ptr(), k,
                             val.data_ptr(), _stream(x))
    else:
        val = flat.float()[idx64].contiguous()
    return idx, val


def sparse_decompress(idx: torch.Tensor, val: torch.Tensor, n: int,
                      out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if out is None:
        out = torch.zeros(n, dtype=torch.float32, device=idx.device)
    else:
        out.zero_()
    k = idx.numel()
    if idx.is_cuda:
        core().sparse_scatter(idx.data_ptr(), val.data_ptr(), k,
                              out.data_ptr(), _stream(idx))
    else:
        core().cpu_sparse_scatter(idx.data_ptr(), val.data_ptr(), k,
                                  out.data_ptr())
    return out


def sparse_error(x: torch.Tensor, idx: torch.Tensor,
                 err: torch.Tensor) -> torch.Tensor:
    """err = x with transmitted coordinates zeroed."""
    err.reshape(-1).copy_(x.reshape(-1))
    k = idx.numel()
    if x.is_cuda:
        core().sparse_error_zero(idx.data_ptr(), k, err.data_ptr(),
                                 _stream(x))
    else:
        err.reshape(-1)[idx.long()] = 0
    return err


# -- fp8 e4m3 wire -----------------------------------------------------------

def fp8_compress(x: torch.Tensor,
                 amax_t: Optional[torch.Tensor] = None
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    """→ (codes uint8 [n], amax float[1]).  OCP e4m3fn with per-partition
    448/amax scaling; GPU and CPU produce identical bytes."""
    n = x.numel()
    if amax_t is None:
        amax_t = norm(x, "max")
    if x.is_cuda:
        _check_gpu(x)
        code = torch.empty(n, dtype=torch.uint8, device=x.device)
        core().fp8_compress(x.data_ptr(), n, amax_t.data_ptr(),
                            code.data_ptr(), _stream(x))
        return code, amax_t
    code = torch.empty(n, dtype=torch.uint8)
    core().cpu_fp8_compress(x.float().contiguous().data_ptr(), n,
                            float(amax_t.item()), code.data_ptr())
    return code, amax_t


def fp8_decompress(code: torch.Tensor, amax_t: torch.Tensor,
                   out: Optional[torch.Tensor] = None) -> torch.Tensor:
    n = code.numel()
    if out is None:
        out = torch.empty(n, dtype=torch.float32, device=code.device)
    if code.is_cuda:
        core().fp8_decompress(code.data_ptr(), n, amax_t.data_ptr(),
                              out.data_ptr(), _stream(code))
    else:
        core().cpu_fp8_decompress(code.data_ptr(), n, float(amax_t.item()),
                                  out.data_ptr())
    return out


# -- dithering --------------------------------------------------------------

def dithering_compress(x: torch.Tensor, s: int, seed: int,
                       natural: bool = False,
                       norm_t: Optional[torch.Tensor] = None
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """→ (codes int8 [n], norm float[1]).  Stochastic quantization
    (reference impl/dithering.cc:51-121); linear partitions use max-norm,
    natural uses L2."""
    n = x.numel()
    mode = "l2" if natural else "max"
    if norm_t is None:
        norm_t = norm(x, mode)
    if x.is_cuda:
        _check_gpu(x)
        code = torch.empty(n, dtype=torch.int8, device=x.device)
        core().dithering_compress(x.data_ptr(), n, s, seed, int(natural),
                                  norm_t.data_ptr(), code.data_ptr(),
                                  _stream(x))
        return code, norm_t
    code = torch.empty(n, dtype=torch.int8)
    core().cpu_dithering_compress(x.float().contiguous().data_ptr(), n, s,
                                  seed, int(natural), float(norm_t.item()),
                                  code.data_ptr())
    return code, norm_t


def dithering_decompress(code: torch.Tensor, norm_t: torch.Tensor, s: int,
                         natural: bool = False,
                         out: Optional[torch.Tensor] = None) -> torch.Tensor:
    n = code.numel()
    if out is None:
        out = torch.empty(n, dtype=torch.float32, device=code.device)
    if code.is_cuda:
        core().dithering_decompress(code.data_ptr(), n, s, int(natural),
                                    norm_t.data_ptr(), out.data_ptr(),
                                    _stream(code))
    else:
        core().cpu_dithering_decompress(code.data_ptr(), n, s, int(natural),
                                        float(norm_t.item()), out.data_ptr())
    return out
