"""Gradient-compression framework: self-registering codec factory with
momentum → error-feedback → compressor decorator chaining (architecture of
reference common/compressor/compressor_registry.cc:39-56 and
common/compressor/compressor.h:53-127; codecs themselves run as HIP
kernels on gfx950, CPU codecs on the server).

Wire formats are defined in ops/csrc/kv.h; ``WireCodec`` ids here must
match.  A compressor instance is per-tensor (it owns error/momentum
state), created from a kwargs dict like the reference's
``compressor_type`` / ``ef_type`` / ``momentum_type`` params
(reference mxnet/__init__.py:250-317).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from .. import ops as K

# wire codec ids (ops/csrc/kv.h WireCodec)
RAW, ONEBIT, TOPK, RANDOMK, DITHER_LINEAR, DITHER_NATURAL, FP8 = range(7)


class Compressed:
    """Wire payload: ordered byte parts + aux word (k for sparse codecs)."""

    __slots__ = ("parts", "aux", "nbytes")

    def __init__(self, parts: List[torch.Tensor], aux: int = 0):
        self.parts = parts
        self.aux = aux
        self.nbytes = sum(p.numel() * p.element_size() for p in parts)


class BaseCompressor:
    codec = RAW
    levels = 0          # init-time parameter shipped to the server

    def compress(self, x: torch.Tensor) -> Compressed:
        raise NotImplementedError

    def decompress(self, payload: torch.Tensor, n: int,
                   aux: int = 0,
                   out: Optional[torch.Tensor] = None) -> torch.Tensor:
        raise NotImplementedError

    def update_error(self, x: torch.Tensor, comp: Compressed,
                     err: torch.Tensor) -> None:
        """err = x - decompress(compress(x)) — overridden with fused
        kernels where available."""
        dec = self.decompress(self._payload_cat(comp), x.numel(), comp.aux)
        torch.sub(x.reshape(-1).float(), dec, out=err.reshape(-1))

    @staticmethod
    def _payload_cat(comp: Compressed) -> torch.Tensor:
        return torch.cat([p.reshape(-1).view(torch.uint8) for p in comp.parts])


class OnebitCompressor(BaseCompressor):
    """Sign bits + L1/n scale (reference impl/onebit.cc)."""
    codec = ONEBIT

    def __init__(self, scaled: bool = True):
        self.scaled = scaled

    def compress(self, x: torch.Tensor) -> Compressed:
        bits, sc = K.onebit_compress(x.reshape(-1))
        if not self.scaled:
            sc = torch.ones_like(sc) * x.numel()
        pad = torch.zeros(1, dtype=torch.float32, device=sc.device)
        self._last = (bits, sc)
        return Compressed([bits.view(torch.uint8),
                           sc.view(torch.uint8), pad.view(torch.uint8)])

    def decompress(self, payload, n, aux=0, out=None):
        nwords = (n + 63) // 64
        bits = payload[:nwords * 8].view(torch.int64)
        sc = payload[nwords * 8:nwords * 8 + 4].view(torch.float32)
        return K.onebit_decompress(bits.contiguous(), sc.contiguous(), n, out)

    def update_error(self, x, comp, err):
        bits, sc = self._last
        K.onebit_error(x.reshape(-1), bits, sc, err.reshape(-1))


class TopkCompressor(BaseCompressor):
    """k largest-|x| (idx, val) pairs (reference impl/topk.cc)."""
    codec = TOPK

    def __init__(self, k: int):
        self.k = max(1, int(k))
        self.levels = self.k

    def compress(self, x: torch.Tensor) -> Compressed:
        k = min(self.k, x.numel())
        idx, val = K.topk_compress(x.reshape(-1), k)
        self._last = idx
        return Compressed([idx.view(torch.uint8), val.view(torch.uint8)],
                          aux=k)

    def decompress(self, payload, n, aux=0, out=None):
        k = aux or self.k
        idx = payload[:k * 4].view(torch.int32).contiguous()
        val = payload[k * 4:k * 8].view(torch.float32).contiguous()
        return K.sparse_decompress(idx, val, n, out)

    def update_error(self, x, comp, err):
        K.sparse_error(x.reshape(-1), self._last, err.reshape(-1))


class RandomkCompressor(BaseCompressor):
    """k pseudo-random (idx, val) pairs, counter-mode RNG so indices are
    reproducible from the seed (reference impl/randomk.cc)."""
    codec = RANDOMK

    def __init__(self, k: int, seed: int = 1):
        self.k = max(1, int(k))
        self.levels = self.k
        self.seed = seed
        self.round = 0

    def compress(self, x: torch.Tensor) -> Compressed:
        k = min(self.k, x.numel())
        # fresh stream every round, identical across ranks
        seed = (self.seed * 0x9E3779B97F4A7C15 + self.round) & (2**64 - 1)
        self.round += 1
        idx, val = K.randomk_compress(x.reshape(-1), k, seed)
        self._last = idx
        return Compressed([idx.view(torch.uint8), val.view(torch.uint8)],
                          aux=k)

    decompress = TopkCompressor.decompress

    def update_error(self, x, comp, err):
        K.sparse_error(x.reshape(-1), self._last, err.reshape(-1))


class DitheringCompressor(BaseCompressor):
    """Stochastic quantization, linear or natural (power-of-2) partitions
    (reference impl/dithering.cc).

    Wire: ``[norm f32][flag u8][body]`` — flag 0 is dense int8 codes,
    flag 1 is the chunked Elias-delta sparse stream (reference
    utils.h:115-250 BitWriter + delta-coded positions).  Quantization
    runs on the GPU; the bit-level (de)coding is a host-side transform
    between staging and the wire (``encode_wire``/``decode_wire``),
    exactly where the reference ran its whole codec."""

    host_wire = True

    def __init__(self, s: int = 64, natural: bool = False, seed: int = 1):
        self.s = int(s)
        self.levels = self.s
        self.natural = natural
        self.codec = DITHER_NATURAL if natural else DITHER_LINEAR
        self.seed = seed
        self.round = 0
        from ..common.config import env_bool
        self.sparse_wire = env_bool("BPS_DITHER_SPARSE", default=True)

    def compress(self, x: torch.Tensor) -> Compressed:
        seed = (self.seed * 0xD6E8FEB86659FD93 + self.round) & (2**64 - 1)
        self.round += 1
        code, norm_t = K.dithering_compress(x.reshape(-1), self.s, seed,
                                            self.natural)
        flag = torch.zeros(1, dtype=torch.uint8, device=x.device)
        return Compressed([norm_t.view(torch.uint8), flag,
                           code.view(torch.uint8)])

    def encode_wire(self, host_payload: torch.Tensor, n: int,
                    out_buf) -> int:
        """Dense host payload → Elias wire in ``out_buf``.  Returns the
        wire length, or -1 when the dense form is smaller (caller pushes
        the dense payload unchanged)."""
        if not self.sparse_wire or out_buf is None:
            return -1
        from ..ops import core
        wlen = core().cpu_dither_encode(
            host_payload.data_ptr() + 5, n, out_buf.data_ptr() + 5,
            min(out_buf.numel(), host_payload.numel()) - 5)
        if wlen < 0:
            return -1
        out_buf[:4] = host_payload[:4]
        out_buf[4] = 1
        return 5 + int(wlen)

    def decode_wire(self, wire: torch.Tensor, n: int,
                    out=None) -> torch.Tensor:
        """Sparse host wire → dense host payload (no-op for flag 0).
        ``out`` (≥ 5+n bytes, ideally pinned — the pipeline reuses its
        push wire buffer, free once the reply exists) avoids a pageable
        H2D of the dense codes."""
        if int(wire[4]) != 1:
            return wire
        from ..ops import core
        if out is not None and out.numel() >= 5 + n \
                and out.data_ptr() != wire.data_ptr():
            dense = out[:5 + n]
        else:
            dense = torch.empty(5 + n, dtype=torch.uint8)
        dense[:4] = wire[:4]
        dense[4] = 0
        body = wire[5:].contiguous()
        core().cpu_dither_decode(body.data_ptr(), body.numel(), n,
                                 dense.data_ptr() + 5)
        return dense

    def decompress(self, payload, n, aux=0, out=None):
        norm_t = payload[:4].view(torch.float32).contiguous()
        code = payload[5:5 + n].view(torch.int8).contiguous()
        return K.dithering_decompress(code, norm_t, self.s, self.natural, out)


class Fp8Compressor(BaseCompressor):
    """OCP e4m3fn wire with per-partition amax scaling (MI355X-native
    addition: gfx950's hardware fp8 format; 4× compression, ~2^-3
    relative precision, deterministic — no RNG)."""
    codec = FP8

    def compress(self, x: torch.Tensor) -> Compressed:
        code, amax = K.fp8_compress(x.reshape(-1).float())
        return Compressed([amax.view(torch.uint8), code])

    def decompress(self, payload, n, aux=0, out=None):
        amax = payload[:4].view(torch.float32).contiguous()
        code = payload[4:4 + n].contiguous()
        return K.fp8_decompress(code, amax, out)


class NesterovMomentum(BaseCompressor):
    """m = μm + g ; g += μm before compression (reference
    impl/nesterov_momentum.cc:39-49) — fused HIP kernel on GPU."""

    def __init__(self, inner: BaseCompressor, mu: float = 0.9):
        self.inner = inner
        self.host_wire = getattr(inner, "host_wire", False)
        self.mu = mu
        self.codec = inner.codec
        self.levels = inner.levels
        self._mom: Optional[torch.Tensor] = None

    def compress(self, x: torch.Tensor) -> Compressed:
        flat = x.reshape(-1)
        if self._mom is None:
            self._mom = torch.zeros_like(flat, dtype=torch.float32)
        K.nesterov_(flat, self._mom, self.mu)
        return self.inner.compress(x)

    def decompress(self, payload, n, aux=0, out=None):
        return self.inner.decompress(payload, n, aux, out)

    def encode_wire(self, host_payload, n, out_buf):
        return self.inner.encode_wire(host_payload, n, out_buf)

    def decode_wire(self, wire, n, out=None):
        return self.inner.decode_wire(wire, n, out)

    def update_error(self, x, comp, err):
        self.inner.update_error(x, comp, err)


class ErrorFeedback(BaseCompressor):
    """Vanilla error feedback: compensate with the previous round's
    residual, then store the new residual (reference
    impl/vanilla_error_feedback.cc:42-65 — the LR-ratio correction there
    is read from the optimizer here, not an mmap'd file)."""

    def __init__(self, inner: BaseCompressor):
        self.inner = inner
        self.host_wire = getattr(inner, "host_wire", False)
        self.codec = inner.codec
        self.levels = inner.levels
        self._err: Optional[torch.Tensor] = None

    def compress(self, x: torch.Tensor) -> Compressed:
        flat = x.reshape(-1)
        if self._err is None:
            self._err = torch.zeros_like(flat, dtype=torch.float32)
        K.axpy_(flat, self._err.to(flat.dtype), 1.0)
        comp = self.inner.compress(flat)
        self.inner.update_error(flat, comp, self._err)
        return comp

    def decompress(self, payload, n, aux=0, out=None):
        return self.inner.decompress(payload, n, aux, out)

    def encode_wire(self, host_payload, n, out_buf):
        return self.inner.encode_wire(host_payload, n, out_buf)

    def decode_wire(self, wire, n, out=None):
        return self.inner.decode_wire(wire, n, out)


_REGISTRY = {
    "onebit": lambda p: OnebitCompressor(
        scaled=str(p.get("onebit_scaling", "true")).lower() != "false"),
    "topk": lambda p: TopkCompressor(int(p.get("compressor_k", 128))),
    "randomk": lambda p: RandomkCompressor(int(p.get("compressor_k", 128)),
                                           int(p.get("seed", 1))),
    "dithering": lambda p: DitheringCompressor(
        int(p.get("compressor_k", 64)),
        str(p.get("partition", "linear")) == "natural",
        int(p.get("seed", 1))),
    "fp8": lambda p: Fp8Compressor(),
}


def create(params: Dict) -> Optional[BaseCompressor]:
    """Build a (decorated) compressor from reference-style kwargs
    (compressor_registry.cc:39-56 chaining: momentum → ef → compressor)."""
    ctype = params.get("compressor_type")
    if not ctype:
        return None
    comp = _REGISTRY[ctype](params)
    if params.get("ef_type", "").lower() in ("vanilla", "1", "true"):
        comp = ErrorFeedback(comp)
    if params.get("momentum_type", "").lower() == "nesterov":
        comp = NesterovMomentum(comp, float(params.get("momentum_mu", 0.9)))
    return comp
