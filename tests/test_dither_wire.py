"""Elias-delta sparse wire for dithering (reference utils.h:115-250 +
impl/dithering.cc:51-121): golden NumPy codec, wire-size wins, natural
s-level sparsity semantics, and end-to-end PS parity over the sparse
wire."""

import numpy as np
import pytest
import torch

from byteps_amd.ops import core


# -- NumPy golden model of the chunked Elias stream --------------------------

CHUNK = 1 << 16


class _BitReader:
    def __init__(self, data: bytes):
        self.data = data
        self.pos = 0

    def bit(self) -> int:
        b = (self.data[self.pos >> 3] >> (7 - (self.pos & 7))) & 1
        self.pos += 1
        return b

    def bits(self, n: int) -> int:
        v = 0
        for _ in range(n):
            v = (v << 1) | self.bit()
        return v

    def elias_delta(self) -> int:
        lb = 0
        while self.bit() == 0:
            lb += 1
        N = (1 << lb) | self.bits(lb)
        nb = N - 1
        x = 1 << nb
        if nb > 0:
            x |= self.bits(nb)
        return x


def numpy_decode(wire: bytes, n: int) -> np.ndarray:
    ce = int.from_bytes(wire[0:4], "little")
    nc = int.from_bytes(wire[4:8], "little")
    sizes = [int.from_bytes(wire[8 + 4 * c:12 + 4 * c], "little")
             for c in range(nc)]
    out = np.zeros(n, dtype=np.int8)
    off = 8 + 4 * nc
    for c in range(nc):
        lo = c * ce
        ln = min(ce, n - lo)
        r = _BitReader(wire[off:off + sizes[c]])
        i = -1
        while True:
            gap = r.elias_delta()
            i += gap
            if i >= ln:
                break
            neg = r.bit()
            mag = r.elias_delta()
            out[lo + i] = -mag if neg else mag
        off += sizes[c]
    return out


def _encode(codes: torch.Tensor) -> bytes:
    n = codes.numel()
    out = torch.zeros(2 * n + 1024, dtype=torch.uint8)
    wlen = core().cpu_dither_encode(codes.data_ptr(), n, out.data_ptr(),
                                    out.numel())
    assert wlen > 0
    return bytes(out[:wlen].numpy().tobytes())


@pytest.mark.parametrize("n,density", [(1000, 0.05), (200000, 0.01),
                                       (70000, 0.5), (64, 1.0), (100, 0.0)])
def test_elias_roundtrip_golden(n, density):
    torch.manual_seed(n)
    codes = torch.zeros(n, dtype=torch.int8)
    nnz = int(n * density)
    if nnz:
        idx = torch.randperm(n)[:nnz]
        vals = torch.randint(1, 127, (nnz,), dtype=torch.int8)
        sign = torch.randint(0, 2, (nnz,)) * 2 - 1
        codes[idx] = (vals * sign.to(torch.int8)).to(torch.int8)
    wire = _encode(codes)
    # C++ decode
    dec = torch.empty(n, dtype=torch.int8)
    wire_t = torch.frombuffer(bytearray(wire), dtype=torch.uint8)
    core().cpu_dither_decode(wire_t.data_ptr(), len(wire), n,
                             dec.data_ptr())
    assert torch.equal(dec, codes)
    # independent NumPy golden decode of the same bytes
    ref = numpy_decode(wire, n)
    assert np.array_equal(ref, codes.numpy())


def test_sparse_wire_smaller_than_dense():
    """At 1% density the Elias stream must be far below the n-byte dense
    wire (the reference's whole motivation, impl/dithering.cc coding)."""
    n = 1 << 20
    torch.manual_seed(3)
    codes = torch.zeros(n, dtype=torch.int8)
    idx = torch.randperm(n)[:n // 100]
    codes[idx] = 3
    wire = _encode(codes)
    assert len(wire) < n // 4, "sparse wire not compact: %d" % len(wire)


def test_natural_dithering_sparsity_and_unbiasedness():
    """s-level natural partitions: values below 2^(1-s)·norm mostly round
    to zero (sparsity), and the quantizer stays unbiased in expectation."""
    from byteps_amd import ops as K
    torch.manual_seed(0)
    n, s = 1 << 16, 4
    x = torch.randn(n)
    code, norm_t = K.dithering_compress(x, s, seed=123, natural=True)
    frac_zero = float((code == 0).float().mean())
    assert frac_zero > 0.5, "expected sparsity at s=4, got %.2f" % frac_zero
    assert int(code.abs().max()) <= s
    # unbiasedness: per-element variance at s=4 is large, but the
    # AGGREGATE bias over n·trials draws must vanish — test that the
    # mean signed error and the mean |decoded| match x statistically
    acc = torch.zeros(n)
    trials = 40
    for t in range(trials):
        c, _ = K.dithering_compress(x, s, seed=1000 + t, natural=True)
        acc += K.dithering_decompress(c, norm_t, s, natural=True)
    mean = acc / trials
    agg_bias = float((mean - x).mean().abs()) / float(x.abs().mean())
    assert agg_bias < 0.05, "aggregate bias %.3f" % agg_bias
    mag_ratio = float(mean.abs().sum()) / float(x.abs().sum())
    assert 0.5 < mag_ratio < 2.0, "magnitude off: %.3f" % mag_ratio


def _dither_ps_worker(rank, world, sparse):
    import os
    os.environ["BPS_DITHER_SPARSE"] = "1" if sparse else "0"
    import byteps_amd.torch as bps
    bps.init()
    torch.manual_seed(0)
    m = torch.nn.Sequential(torch.nn.Linear(64, 256),
                            torch.nn.Linear(256, 16))
    opt = bps.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05),
        named_parameters=m.named_parameters(),
        compression_params={"compressor_type": "dithering",
                            "partition": "natural",
                            "ef_type": "vanilla",
                            "compressor_k": 4})
    torch.manual_seed(42)
    x = torch.randn(8, 64)
    y = torch.randn(8, 16)
    for _ in range(3):
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    out = [p.detach().clone() for p in m.parameters()]
    bps.shutdown()
    return out


def test_ps_dithering_sparse_wire_end_to_end():
    """PS round-trip with the Elias wire (natural, s=4 → sparse) must be
    numerically identical to the dense wire."""
    from mp_util import run_in_processes
    from byteps_amd.ops import _core
    results = {}
    for sparse in (True, False):
        srv = _core.Server(0, 2, False)
        srv.start()
        try:
            env = {"BPS_FORCE_DISTRIBUTED": "1",
                   "BPS_SERVER_URIS": "127.0.0.1:%d" % srv.port,
                   "BPS_NUM_SERVER": "1",
                   "BPS_MIN_COMPRESS_BYTES": "0",
                   # force the TCP lane so the Elias wire actually
                   # engages (colocated shm skips bit-level coding)
                   "BPS_ENABLE_IPC": "0"}
            results[sparse] = run_in_processes(_dither_ps_worker, 1,
                                               sparse, extra_env=env)[0]
        finally:
            srv.stop()
    for a, b in zip(results[True], results[False]):
        assert torch.equal(a, b), "sparse wire changed numerics"
