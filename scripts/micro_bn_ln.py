"""Microbench the fused BN / LN kernels at ResNet-50 / BERT-large shapes.

Prints per-kernel-group wall time (hipEvent) for bandwidth math; run
under `rocprofv3 --pmc FETCH_SIZE,SQ_LDS_BANK_CONFLICT,SQ_WAVES` for the
HBM-read bytes and LDS behavior (PMC runs must not add trace flags —
gpurun refuses the combination)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from byteps_amd import ops as K  # noqa: E402

core = K.core()
ITERS = 50


def timed(label, fn, traffic_bytes):
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(ITERS):
        fn()
    e.record()
    torch.cuda.synchronize()
    ms = s.elapsed_time(e) / ITERS
    print("%-28s %8.1f us   %6.2f TB/s (of %.0f MB/iter)"
          % (label, ms * 1e3, traffic_bytes / ms / 1e9,
             traffic_bytes / 1e6), flush=True)


def bench_bn(N, C, H, W):
    M = N * H * W
    x = torch.randn(N * H * W * C, device="cuda").to(torch.bfloat16)
    y = torch.empty_like(x)
    dy = torch.randn_like(x)
    dx = torch.empty_like(x)
    mask = torch.empty(M * C // 8, dtype=torch.uint8, device="cuda")
    nb = core.BN_RED_BLOCKS
    partial = torch.empty(nb * 2 * C, device="cuda")
    mean = torch.empty(C, device="cuda")
    invstd = torch.empty(C, device="cuda")
    rm = torch.zeros(C, device="cuda")
    rv = torch.ones(C, device="cuda")
    g = torch.ones(C, device="cuda")
    b = torch.zeros(C, device="cuda")
    sums2 = torch.empty(2 * C, device="cuda")
    st = torch.cuda.current_stream().cuda_stream
    nbytes = M * C * 2
    tag = "bn[%d,%d,%d,%d]" % (N, C, H, W)

    timed(tag + " fwd_reduce",
          lambda: core.bn_reduce(x.data_ptr(), M, C, partial.data_ptr(), st),
          nbytes)
    core.bn_finalize(partial.data_ptr(), M, C, 1e-5, 0.1, mean.data_ptr(),
                     invstd.data_ptr(), rm.data_ptr(), rv.data_ptr(), 1, st)
    timed(tag + " fwd_apply(+res,relu)",
          lambda: core.bn_fwd_apply(x.data_ptr(), x.data_ptr(), y.data_ptr(),
                                    M, C, mean.data_ptr(), invstd.data_ptr(),
                                    g.data_ptr(), b.data_ptr(), 1,
                                    mask.data_ptr(), st),
          nbytes * 3)
    timed(tag + " bwd_reduce",
          lambda: core.bn_bwd_reduce(x.data_ptr(), dy.data_ptr(),
                                     mask.data_ptr(), M, C, mean.data_ptr(),
                                     invstd.data_ptr(), partial.data_ptr(),
                                     1, st),
          nbytes * 2)
    core.bn_fold(partial.data_ptr(), M, C, sums2.data_ptr(), st)
    timed(tag + " bwd_apply(+dres,relu)",
          lambda: core.bn_bwd_apply(x.data_ptr(), dy.data_ptr(),
                                    mask.data_ptr(), dx.data_ptr(),
                                    y.data_ptr(), M, C, mean.data_ptr(),
                                    invstd.data_ptr(), g.data_ptr(),
                                    sums2.data_ptr(), 1, st),
          nbytes * 4)


def bench_ln(M, C):
    x = torch.randn(M * C, device="cuda").to(torch.bfloat16)
    y = torch.empty_like(x)
    dy = torch.randn_like(x)
    dx = torch.empty_like(x)
    g = torch.ones(C, device="cuda")
    b = torch.zeros(C, device="cuda")
    mean = torch.empty(M, device="cuda")
    invstd = torch.empty(M, device="cuda")
    nb = core.LN_RED_BLOCKS
    partial = torch.empty(nb * 2 * C, device="cuda")
    st = torch.cuda.current_stream().cuda_stream
    nbytes = M * C * 2
    tag = "ln[%d,%d]" % (M, C)
    timed(tag + " fwd",
          lambda: core.ln_fwd(x.data_ptr(), g.data_ptr(), b.data_ptr(),
                              y.data_ptr(), M, C, 1e-12, mean.data_ptr(),
                              invstd.data_ptr(), st),
          nbytes * 2)
    timed(tag + " bwd",
          lambda: core.ln_bwd(x.data_ptr(), dy.data_ptr(), g.data_ptr(),
                              mean.data_ptr(), invstd.data_ptr(),
                              dx.data_ptr(), M, C, partial.data_ptr(), st),
          nbytes * 3)


if __name__ == "__main__":
    bench_bn(64, 64, 112, 112)     # ResNet-50 stem
    bench_bn(64, 256, 56, 56)      # layer1 width
    bench_bn(64, 2048, 7, 7)       # layer4 width
    bench_ln(64 * 128, 1024)       # BERT-large b64 s128
