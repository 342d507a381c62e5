#!/usr/bin/env python3
"""Same-box A/B of the fused post-collective epilogue kernel
(bps_cast_scale_many) against the torch ops it replaces
(per-bucket copy_ + _foreach_div_) on ResNet-50-shaped buckets.

Run on a GPU box:  python scripts/micro_epilogue.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from byteps_amd import ops as K  # noqa: E402


def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    dev = torch.device("cuda")
    # ResNet-50 grads: ~25.6M fp32 elems in 16MiB buckets → 7 buckets
    sizes = [4194304] * 6 + [448512]
    sizes = [s - s % 64 for s in sizes]
    world = 8

    # case 1: plain averaging (in-place divide)
    bufs = [torch.randn(s, device=dev) for s in sizes]
    desc, total_vec, _ = K.build_cast_scale_desc(bufs, bufs)

    def fused_divide():
        K.cast_scale_many_(desc, len(bufs), total_vec, 1.0 / world,
                           torch.float32, torch.float32, dev)

    def torch_divide():
        torch._foreach_div_(bufs, float(world))

    t_f = bench(fused_divide)
    t_t = bench(torch_divide)
    nbytes = sum(s * 4 * 2 for s in sizes)  # read + write
    print("divide-only   : fused %7.1f us (%.2f TB/s)  _foreach_div_ %7.1f us (%.2f TB/s)"
          % (t_f, nbytes / t_f / 1e6, t_t, nbytes / t_t / 1e6))

    # case 2: bf16 wire cast-back + averaging (fused) vs copy_ + div_
    scratch = [torch.randn(s, device=dev).bfloat16() for s in sizes]
    desc2, total_vec2, _ = K.build_cast_scale_desc(bufs, scratch)

    def fused_wire():
        K.cast_scale_many_(desc2, len(bufs), total_vec2, 1.0 / world,
                           torch.bfloat16, torch.float32, dev)

    def torch_wire():
        for b, s in zip(bufs, scratch):
            b.copy_(s)
        torch._foreach_div_(bufs, float(world))

    t_f2 = bench(fused_wire)
    t_t2 = bench(torch_wire)
    nbytes2 = sum(s * (2 + 4) for s in sizes)
    print("bf16 cast+avg : fused %7.1f us (%.2f TB/s)  copy_+div_    %7.1f us"
          % (t_f2, nbytes2 / t_f2 / 1e6, t_t2))


if __name__ == "__main__":
    main()
