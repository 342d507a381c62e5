"""Microbench on MI355X: fused LN vs torch LN (autocast), and SDPA
backend comparison at BERT-large shapes (b64 s128 h16 d64)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from byteps_amd.torch.fused_ln import FusedLayerNorm


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    M, C = 64 * 128, 1024
    x16 = torch.randn(M, C, device="cuda", dtype=torch.bfloat16,
                      requires_grad=True)
    x32 = x16.detach().float().requires_grad_(True)
    fused = FusedLayerNorm(C).cuda()
    ln = torch.nn.LayerNorm(C).cuda()
    g16 = torch.randn(M, C, device="cuda", dtype=torch.bfloat16)

    def fused_fb():
        y = fused(x16)
        y.backward(g16)
        x16.grad = None

    def torch_fb():
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = ln(x16)
        y.backward(g16.float())
        x16.grad = None

    print("LN fused fwd+bwd: %.1f us" % timeit(fused_fb))
    print("LN torch fwd+bwd: %.1f us" % timeit(torch_fb))

    # kernel-level (no autograd engine): forward only under no_grad
    from byteps_amd import ops as K
    core = K.core()
    xk = x16.detach()
    yk = torch.empty_like(xk)
    mean = torch.empty(M, device="cuda")
    invstd = torch.empty(M, device="cuda")
    wf = fused.weight.detach().float().contiguous()
    bf = fused.bias.detach().float().contiguous()
    s = torch.cuda.current_stream().cuda_stream

    def fused_fwd_kernel():
        core.ln_fwd(xk.data_ptr(), wf.data_ptr(), bf.data_ptr(),
                    yk.data_ptr(), M, C, 1e-12, mean.data_ptr(),
                    invstd.data_ptr(), s)

    def torch_fwd_kernel():
        with torch.no_grad():
            F.layer_norm(xk.float(), (C,), wf, bf, 1e-12)

    def torch_fwd_bf16():
        with torch.no_grad():
            F.layer_norm(xk, (C,), wf.to(torch.bfloat16),
                         bf.to(torch.bfloat16), 1e-12)

    print("LN fused fwd kernel:      %.1f us" % timeit(fused_fwd_kernel, 200))
    print("LN torch fwd fp32(+cast): %.1f us" % timeit(torch_fwd_kernel, 200))
    print("LN torch fwd bf16:        %.1f us" % timeit(torch_fwd_bf16, 200))

    # SDPA backends
    B, h, S, d = 64, 16, 128, 64
    from torch.nn.attention import SDPBackend, sdpa_kernel
    q = torch.randn(B, h, S, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    go = torch.randn_like(q)
    for name, be in [("flash", SDPBackend.FLASH_ATTENTION),
                     ("efficient", SDPBackend.EFFICIENT_ATTENTION),
                     ("math", SDPBackend.MATH)]:
        def fb():
            with sdpa_kernel(be):
                o = F.scaled_dot_product_attention(q, k, v)
            o.backward(go)
            q.grad = k.grad = v.grad = None
        try:
            print("SDPA %-9s fwd+bwd: %.1f us" % (name, timeit(fb)))
        except Exception as e:
            print("SDPA %-9s failed: %s" % (name, type(e).__name__))


if __name__ == "__main__":
    main()
