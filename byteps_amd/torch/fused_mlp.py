"""hipBLASLt epilogue-fused MLP block (Linear → tanh-GELU → Linear) for
the BERT encoder.

The eager path pays a standalone GELU-backward kernel plus two bias-grad
reductions per layer (~2 ms/step on BERT-large,
profiles/bert_large_final_kernels.txt).  This hipBLASLt build for gfx950
supports DGELU (NN) and BGRADB (NT) epilogues but no AUX-output forward
(probed — scripts/probe_blaslt.cc), so the FORWARD stays on torch's
TunableOp-tuned addmm + gelu (keeping the pre-GELU activation) and the
BACKWARD goes through hipBLASLt with fused bias-grads:

  dgrad2: dY1 = gelu_backward(dY2·W2, H)   (lt GEMM + aten kernel —
          the DGELU epilogue exists but measured 4x slower than the
          plain GEMM and numerically off on this hipBLASLt build,
          profiles/MEASUREMENTS.md)
  wgrad2: dW2 = dY2ᵀ·Y1  with db2 = Σ dY2  (BGRADB)
  wgrad1: dW1 = dY1ᵀ·X   with db1 = Σ dY1  (BGRADB)
  dgrad1: dX  = dY1·W1

Measured (same-box A/B, BERT-large b64 s128): 198.1k tok/s fused vs
195.1k eager (+1.5%) once the DGELU epilogue was dropped — the fused
bias-grads + mini-autotuned GEMMs are a small net win, so this is the
default (BPS_FUSED_MLP=0 reverts)."""

from __future__ import annotations

import torch
import torch.nn.functional as F

from .. import ops as _ops


def _stream(t: torch.Tensor) -> int:
    return torch.cuda.current_stream(t.device).cuda_stream


class _FusedMLPFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        # x [M, K] bf16 (2-D); w1 [I, K]; w2 [H, I]; biases bf16
        h = F.linear(x, w1, b1)                       # pre-GELU, saved
        y1 = F.gelu(h, approximate="tanh")
        y2 = F.linear(y1, w2, b2)
        ctx.save_for_backward(x, w1, w2, h, y1)
        return y2

    @staticmethod
    def backward(ctx, dy2):
        core = _ops.core()
        x, w1, w2, h, y1 = ctx.saved_tensors
        M, K = x.shape
        I = w1.shape[0]
        H = w2.shape[0]
        dy2 = dy2.contiguous()
        s = _stream(dy2)
        dev = x.device
        # dW2 [H, I] with db2 fused (BGRADB)
        dw2 = torch.empty(H, I, dtype=torch.bfloat16, device=dev)
        db2 = torch.empty(H, dtype=torch.bfloat16, device=dev)
        core.lt_gemm_wgrad(dy2.data_ptr(), y1.data_ptr(), dw2.data_ptr(),
                           db2.data_ptr(), M, H, I, s)
        # dY1 [M, I] = gelu'(H) ⊙ (dY2·W2)
        t = torch.empty(M, I, dtype=torch.bfloat16, device=dev)
        core.lt_gemm_dgrad(dy2.data_ptr(), w2.data_ptr(), t.data_ptr(),
                           M, H, I, s)
        dy1 = torch.ops.aten.gelu_backward(t, h, approximate="tanh")
        dy1 = dy1.contiguous()
        # dW1 [I, K] with db1 fused
        dw1 = torch.empty(I, K, dtype=torch.bfloat16, device=dev)
        db1 = torch.empty(I, dtype=torch.bfloat16, device=dev)
        core.lt_gemm_wgrad(dy1.data_ptr(), x.data_ptr(), dw1.data_ptr(),
                           db1.data_ptr(), M, I, K, s)
        # dX [M, K]
        dx = torch.empty(M, K, dtype=torch.bfloat16, device=dev)
        core.lt_gemm_dgrad(dy1.data_ptr(), w1.data_ptr(), dx.data_ptr(),
                           M, I, K, s)
        return dx, dw1, db1, dw2, db2


def fused_mlp(x: torch.Tensor, w1, b1, w2, b2) -> torch.Tensor:
    """x [..., K] → [..., H].  fp32 weights/biases are cast to bf16 here
    (recorded — grads flow back to the fp32 params exactly as under
    autocast's cast)."""
    shape = x.shape
    x2 = x.reshape(-1, shape[-1])
    if x2.dtype != torch.bfloat16:
        x2 = x2.to(torch.bfloat16)
    if not x2.is_contiguous():
        x2 = x2.contiguous()
    w1b = w1.to(torch.bfloat16)
    w2b = w2.to(torch.bfloat16)
    b1b = b1.to(torch.bfloat16)
    b2b = b2.to(torch.bfloat16)
    y = _FusedMLPFn.apply(x2, w1b, b1b, w2b, b2b)
    return y.reshape(*shape[:-1], y.shape[-1])


def fused_mlp_available() -> bool:
    return _ops.have_core() and torch.cuda.is_available() \
        and hasattr(_ops.core(), "lt_gemm_dgelu")
