"""hipBLASLt epilogue-fused MLP block (Linear → tanh-GELU → Linear) for
the BERT encoder.

The eager path pays separate GELU fwd/bwd kernels plus standalone
bias-grad reductions every layer (~2-3 ms/step total on BERT-large,
profiles/bert_large_final_kernels.txt).  Here they ride the GEMM
epilogues (blaslt.cc):

  fwd : Y1 = GELU(X·W1ᵀ+b1) storing the pre-GELU aux;  Y2 = Y1·W2ᵀ+b2
  bwd : dY1 = dGELU(aux)⊙(dY2·W2) with db1 fused;  dW2 with db2 fused;
        dW1, dX plain hipBLASLt GEMMs.

Numerics: identical operator set to
``fc2(F.gelu(fc1(x), approximate='tanh'))`` under bf16 autocast — the
epilogue GELU is the same tanh approximation; grads accumulate in the
GEMM's fp32 compute, bias grads in fp32.
"""

from __future__ import annotations

import torch

from .. import ops as _ops


def _stream(t: torch.Tensor) -> int:
    return torch.cuda.current_stream(t.device).cuda_stream


class _FusedMLPFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        # x [M, K] bf16 (2-D, caller flattens); w1 [I, K]; w2 [H, I]
        core = _ops.core()
        M, K = x.shape
        I = w1.shape[0]
        H = w2.shape[0]
        y1 = torch.empty(M, I, dtype=torch.bfloat16, device=x.device)
        aux = torch.empty(M, I, dtype=torch.bfloat16, device=x.device)
        core.lt_gemm_gelu_aux(x.data_ptr(), w1.data_ptr(), b1.data_ptr(),
                              y1.data_ptr(), aux.data_ptr(), M, I, K,
                              _stream(x))
        y2 = torch.empty(M, H, dtype=torch.bfloat16, device=x.device)
        core.lt_gemm_bias(y1.data_ptr(), w2.data_ptr(), b2.data_ptr(),
                          y2.data_ptr(), M, H, I, _stream(x))
        ctx.save_for_backward(x, w1, w2, y1, aux)
        return y2

    @staticmethod
    def backward(ctx, dy2):
        core = _ops.core()
        x, w1, w2, y1, aux = ctx.saved_tensors
        M, K = x.shape
        I = w1.shape[0]
        H = w2.shape[0]
        dy2 = dy2.contiguous()
        s = _stream(dy2)
        # dW2 [H, I] + db2 fused
        dw2 = torch.empty(H, I, dtype=torch.bfloat16, device=x.device)
        db2 = torch.empty(H, dtype=torch.float32, device=x.device)
        core.lt_gemm_wgrad(dy2.data_ptr(), y1.data_ptr(), dw2.data_ptr(),
                           db2.data_ptr(), M, H, I, s)
        # dY1 [M, I] = dGELU(aux) ⊙ (dY2·W2), db1 fused
        dy1 = torch.empty(M, I, dtype=torch.bfloat16, device=x.device)
        db1 = torch.empty(I, dtype=torch.float32, device=x.device)
        core.lt_gemm_dgelu_bgrad(dy2.data_ptr(), w2.data_ptr(),
                                 aux.data_ptr(), dy1.data_ptr(),
                                 db1.data_ptr(), M, H, I, s)
        # dW1 [I, K]
        dw1 = torch.empty(I, K, dtype=torch.bfloat16, device=x.device)
        core.lt_gemm_wgrad(dy1.data_ptr(), x.data_ptr(), dw1.data_ptr(), 0,
                           M, I, K, s)
        # dX [M, K]
        dx = torch.empty(M, K, dtype=torch.bfloat16, device=x.device)
        core.lt_gemm_dgrad(dy1.data_ptr(), w1.data_ptr(), dx.data_ptr(),
                           M, I, K, s)
        return dx, dw1, db1, dw2, db2


def fused_mlp(x: torch.Tensor, w1, b1, w2, b2) -> torch.Tensor:
    """x [..., K] bf16 → [..., H].  Weights/bias may be fp32 (cast here,
    recorded — grads flow back to the fp32 params exactly as under
    autocast's cast)."""
    shape = x.shape
    x2 = x.reshape(-1, shape[-1])
    if x2.dtype != torch.bfloat16:
        x2 = x2.to(torch.bfloat16)
    if not x2.is_contiguous():
        x2 = x2.contiguous()
    w1b = w1.to(torch.bfloat16)
    w2b = w2.to(torch.bfloat16)
    b1f = b1.float()
    b2f = b2.float()
    y = _FusedMLPFn.apply(x2, w1b, b1f, w2b, b2f)
    return y.reshape(*shape[:-1], y.shape[-1])


def fused_mlp_available() -> bool:
    return _ops.have_core() and torch.cuda.is_available() \
        and hasattr(_ops.core(), "lt_gemm_gelu_aux")
