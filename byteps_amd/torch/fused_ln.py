"""Fused LayerNorm for bf16 on gfx950 (ops/csrc/ln.hip).

torch's layer_norm under autocast upcasts activations to fp32 — the
BERT-large profile showed the fp32 LN kernels plus bf16↔fp32 casts
around every call at ~8% of the step.  This module keeps bf16 storage
with fp32 row statistics in one pass each way.  Falls back to
``F.layer_norm`` on CPU or unsupported hidden sizes.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops as K


def _stream(t: torch.Tensor) -> int:
    return torch.cuda.current_stream(t.device).cuda_stream


class _FusedLNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        C = x.shape[-1]
        M = x.numel() // C
        core = K.core()
        x = x.contiguous()
        y = torch.empty_like(x)
        mean = torch.empty(M, dtype=torch.float32, device=x.device)
        invstd = torch.empty(M, dtype=torch.float32, device=x.device)
        wf = weight.float().contiguous()
        bf = bias.float().contiguous()
        core.ln_fwd(x.data_ptr(), wf.data_ptr(), bf.data_ptr(), y.data_ptr(),
                    M, C, eps, mean.data_ptr(), invstd.data_ptr(),
                    _stream(x))
        ctx.save_for_backward(x, mean, invstd, wf)
        ctx.dims = (M, C)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mean, invstd, wf = ctx.saved_tensors
        M, C = ctx.dims
        core = K.core()
        dy = dy.contiguous()
        dx = torch.empty_like(x)
        nb = core.LN_RED_BLOCKS
        partial = torch.empty(nb * 2 * C, dtype=torch.float32,
                              device=x.device)
        core.ln_bwd(x.data_ptr(), dy.data_ptr(), wf.data_ptr(),
                    mean.data_ptr(), invstd.data_ptr(), dx.data_ptr(), M, C,
                    partial.data_ptr(), _stream(x))
        sums2 = torch.empty(2 * C, dtype=torch.float32, device=x.device)
        core.ln_fold(partial.data_ptr(), M, C, sums2.data_ptr(), _stream(x))
        dbeta = sums2[:C]
        dgamma = sums2[C:]
        return dx, dgamma, dbeta, None


class FusedLayerNorm(nn.Module):
    """Drop-in ``nn.LayerNorm`` over the last dimension."""

    def __init__(self, hidden: int, eps: float = 1e-12):
        super().__init__()
        self.hidden = hidden
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden))
        self.bias = nn.Parameter(torch.zeros(hidden))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if (x.is_cuda and x.dtype == torch.bfloat16
                and K.have_core() and K.core().ln_supported(self.hidden)):
            return _FusedLNFunction.apply(x, self.weight, self.bias,
                                          self.eps)
        if x.dtype != self.weight.dtype:
            return F.layer_norm(x.float(), (self.hidden,), self.weight,
                                self.bias, self.eps).to(x.dtype)
        return F.layer_norm(x, (self.hidden,), self.weight, self.bias,
                            self.eps)
