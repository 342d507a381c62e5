"""Fused BatchNorm(+residual)(+ReLU) for NHWC bf16 on gfx950.

Profiling ResNet-50 bf16 on MI355X showed MIOpen's spatial batchnorm
(3 kernels fwd + 3 bwd, unfused ReLU and residual adds around them) at
~30% of the step (profiles/resnet50_steady_state.md).  This module fuses
normalize+affine+residual+ReLU into one memory pass each way using the
hand-written CDNA4 kernels in ops/csrc/bn.hip.

Falls back to the native torch ops on CPU, for unsupported channel
counts (C % 8 != 0 or C > 2048), or for non-bf16/non-channels-last
inputs — and raises if the HIP extension is missing on a GPU machine
(ops.core() enforces that).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops as K


def _stream(t: torch.Tensor) -> int:
    return torch.cuda.current_stream(t.device).cuda_stream


def _nhwc_ok(x: torch.Tensor, C: int) -> bool:
    return (x.is_cuda and x.dtype == torch.bfloat16 and C % 8 == 0
            and C <= 2048 and x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last))


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                momentum, eps, relu, training):
        N, C, H, W = x.shape
        M = N * H * W
        core = K.core()
        dev = x.device
        s = _stream(x)
        y = torch.empty_like(x)
        if training:
            nb = core.BN_RED_BLOCKS
            partial = torch.empty(nb * 2 * C, dtype=torch.float32,
                                  device=dev)
            core.bn_reduce(x.data_ptr(), M, C, partial.data_ptr(), s)
            mean = torch.empty(C, dtype=torch.float32, device=dev)
            invstd = torch.empty(C, dtype=torch.float32, device=dev)
            core.bn_finalize(partial.data_ptr(), M, C, eps, momentum,
                             mean.data_ptr(), invstd.data_ptr(),
                             running_mean.data_ptr(), running_var.data_ptr(),
                             1, s)
        else:
            mean = running_mean.float()
            invstd = torch.rsqrt(running_var.float() + eps)
        wf = weight.float().contiguous()
        bf = bias.float().contiguous()
        # relu path emits a 1-bit activation mask (1 byte / 8 channels) so
        # backward never re-reads y
        mask = torch.empty(M * (C // 8), dtype=torch.uint8, device=dev) \
            if relu else None
        core.bn_fwd_apply(x.data_ptr(),
                          residual.data_ptr() if residual is not None else 0,
                          y.data_ptr(), M, C, mean.data_ptr(),
                          invstd.data_ptr(), wf.data_ptr(), bf.data_ptr(),
                          int(relu), mask.data_ptr() if relu else 0, s)
        ctx.save_for_backward(x, mean, invstd, wf,
                              mask if mask is not None else x.new_empty(0))
        ctx.relu = relu
        ctx.has_res = residual is not None
        ctx.dims = (M, C)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mean, invstd, wf, mask = ctx.saved_tensors
        M, C = ctx.dims
        core = K.core()
        dy = dy.contiguous(memory_format=torch.channels_last)
        s = _stream(x)
        nb = core.BN_RED_BLOCKS
        mask_ptr = mask.data_ptr() if ctx.relu else 0
        partial = torch.empty(nb * 2 * C, dtype=torch.float32,
                              device=x.device)
        core.bn_bwd_reduce(x.data_ptr(), dy.data_ptr(), mask_ptr, M, C,
                           mean.data_ptr(), invstd.data_ptr(),
                           partial.data_ptr(), int(ctx.relu), s)
        sums2 = torch.empty(2 * C, dtype=torch.float32, device=x.device)
        core.bn_fold(partial.data_ptr(), M, C, sums2.data_ptr(), s)
        dx = torch.empty_like(x)
        dres = torch.empty_like(x) if ctx.has_res else None
        core.bn_bwd_apply(x.data_ptr(), dy.data_ptr(), mask_ptr,
                          dx.data_ptr(),
                          dres.data_ptr() if dres is not None else 0, M, C,
                          mean.data_ptr(), invstd.data_ptr(), wf.data_ptr(),
                          sums2.data_ptr(), int(ctx.relu), s)
        dbeta = sums2[:C]
        dgamma = sums2[C:]
        return (dx, dres, dgamma, dbeta, None, None, None, None, None, None)


class FusedBNReLU(nn.Module):
    """Drop-in BatchNorm2d with optional fused residual add and ReLU.

    ``forward(x, residual=None)`` computes
    ``relu(bn(x) + residual)`` (relu/residual per constructor flags) in a
    single kernel pass on gfx950.
    """

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, relu: bool = True):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        # kept for state_dict parity with nn.BatchNorm2d but NOT bumped
        # per step on the GPU path — with fixed momentum it is unused,
        # and the per-call long-add kernel showed up in profiles
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        if _nhwc_ok(x, self.num_features) and (
                residual is None or
                residual.is_contiguous(memory_format=torch.channels_last)):
            res = residual.to(torch.bfloat16) if residual is not None else None
            return _FusedBNFunction.apply(
                x, res, self.weight, self.bias, self.running_mean,
                self.running_var, self.momentum, self.eps, self.relu,
                self.training)
        # reference fallback (CPU tests / unsupported shapes)
        out = F.batch_norm(x, self.running_mean, self.running_var,
                           self.weight, self.bias, self.training,
                           self.momentum, self.eps)
        if residual is not None:
            out = out + residual
        return F.relu(out, inplace=True) if self.relu else out
