"""Intra-node (plugin-level) wire compression: none / fp16 / bf16.

Parity with reference byteps/torch/compression.py:34-76 (which offered
none and fp16).  bf16 is added because it is the native MI355X training
dtype — casting fp32 gradients to bf16 halves xGMI/PCIe bytes with the
same exponent range.  (The *codec* compressors — onebit/topk/randomk/
dithering — live in :mod:`byteps_amd.compression`.)
"""

import torch


class NoneCompressor:
    @staticmethod
    def compress(tensor):
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor


class FP16Compressor:
    @staticmethod
    def compress(tensor):
        if tensor.dtype in (torch.float32, torch.float64):
            return tensor.to(torch.float16), tensor.dtype
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor.to(ctx) if ctx is not None else tensor


class BF16Compressor:
    @staticmethod
    def compress(tensor):
        if tensor.dtype in (torch.float32, torch.float64):
            return tensor.to(torch.bfloat16), tensor.dtype
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor.to(ctx) if ctx is not None else tensor


class Compression:
    """Namespace matching the reference API (torch/compression.py)."""
    none = NoneCompressor
    fp16 = FP16Compressor
    bf16 = BF16Compressor
