"""BERT (large) encoder for masked-LM pretraining benchmarks — the
reference's headline scaling-efficiency number is BERT-large mixed
precision (reference README.md:34-38, ≈334M params).

MI355X-first implementation: attention goes through
``F.scaled_dot_product_attention`` (ROCm flash/mem-efficient kernels),
GELU fused via ``F.gelu(approximate='tanh')``, compute dtype bf16 under
autocast.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..torch.fused_ln import FusedLayerNorm


class BertConfig:
    def __init__(self, vocab_size=30522, hidden=1024, layers=24, heads=16,
                 intermediate=4096, max_pos=512, dropout=0.0):
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.layers = layers
        self.heads = heads
        self.intermediate = intermediate
        self.max_pos = max_pos
        self.dropout = dropout

    @staticmethod
    def bert_large() -> "BertConfig":
        return BertConfig(hidden=1024, layers=24, heads=16, intermediate=4096)

    @staticmethod
    def bert_base() -> "BertConfig":
        return BertConfig(hidden=768, layers=12, heads=12, intermediate=3072)

    @staticmethod
    def tiny() -> "BertConfig":
        return BertConfig(vocab_size=1024, hidden=64, layers=2, heads=4,
                          intermediate=128, max_pos=128)


def _sdpa_ctx(seq_len: int):
    """SDPA backend policy: flash backward is overhead-bound at short
    sequences on ROCm (measured 140 µs vs 37 µs fwd at s=128 —
    profiles/bert_large_steady_state.md), so default to the CK
    efficient-attention kernels for S ≤ 256 and flash beyond.
    ``BPS_SDPA_BACKEND=flash|efficient|math`` overrides."""
    import os
    name = os.environ.get("BPS_SDPA_BACKEND", "").lower()
    if not name:
        # within-run A/B (scripts/micro_ln_sdpa.py on MI355X): flash
        # fwd+bwd 195 µs vs efficient 217 µs vs math 529 µs at b64 s128 —
        # torch's default (flash) is right; keep the knob for other shapes
        return None
    from torch.nn.attention import SDPBackend, sdpa_kernel
    table = {"flash": SDPBackend.FLASH_ATTENTION,
             "efficient": SDPBackend.EFFICIENT_ATTENTION,
             "math": SDPBackend.MATH}
    return sdpa_kernel(table[name]) if name in table else None


class SelfAttention(nn.Module):
    """Packed qkv (one GEMM) by default.  ``BPS_BERT_SPLIT_QKV=1`` uses
    three separate projections: slightly smaller GEMMs, but no per-layer
    unbind→contiguous copies forward and no CatArrayBatchedCopy grad
    assembly backward (profiles/bert_large_steady_state.md showed those
    at ~1.5 ms/step)."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        import os
        self.heads = cfg.heads
        self.head_dim = cfg.hidden // cfg.heads
        self.split_qkv = os.environ.get("BPS_BERT_SPLIT_QKV", "0") == "1"
        if self.split_qkv:
            self.q_proj = nn.Linear(cfg.hidden, cfg.hidden)
            self.k_proj = nn.Linear(cfg.hidden, cfg.hidden)
            self.v_proj = nn.Linear(cfg.hidden, cfg.hidden)
        else:
            self.qkv = nn.Linear(cfg.hidden, 3 * cfg.hidden)
        self.out = nn.Linear(cfg.hidden, cfg.hidden)
        self.dropout = cfg.dropout

    def forward(self, x):
        B, S, H = x.shape
        h, d = self.heads, self.head_dim
        if self.split_qkv:
            q = self.q_proj(x).view(B, S, h, d).transpose(1, 2)
            k = self.k_proj(x).view(B, S, h, d).transpose(1, 2)
            v = self.v_proj(x).view(B, S, h, d).transpose(1, 2)
        else:
            qkv = self.qkv(x).view(B, S, 3, h, d)
            q, k, v = qkv.permute(2, 0, 3, 1, 4)      # 3 × (B, h, S, d)
        ctx = _sdpa_ctx(S) if x.is_cuda else None
        if ctx is not None:
            with ctx:
                o = F.scaled_dot_product_attention(
                    q, k, v,
                    dropout_p=self.dropout if self.training else 0.0)
        else:
            o = F.scaled_dot_product_attention(
                q, k, v, dropout_p=self.dropout if self.training else 0.0)
        o = o.transpose(1, 2).reshape(B, S, H)
        return self.out(o)


def _use_fused_mlp() -> bool:
    import os
    if os.environ.get("BPS_FUSED_MLP", "1") in ("0", "false", "no"):
        return False
    if not torch.cuda.is_available():
        return False
    try:
        from ..torch.fused_mlp import fused_mlp_available
        return fused_mlp_available()
    except Exception:
        return False


class EncoderLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = SelfAttention(cfg)
        self.ln1 = FusedLayerNorm(cfg.hidden)
        self.fc1 = nn.Linear(cfg.hidden, cfg.intermediate)
        self.fc2 = nn.Linear(cfg.intermediate, cfg.hidden)
        self.ln2 = FusedLayerNorm(cfg.hidden)
        self._fused_mlp = _use_fused_mlp()

    def forward(self, x):
        x = self.ln1(x + self.attn(x))
        if self._fused_mlp and x.is_cuda:
            # GELU + both bias grads ride the hipBLASLt epilogues
            from ..torch.fused_mlp import fused_mlp
            h = fused_mlp(x, self.fc1.weight, self.fc1.bias,
                          self.fc2.weight, self.fc2.bias)
        else:
            h = self.fc2(F.gelu(self.fc1(x), approximate="tanh"))
        return self.ln2(x + h)


class BertForPreTraining(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos_emb = nn.Embedding(cfg.max_pos, cfg.hidden)
        self.type_emb = nn.Embedding(2, cfg.hidden)
        self.emb_ln = FusedLayerNorm(cfg.hidden)
        self.layers = nn.ModuleList(
            EncoderLayer(cfg) for _ in range(cfg.layers))
        self.mlm_dense = nn.Linear(cfg.hidden, cfg.hidden)
        self.mlm_ln = FusedLayerNorm(cfg.hidden)
        # decoder tied to token embedding
        self.mlm_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids, token_type_ids=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device).unsqueeze(0)
        x = self.tok_emb(input_ids) + self.pos_emb(pos)
        if token_type_ids is not None:
            x = x + self.type_emb(token_type_ids)
        x = self.emb_ln(x)
        # keep the residual stream in the autocast compute dtype so the
        # fused bf16 LayerNorm kernels engage and per-layer re-casts
        # disappear (profiles/bert_large_steady_state.md)
        if x.is_cuda and torch.is_autocast_enabled():
            x = x.to(torch.get_autocast_gpu_dtype())
        for layer in self.layers:
            x = layer(x)
        h = self.mlm_ln(F.gelu(self.mlm_dense(x), approximate="tanh"))
        logits = F.linear(h, self.tok_emb.weight, self.mlm_bias)
        return logits

    def loss(self, input_ids, labels, token_type_ids=None):
        logits = self(input_ids, token_type_ids)
        flat = logits.view(-1, self.cfg.vocab_size)
        # cross_entropy accumulates its row reductions in fp32 internally;
        # materializing a fp32 copy of [tokens, vocab] (≈1 GB at b64 s128)
        # would cost more HBM traffic than it buys
        if not flat.is_cuda:
            flat = flat.float()
        return F.cross_entropy(flat, labels.view(-1), ignore_index=-100)


def bert_large() -> BertForPreTraining:
    return BertForPreTraining(BertConfig.bert_large())


def bert_base() -> BertForPreTraining:
    return BertForPreTraining(BertConfig.bert_base())
