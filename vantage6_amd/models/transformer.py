"""Transformer models for configs 3 and 4.

- ``bert_base``: 12-layer post-LN encoder (hidden 768, 12 heads, MLM-style
  token-classification head) — the DP-SGD config.
- ``llama_1b``: Llama-3.2-1B-shaped decoder (16 layers, hidden 2048, GQA
  32/8 heads, SwiGLU 8192, RMSNorm, RoPE) — the secure-aggregation config.

All Linear layers run on the hand-written MFMA GEMM (ops.gemm.Linear);
norms run on the hand-written LayerNorm/RMSNorm HIP kernels; attention uses
torch SDPA (library path). bf16 throughout.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.activations import rope, rope_tables, swiglu
import os as _os

from ..ops.attention import sdpa as hip_sdpa

_DENSE_QKV = _os.environ.get("V6AMD_BERT_DENSE_QKV", "0") == "1"
from ..ops.embedding import Embedding as GraphSafeEmbedding
from ..ops.gemm import Linear as MfmaLinear
from ..ops.norms import LayerNorm, RMSNorm


# ---------------------------------------------------------------- BERT-base
class BertLayer(nn.Module):
    def __init__(self, hidden: int, heads: int, ffn: int):
        super().__init__()
        self.heads = heads
        self.qkv = MfmaLinear(hidden, 3 * hidden)
        self.proj = MfmaLinear(hidden, hidden)
        self.ln1 = LayerNorm(hidden)
        self.fc1 = MfmaLinear(hidden, ffn)
        self.fc2 = MfmaLinear(ffn, hidden)
        self.ln2 = LayerNorm(hidden)

    def forward(self, x):
        B, S, H = x.shape
        qkv = self.qkv(x).reshape(B, S, 3, self.heads, H // self.heads)
        if _DENSE_QKV:
            # one copy makes q/k/v dense so the hand-written attention's
            # strided writes stay compact (A/B experiment, see NOTES)
            q, k, v = qkv.permute(2, 0, 3, 1, 4).contiguous().unbind(0)
        else:
            q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        att = hip_sdpa(q, k, v)
        att = att.transpose(1, 2).reshape(B, S, H)
        x = self.ln1(x + self.proj(att))
        x = self.ln2(x + self.fc2(F.gelu(self.fc1(x))))
        return x


class BertBase(nn.Module):
    def __init__(self, vocab: int = 30522, hidden: int = 768, layers: int = 12,
                 heads: int = 12, ffn: int = 3072, max_seq: int = 512,
                 vocab_pad: int | None = None):
        """``vocab_pad`` (megatron-style) pads ONLY the LM head's output
        to a %8 width (e.g. 30720) so the fused bf16 cross-entropy kernel
        can engage; pad logits are masked to -inf in the loss, so the
        distribution over real classes is mathematically identical
        (exp(-inf)=0). Default off — a round-3 measurement lead."""
        super().__init__()
        self.vocab_logical = vocab
        self.tok = GraphSafeEmbedding(vocab, hidden)
        self.pos = GraphSafeEmbedding(max_seq, hidden)
        self.ln = LayerNorm(hidden)
        self.blocks = nn.ModuleList(BertLayer(hidden, heads, ffn) for _ in range(layers))
        self.head = MfmaLinear(hidden, vocab_pad or vocab, bias=False)

    def forward(self, input_ids):
        B, S = input_ids.shape
        # batch-shaped position ids so per-sample grads stay separable (DP)
        pos = torch.arange(S, device=input_ids.device)[None].expand(B, S)
        x = self.ln(self.tok(input_ids) + self.pos(pos))
        for blk in self.blocks:
            x = blk(x)
        return self.head(x)


# ----------------------------------------------------------------- Llama-1B
_ROPE_CACHE: dict = {}


def _rope(q, k, base: float = 500000.0):
    """Rotary embeddings via the hand-written HIP kernel; cos/sin tables
    cached per (seq, head_dim, device)."""
    B, Hn, S, D = q.shape
    key = (S, D, q.device)
    if key not in _ROPE_CACHE:
        _ROPE_CACHE[key] = rope_tables(S, D, base, q.device)
    coss, sins = _ROPE_CACHE[key]
    return rope(q, coss, sins), rope(k, coss, sins)


class LlamaLayer(nn.Module):
    def __init__(self, hidden: int, heads: int, kv_heads: int, ffn: int):
        super().__init__()
        self.heads, self.kv_heads = heads, kv_heads
        self.head_dim = hidden // heads
        self.q = MfmaLinear(hidden, hidden, bias=False)
        self.k = MfmaLinear(hidden, kv_heads * self.head_dim, bias=False)
        self.v = MfmaLinear(hidden, kv_heads * self.head_dim, bias=False)
        self.o = MfmaLinear(hidden, hidden, bias=False)
        self.gate = MfmaLinear(hidden, ffn, bias=False)
        self.up = MfmaLinear(hidden, ffn, bias=False)
        self.down = MfmaLinear(ffn, hidden, bias=False)
        self.norm1 = RMSNorm(hidden)
        self.norm2 = RMSNorm(hidden)

    def forward(self, x):
        B, S, H = x.shape
        h = self.norm1(x)
        q = self.q(h).reshape(B, S, self.heads, self.head_dim).transpose(1, 2)
        k = self.k(h).reshape(B, S, self.kv_heads, self.head_dim).transpose(1, 2)
        v = self.v(h).reshape(B, S, self.kv_heads, self.head_dim).transpose(1, 2)
        q, k = _rope(q, k)
        att = hip_sdpa(q, k, v, causal=True)
        att = att.transpose(1, 2).reshape(B, S, H)
        x = x + self.o(att)
        h = self.norm2(x)
        x = x + self.down(swiglu(self.gate(h), self.up(h)))
        return x


class Llama(nn.Module):
    def __init__(self, vocab: int = 128256, hidden: int = 2048, layers: int = 16,
                 heads: int = 32, kv_heads: int = 8, ffn: int = 8192):
        super().__init__()
        self.tok = GraphSafeEmbedding(vocab, hidden)
        self.blocks = nn.ModuleList(
            LlamaLayer(hidden, heads, kv_heads, ffn) for _ in range(layers)
        )
        self.norm = RMSNorm(hidden)
        self.head = MfmaLinear(hidden, vocab, bias=False)

    def forward(self, input_ids):
        x = self.tok(input_ids)
        for blk in self.blocks:
            x = blk(x)
        return self.head(self.norm(x))


class ViT(nn.Module):
    """Vision Transformer (ViT-B/16 default) on the same MFMA stack as
    BERT: patch embedding is an MfmaLinear over unfolded patches (no
    large-kernel conv needed), then pre-LN-free BertLayer encoder blocks,
    class token + learned positions, MFMA classification head."""

    def __init__(self, image_size: int = 224, patch: int = 16,
                 hidden: int = 768, layers: int = 12, heads: int = 12,
                 ffn: int = 3072, num_classes: int = 1000):
        super().__init__()
        assert image_size % patch == 0
        self.patch = patch
        n_tok = (image_size // patch) ** 2 + 1  # + class token
        self.embed = MfmaLinear(3 * patch * patch, hidden)
        self.cls = nn.Parameter(torch.zeros(1, 1, hidden))
        self.pos = nn.Parameter(torch.zeros(1, n_tok, hidden) .normal_(0, 0.02))
        self.blocks = nn.ModuleList(BertLayer(hidden, heads, ffn)
                                    for _ in range(layers))
        self.ln = LayerNorm(hidden)
        self.head = MfmaLinear(hidden, num_classes)

    def forward(self, x):
        B, C, H, W = x.shape
        p = self.patch
        # [B,C,H,W] -> [B, (H/p)(W/p), C*p*p] patch unfold
        x = (x.reshape(B, C, H // p, p, W // p, p)
             .permute(0, 2, 4, 1, 3, 5).reshape(B, -1, C * p * p))
        x = self.embed(x)
        x = torch.cat([self.cls.expand(B, 1, -1).to(x.dtype), x], dim=1)
        x = x + self.pos[:, : x.shape[1]].to(x.dtype)
        for blk in self.blocks:
            x = blk(x)
        return self.head(self.ln(x[:, 0]))


def vit_b16(dtype=torch.bfloat16, device="cpu", **kw) -> ViT:
    m = ViT(**kw)
    return m.to(device=device, dtype=dtype)


def bert_base(dtype=torch.bfloat16, device="cpu", **kw) -> BertBase:
    m = BertBase(**kw)
    return m.to(device=device, dtype=dtype)


def llama_1b(dtype=torch.bfloat16, device="cpu", **kw) -> Llama:
    m = Llama(**kw)
    return m.to(device=device, dtype=dtype)
