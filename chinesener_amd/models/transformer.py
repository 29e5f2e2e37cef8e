"""Vanilla Transformer encoder + TENER relative-position encoder.

Parity with reference tools/transformer/{modules,encoder,tener}.py:
post-LN residual blocks (:58-65), relu FFN (:68-80), additive key mask
(:118-130), sinusoidal absolute positions (:177-197); TENER relative
attention with learnable per-head u/v biases, no key projection and
unscaled scores (tener.py:12-48, :94). The shift trick (:51-74) is
replaced by direct R[j-i] indexing inside the kernel.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops
from ..ops import reference as ref


class FFN(nn.Module):
    def __init__(self, d_model: int, d_ffn: int, dropout: float):
        super().__init__()
        self.fc1 = nn.Linear(d_model, d_ffn)
        self.fc2 = nn.Linear(d_ffn, d_model)
        self.dropout = nn.Dropout(dropout)
        self.ln_w = nn.Parameter(torch.ones(d_model))
        self.ln_b = nn.Parameter(torch.zeros(d_model))

    def forward(self, x):
        f = self.dropout(self.fc2(torch.relu(self.fc1(x))))
        return ops.add_layernorm(f, x, self.ln_w, self.ln_b)


class EncoderLayer(nn.Module):
    """Vanilla MHA block, d_model=160, 8 heads by default
    (model/transformer_crf_bichar.py:57)."""

    def __init__(self, d_model: int, n_heads: int, d_ffn: int, dropout: float):
        super().__init__()
        self.n_heads = n_heads
        self.head_dim = d_model // n_heads
        self.qkv = nn.Linear(d_model, 3 * d_model)
        self.out = nn.Linear(d_model, d_model)
        self.dropout = nn.Dropout(dropout)
        self.ln_w = nn.Parameter(torch.ones(d_model))
        self.ln_b = nn.Parameter(torch.zeros(d_model))
        self.ffn = FFN(d_model, d_ffn, dropout)

    def forward(self, x, mask, lens=None):
        B, L, D = x.shape
        qkv = self.qkv(x).reshape(B, L, 3, self.n_heads, self.head_dim)
        ctx = ops.attention_qkv(qkv, mask=mask, lens=lens)
        a = self.dropout(self.out(ctx.reshape(B, L, D)))
        x = ops.add_layernorm(a, x, self.ln_w, self.ln_b)
        return self.ffn(x)


class TransformerEncoder(nn.Module):
    def __init__(self, d_model: int = 160, n_heads: int = 8, d_ffn: int = 320,
                 n_layers: int = 2, dropout: float = 0.2, max_len: int = 512):
        super().__init__()
        self.layers = nn.ModuleList(
            EncoderLayer(d_model, n_heads, d_ffn, dropout) for _ in range(n_layers))
        self.register_buffer("pos_table",
                             ref.sinusoidal_table(max_len, d_model), persistent=False)
        self.dropout = nn.Dropout(dropout)

    def forward(self, x, mask, lens=None):
        L = x.shape[1]
        x = self.dropout(x + self.pos_table[:L].to(x.dtype))
        for layer in self.layers:
            x = layer(x, mask, lens)
        return x


class TenerLayer(nn.Module):
    """Relative MHA: no key projection (tener.py:94), unscaled scores,
    learnable per-head u/v (tener.py:12-48)."""

    def __init__(self, d_model: int, n_heads: int, d_ffn: int, dropout: float):
        super().__init__()
        self.n_heads = n_heads
        self.head_dim = d_model // n_heads
        self.qv = nn.Linear(d_model, 2 * d_model)    # Q and V projections only
        self.out = nn.Linear(d_model, d_model)
        self.u = nn.Parameter(torch.zeros(n_heads, self.head_dim))
        self.v = nn.Parameter(torch.zeros(n_heads, self.head_dim))
        nn.init.normal_(self.u, std=0.02)
        nn.init.normal_(self.v, std=0.02)
        self.dropout = nn.Dropout(dropout)
        self.ln_w = nn.Parameter(torch.ones(d_model))
        self.ln_b = nn.Parameter(torch.zeros(d_model))
        self.ffn = FFN(d_model, d_ffn, dropout)

    def forward(self, x, mask, rel):
        B, L, D = x.shape
        qv = self.qv(x).reshape(B, L, 2, self.n_heads, self.head_dim)
        q, v = qv[:, :, 0].transpose(1, 2), qv[:, :, 1].transpose(1, 2)
        k = x.reshape(B, L, self.n_heads, self.head_dim).transpose(1, 2)
        ctx = ops.tener_attention(q, k, v, self.u, self.v, rel, mask)
        a = self.dropout(self.out(ctx.transpose(1, 2).reshape(B, L, D)))
        x = ops.add_layernorm(a, x, self.ln_w, self.ln_b)
        return self.ffn(x)


class TenerEncoder(nn.Module):
    def __init__(self, d_model: int = 160, n_heads: int = 8, d_ffn: int = 320,
                 n_layers: int = 2, dropout: float = 0.2):
        super().__init__()
        self.head_dim = d_model // n_heads
        self.layers = nn.ModuleList(
            TenerLayer(d_model, n_heads, d_ffn, dropout) for _ in range(n_layers))
        self.dropout = nn.Dropout(dropout)
        self._rel_cache = {}

    def _rel(self, L, device, dtype):
        key = (L, device, dtype)
        if key not in self._rel_cache:
            self._rel_cache[key] = ref.relative_table(
                L, self.head_dim, device=device).to(dtype)
        return self._rel_cache[key]

    def forward(self, x, mask, lens=None):
        x = self.dropout(x)
        rel = self._rel(x.shape[1], x.device, x.dtype)
        for layer in self.layers:
            x = layer(x, mask, rel)
        return x
