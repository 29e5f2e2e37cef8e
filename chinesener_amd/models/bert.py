"""BERT encoder, MI355X-first.

Capability parity with the reference's use of bert_base.bert.modeling
BertModel (tools/layer.py:63-81; config pretrain_model/ch_google/
bert_config.json: 12L-768H-12A, gelu, vocab 21128). Re-designed for the
hardware: the projection/FFN GEMMs go through rocBLAS/hipBLASLt
(torch.matmul); attention core, residual+LayerNorm and bias+GELU are the
hand-written gfx950 kernels behind chinesener_amd.ops.
Compute dtype is bf16 on GPU; LayerNorm statistics are fp32 inside the
kernel.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from .. import ops


@dataclass
class BertConfig:
    vocab_size: int = 21128
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    hidden_dropout_prob: float = 0.1
    attention_probs_dropout_prob: float = 0.1
    layer_norm_eps: float = 1e-12
    initializer_range: float = 0.02


class BertEmbeddings(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.word = nn.Embedding(cfg.vocab_size, cfg.hidden_size, padding_idx=0)
        self.position = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.token_type = nn.Embedding(cfg.type_vocab_size, cfg.hidden_size)
        self.ln_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.ln_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.dropout = nn.Dropout(cfg.hidden_dropout_prob)
        self.eps = cfg.layer_norm_eps

    def forward(self, token_ids, token_type_ids=None):
        B, L = token_ids.shape
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(token_ids)
        # fused 3-table gather-sum (K1); torch fallback off-GPU
        x = ops.embed3(self.word.weight, self.position.weight,
                       self.token_type.weight, token_ids, token_type_ids)
        x = ops.layernorm(x, self.ln_weight, self.ln_bias, self.eps)
        return self.dropout(x)


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        H = cfg.hidden_size
        self.n_heads = cfg.num_attention_heads
        self.head_dim = H // cfg.num_attention_heads
        self.qkv = nn.Linear(H, 3 * H)       # fused QKV projection (one GEMM)
        self.attn_out = nn.Linear(H, H)
        self.ln1_w = nn.Parameter(torch.ones(H))
        self.ln1_b = nn.Parameter(torch.zeros(H))
        self.ffn_in = nn.Linear(H, cfg.intermediate_size, bias=False)
        self.ffn_in_bias = nn.Parameter(torch.zeros(cfg.intermediate_size))
        self.ffn_out = nn.Linear(cfg.intermediate_size, H)
        self.ln2_w = nn.Parameter(torch.ones(H))
        self.ln2_b = nn.Parameter(torch.zeros(H))
        self.dropout = nn.Dropout(cfg.hidden_dropout_prob)
        self.p_drop = cfg.hidden_dropout_prob
        self.p_attn_drop = cfg.attention_probs_dropout_prob
        self.eps = cfg.layer_norm_eps

    def forward(self, x, mask, lens):
        B, L, H = x.shape
        qkv = ops.linear(x, self.qkv.weight, self.qkv.bias) \
            .reshape(B, L, 3, self.n_heads, self.head_dim)
        ctx = ops.attention_qkv(qkv, mask=mask, lens=lens,
                                p_drop=self.p_attn_drop,
                                training=self.training).reshape(B, L, H)
        a = ops.linear(ctx, self.attn_out.weight, self.attn_out.bias)
        x = ops.dropout_add_layernorm(a, x, self.ln1_w, self.ln1_b, self.eps,
                                      self.p_drop, self.training)
        f = ops.bias_gelu(ops.linear(x, self.ffn_in.weight), self.ffn_in_bias)
        f = ops.linear(f, self.ffn_out.weight, self.ffn_out.bias)
        return ops.dropout_add_layernorm(f, x, self.ln2_w, self.ln2_b,
                                         self.eps, self.p_drop, self.training)


class BertModel(nn.Module):
    """Returns the last-layer sequence output [B,L,H] (the reference's
    pretrain_bert_embedding returns sequence output + dropout,
    tools/layer.py:63-81)."""

    def __init__(self, cfg: BertConfig | None = None):
        super().__init__()
        self.cfg = cfg or BertConfig()
        self.embeddings = BertEmbeddings(self.cfg)
        self.layers = nn.ModuleList(
            BertLayer(self.cfg) for _ in range(self.cfg.num_hidden_layers))
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=self.cfg.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)
            if isinstance(m, nn.Embedding) and m.padding_idx is not None:
                with torch.no_grad():
                    m.weight[m.padding_idx].zero_()

    def forward(self, token_ids, mask=None, token_type_ids=None):
        lens = mask.long().sum(1) if mask is not None else None
        x = self.embeddings(token_ids, token_type_ids)
        for layer in self.layers:
            x = layer(x, mask, lens)
        return x


class BertMlmHead(nn.Module):
    """MLM head: transform dense + layernorm + tied-embedding logits
    (reference get_masked_lm_output, augment_mlm.py:49-72)."""

    def __init__(self, bert: BertModel):
        super().__init__()
        cfg = bert.cfg
        self.transform = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.ln_w = nn.Parameter(torch.ones(cfg.hidden_size))
        self.ln_b = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.word_embedding = bert.embeddings.word.weight  # tied
        self.output_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.eps = cfg.layer_norm_eps

    def forward(self, seq_out: torch.Tensor) -> torch.Tensor:
        x = torch.nn.functional.gelu(self.transform(seq_out))
        x = ops.layernorm(x, self.ln_w, self.ln_b, self.eps)
        return x @ self.word_embedding.to(x.dtype).T + \
            self.output_bias.to(x.dtype)


class BertEmbeddingOnly(nn.Module):
    """Wordpiece embedding + position postprocessor only, no encoder —
    the reference's bert_token_embedding (tools/layer.py:84-109), used by
    bert_bilstm_crf_bigram with use_bert=False."""

    def __init__(self, cfg: BertConfig | None = None):
        super().__init__()
        self.embeddings = BertEmbeddings(cfg or BertConfig())

    def forward(self, token_ids):
        return self.embeddings(token_ids)
