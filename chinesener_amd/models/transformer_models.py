"""Transformer/TENER CRF graphs (reference model/transformer_crf_bichar.py
:8-69 and model/transformer_tener_crf_bichar.py:8-62): char+bichar emb
concat -> linear project to d_model -> encoder -> dense -> CRF."""
from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from .base import ModelOutput, NerModel
from .layers import CRF, TokenEmbedding
from .transformer import TenerEncoder, TransformerEncoder


class TransformerCrfBichar(NerModel):
    encoder_cls = TransformerEncoder

    def __init__(self, params: Dict):
        super().__init__(params)
        dim = params.get("embedding_dim", 50)
        bdim = params.get("bichar_dim", 50)
        tp = params.get("transformer_params", {})
        d_model = tp.get("d_model", 160)
        self.char_emb = TokenEmbedding(params["vocab_size"], dim,
                                       params.get("embedding"))
        self.bichar_emb = TokenEmbedding(params.get("bichar_vocab_size", 50000),
                                         bdim, params.get("bichar_embedding"))
        self.project = nn.Linear(dim + bdim, d_model)
        self.encoder = self.encoder_cls(
            d_model=d_model, n_heads=tp.get("num_head", 8),
            d_ffn=tp.get("ffn_hidden", 320), n_layers=tp.get("encode_attention_layers", 2),
            dropout=params.get("dropout_rate", 0.2))
        self.logits = nn.Linear(d_model, params["label_size"])
        self.crf = CRF(params["label_size"])

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        emb = torch.cat([self.char_emb(features["token_ids"]),
                         self.bichar_emb(features["bichar_ids"])], dim=-1)
        x = self.project(emb)
        lens = features["mask"].long().sum(1)
        x = self.encoder(x, features["mask"], lens)
        logits = self.logits(x)
        loss = None
        if "label_ids" in features:
            loss = self.crf.neg_log_likelihood(
                logits, features["label_ids"], features["mask"]) \
                / features["token_ids"].shape[0]
        pred = self.crf.decode(logits, features["mask"]) if compute_pred else None
        return ModelOutput(loss, pred, logits=logits)


class TransformerTenerCrfBichar(TransformerCrfBichar):
    """Same input path, TENER relative-position encoder (kernel K9)."""
    encoder_cls = TenerEncoder
