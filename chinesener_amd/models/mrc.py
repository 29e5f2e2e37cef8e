"""MRC-style NER: [CLS]+query+[SEP]+text -> BIO tagging over the text
region (reference mrc/model.py:7-106; dataset expansion x3 tag queries,
mrc/dataset.py:90-203)."""
from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from .. import ops
from .base import ModelOutput, NerModel
from .bert import BertConfig, BertModel

TAG2QUERY = {   # natural-language queries per entity type (mrc/dataset.py:12-16)
    "PER": "找出人名和虚构的人物形象",
    "LOC": "找出国家城市山川等抽象或具体的地点",
    "ORG": "找出公司商业机构社会组织等组织机构",
}
MRC_LABELS = {"O": 0, "B": 1, "I": 2}


class MrcBio(NerModel):
    """BERT -> dense(3) BIO logits; CE masked to the text region."""

    def __init__(self, params: Dict):
        super().__init__(params)
        cfg = params.get("bert_config") or BertConfig(
            vocab_size=params.get("vocab_size", 21128))
        self.bert = BertModel(cfg)
        self.dropout = nn.Dropout(params.get("dropout_rate", 0.2))
        self.logits = nn.Linear(cfg.hidden_size, 3)

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        seq = self.bert(features["token_ids"], features["mask"],
                        features.get("segment_ids"))
        logits = self.logits(self.dropout(seq))
        loss = None
        text_mask = features.get("text_mask", features["mask"])
        if "label_ids" in features:
            loss = ops.masked_cross_entropy(logits, features["label_ids"],
                                            text_mask)
        pred = logits.argmax(-1) * text_mask if compute_pred else None
        return ModelOutput(loss, pred, logits=logits)
