"""MRC span-pointer variant (reference mrc/archive.py:128-279, the
legacy start/end/span formulation): per-position start/end heads plus an
O(L^2) span-match head over candidate (start, end) pairs, with a
weighted 3-part loss and span extraction."""
from __future__ import annotations

from typing import Dict, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..models.base import ModelOutput, NerModel
from ..models.bert import BertConfig, BertModel


class MrcSpan(NerModel):
    """BERT -> start/end logits [B,L,2] + span matrix [B,L,L]."""

    def __init__(self, params: Dict):
        super().__init__(params)
        cfg = params.get("bert_config") or BertConfig(
            vocab_size=params.get("vocab_size", 21128))
        self.bert = BertModel(cfg)
        H = cfg.hidden_size
        self.dropout = nn.Dropout(params.get("dropout_rate", 0.2))
        self.start_head = nn.Linear(H, 1)
        self.end_head = nn.Linear(H, 1)
        # span head: gelu(dense([h_i ; h_j])) -> 1 (archive.py tiled concat)
        self.span_proj = nn.Linear(2 * H, H)
        self.span_out = nn.Linear(H, 1)
        w = params.get("loss_weights", (1.0, 1.0, 1.0))
        self.w_start, self.w_end, self.w_span = w

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        seq = self.dropout(self.bert(features["token_ids"], features["mask"],
                                     features.get("segment_ids")))
        B, L, H = seq.shape
        start_logits = self.start_head(seq).squeeze(-1)      # [B,L]
        end_logits = self.end_head(seq).squeeze(-1)          # [B,L]
        # span matrix from pairwise concat (tiled like archive.py:169-183)
        hi = seq.unsqueeze(2).expand(B, L, L, H)
        hj = seq.unsqueeze(1).expand(B, L, L, H)
        span_logits = self.span_out(
            F.gelu(self.span_proj(torch.cat([hi, hj], -1)))).squeeze(-1)

        text_mask = features.get("text_mask", features["mask"]).float()
        loss = None
        if "start_ids" in features:
            bce = F.binary_cross_entropy_with_logits
            m = text_mask
            loss_s = (bce(start_logits, features["start_ids"].float(),
                          reduction="none") * m).sum() / m.sum().clamp(min=1)
            loss_e = (bce(end_logits, features["end_ids"].float(),
                          reduction="none") * m).sum() / m.sum().clamp(min=1)
            pair_mask = m.unsqueeze(2) * m.unsqueeze(1)
            # only upper-triangular candidates (start <= end)
            triu = torch.triu(torch.ones(L, L, device=seq.device))
            pair_mask = pair_mask * triu
            loss_sp = (bce(span_logits, features["span_ids"].float(),
                           reduction="none") * pair_mask).sum() \
                / pair_mask.sum().clamp(min=1)
            loss = (self.w_start * loss_s + self.w_end * loss_e
                    + self.w_span * loss_sp)
        pred = None
        if compute_pred:
            pred = extract_spans_pred(start_logits, end_logits, span_logits,
                                      text_mask)
        return ModelOutput(loss, pred, logits=span_logits)


def extract_spans_pred(start_logits, end_logits, span_logits, text_mask,
                       threshold: float = 0.0) -> torch.Tensor:
    """Candidate filtering (archive.py:216-246): positions with start>thr
    pair with later end>thr; keep pairs whose span logit > thr. Returns a
    [B,L] BIO-ish id tensor (1=B, 2=I, 0=O) for trainer compatibility."""
    B, L = start_logits.shape
    out = torch.zeros(B, L, dtype=torch.long, device=start_logits.device)
    starts = (start_logits > threshold) & text_mask.bool()
    ends = (end_logits > threshold) & text_mask.bool()
    for b in range(B):
        s_idx = starts[b].nonzero(as_tuple=True)[0]
        e_idx = ends[b].nonzero(as_tuple=True)[0]
        for s in s_idx.tolist():
            cand = e_idx[e_idx >= s]
            if len(cand) == 0:
                continue
            e = int(cand[0])
            if float(span_logits[b, s, e]) > threshold:
                out[b, s] = 1
                if e > s:
                    out[b, s + 1:e + 1] = 2
    return out


def make_span_labels(label_ids: torch.Tensor) -> Tuple[torch.Tensor, ...]:
    """BIO label ids (0=O,1=B,2=I) -> start/end/span supervision tensors
    (archive.py numpy span alignment, :249-279)."""
    B, L = label_ids.shape
    start = (label_ids == 1).long()
    end = torch.zeros_like(label_ids)
    span = torch.zeros(B, L, L, dtype=torch.long, device=label_ids.device)
    for b in range(B):
        s = None
        for t in range(L):
            v = int(label_ids[b, t])
            if v == 1:
                if s is not None:
                    end[b, t - 1] = 1
                    span[b, s, t - 1] = 1
                s = t
            elif v == 0 and s is not None:
                end[b, t - 1] = 1
                span[b, s, t - 1] = 1
                s = None
        if s is not None:
            end[b, L - 1] = 1
            span[b, s, L - 1] = 1
    return start, end, span
