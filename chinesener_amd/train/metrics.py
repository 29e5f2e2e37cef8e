"""Streaming eval metrics: overall accuracy + per-tag P/R/F1, masked to
real tokens excluding CLS/SEP (reference get_eval_metrics/calc_metrics,
tools/train_utils.py:105-142; the reference's mask is sign(label)-based
:109-111 — here an explicit special-token exclusion on the same ids)."""
from __future__ import annotations

from typing import Dict, Optional

import torch


class TagMetrics:
    """Accumulates a confusion matrix over real (non-special) tokens."""

    def __init__(self, label_size: int, idx2tag: Optional[Dict[int, str]] = None):
        self.label_size = label_size
        self.idx2tag = idx2tag or {}
        self.conf = torch.zeros(label_size, label_size, dtype=torch.long)
        self.special = {i for i, t in (idx2tag or {}).items()
                        if t in ("[PAD]", "[CLS]", "[SEP]")}

    def update(self, preds: torch.Tensor, labels: torch.Tensor,
               mask: torch.Tensor) -> None:
        keep = mask.bool()
        for s in self.special:
            keep &= labels != s
        p = preds[keep].reshape(-1).cpu()
        y = labels[keep].reshape(-1).cpu()
        idx = y * self.label_size + p
        binc = torch.bincount(idx, minlength=self.label_size ** 2)
        self.conf += binc.reshape(self.label_size, self.label_size)

    def compute(self) -> Dict[str, float]:
        conf = self.conf.double()
        total = conf.sum().clamp(min=1)
        acc = conf.diag().sum() / total
        out = {"accuracy": float(acc)}
        tp = conf.diag()
        support = conf.sum(1)
        predicted = conf.sum(0)
        precision = tp / predicted.clamp(min=1)
        recall = tp / support.clamp(min=1)
        f1 = 2 * precision * recall / (precision + recall).clamp(min=1e-12)
        for i in range(self.label_size):
            if i in self.special or support[i] == 0:
                continue
            tag = self.idx2tag.get(i, str(i))
            out[f"{tag}_precision"] = float(precision[i])
            out[f"{tag}_recall"] = float(recall[i])
            out[f"{tag}_f1"] = float(f1[i])
        # micro over non-special, non-O tags
        keep = [i for i in range(self.label_size)
                if i not in self.special and self.idx2tag.get(i) != "O"]
        if keep:
            k = torch.tensor(keep)
            mtp = tp[k].sum()
            mp = mtp / predicted[k].sum().clamp(min=1)
            mr = mtp / support[k].sum().clamp(min=1)
            out["micro_f1"] = float(2 * mp * mr / max(1e-12, float(mp + mr)))
        return out
