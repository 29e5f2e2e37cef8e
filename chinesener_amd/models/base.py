"""Model base types: every graph is an nn.Module whose forward takes the
feature dict and returns ModelOutput — the torch analog of the
reference's ``build_graph(features, labels, params, is_training) ->
(loss, pred_ids[, task_ids])`` contract (model/bert_bilstm_crf.py:8-34,
model/bert_bilstm_crf_mtl.py:8-66)."""
from __future__ import annotations

from typing import Dict, NamedTuple, Optional

import torch
import torch.nn as nn


class ModelOutput(NamedTuple):
    loss: Optional[torch.Tensor]
    pred_ids: Optional[torch.Tensor] = None
    task_ids: Optional[torch.Tensor] = None
    logits: Optional[torch.Tensor] = None


class NerModel(nn.Module):
    """Base: holds resolved params; subclasses implement forward()."""

    def __init__(self, params: Dict):
        super().__init__()
        self.params = dict(params)

    def forward(self, features: Dict[str, torch.Tensor],
                compute_pred: bool = False) -> ModelOutput:
        raise NotImplementedError


class GradReverse(torch.autograd.Function):
    """Gradient reversal (reference FlipGradientBuilder,
    tools/train_utils.py:47-63): identity forward, -lambda * grad backward."""

    @staticmethod
    def forward(ctx, x, lam: float):
        ctx.lam = lam
        return x.view_as(x)

    @staticmethod
    def backward(ctx, g):
        return -ctx.lam * g, None


def flip_gradient(x: torch.Tensor, lam: float = 1.0) -> torch.Tensor:
    return GradReverse.apply(x, lam)
