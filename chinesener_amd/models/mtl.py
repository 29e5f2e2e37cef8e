"""Multi-task + adversarial shared-BERT graphs (reference
model/bert_bilstm_crf_mtl.py:8-83 and model/bert_bilstm_crf_adv.py:9-105).
"""
from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from .base import ModelOutput, NerModel, flip_gradient
from .bert import BertConfig, BertModel
from .layers import CRF, BiLSTM


def _lens(features):
    return features["mask"].long().sum(1)


class TaskTower(nn.Module):
    """Per-task BiLSTM + dense + CRF tower."""

    def __init__(self, input_size: int, hidden: int, label_size: int,
                 activation: str = "relu", keep_prob: float = 0.8):
        super().__init__()
        self.bilstm = BiLSTM(input_size, hidden, activation, keep_prob)
        self.logits = nn.Linear(2 * hidden, label_size)
        self.crf = CRF(label_size)

    def forward(self, seq, lens):
        h = self.bilstm(seq, lens)
        return h, self.logits(h)


class BertBilstmCrfMtl(NerModel):
    """Shared BERT -> per-task towers; per-task masked NLL weighted sum /
    batch; joint decode via where(task_ids==0, p0, p1)
    (reference :29-66). ``asymmetry`` concatenates task0's lstm output into
    task1's tower input (:51-53)."""

    def __init__(self, params: Dict):
        super().__init__(params)
        cfg = params.get("bert_config") or BertConfig(
            vocab_size=params.get("vocab_size", 21128))
        self.bert = BertModel(cfg)
        self.dropout = nn.Dropout(params.get("embedding_dropout", 0.2))
        self.task_list = params.get("task_list", ["task0", "task1"])
        self.task_weight = params.get("task_weight", (0.5, 0.5))
        self.asymmetry = params.get("asymmetry", False)
        rnn = params.get("rnn_params", {})
        hidden = rnn.get("hidden_units_list", [128])[0]
        act = rnn.get("cell_activation", "relu")
        keep = rnn.get("keep_prob_list", [0.8])[0]
        sizes = self._label_sizes(params)
        self.label_sizes = sizes
        self.tower0 = TaskTower(cfg.hidden_size, hidden, sizes[0], act, keep)
        in1 = cfg.hidden_size + (2 * hidden if self.asymmetry else 0)
        self.tower1 = TaskTower(in1, hidden, sizes[1], act, keep)

    def _label_sizes(self, params):
        sizes = []
        for t in params.get("task_list", []):
            sub = params.get(t)
            if isinstance(sub, dict) and "label_size" in sub:
                sizes.append(sub["label_size"])
        if len(sizes) != 2:
            sizes = [params.get("label_size") or 10] * 2
        return sizes

    def _shared(self, features):
        seq = self.bert(features["token_ids"], features["mask"])
        return self.dropout(seq)

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        seq = self._shared(features)
        lens = _lens(features)
        task_ids = features.get("task_ids")
        if task_ids is None:
            task_ids = torch.zeros_like(features["token_ids"])
        sample_task = task_ids[:, 0]                        # [B]
        B = seq.shape[0]

        h0, logit0 = self.tower0(seq, lens)
        in1 = torch.cat([seq, h0], dim=-1) if self.asymmetry else seq
        _, logit1 = self.tower1(in1, lens)

        loss = None
        if "label_ids" in features:
            loss = self._mtl_loss(features, logit0, logit1, sample_task, B)
        pred = None
        if compute_pred:
            p0 = self.tower0.crf.decode(logit0, features["mask"])
            p1 = self.tower1.crf.decode(logit1, features["mask"])
            pred = torch.where(sample_task[:, None] == 0, p0, p1)
        return ModelOutput(loss, pred, task_ids=task_ids)

    def _mtl_loss(self, features, logit0, logit1, sample_task, B):
        # per-task masked NLL: -ll per sample, masked by task membership
        # (reference sums (-ll * mask) * weight / batch_size, :38-63)
        from .. import ops
        labels = features["label_ids"]
        # other-task samples are masked out of each tower's loss; clamp their
        # (other-tagset) label ids so the gather stays in range
        ll0 = ops.crf_nll(logit0, labels.clamp(max=self.label_sizes[0] - 1),
                          features["mask"], self.tower0.crf.transitions)
        ll1 = ops.crf_nll(logit1, labels.clamp(max=self.label_sizes[1] - 1),
                          features["mask"], self.tower1.crf.transitions)
        m0 = (sample_task == 0).to(ll0.dtype)
        m1 = (sample_task == 1).to(ll1.dtype)
        w0, w1 = self.task_weight
        return ((-ll0 * m0).sum() * w0 + (-ll1 * m1).sum() * w1) / B


class BertBilstmCrfAdv(BertBilstmCrfMtl):
    """Adds a shared BiLSTM + max-pool -> flip_gradient -> dense(2) task
    discriminator (reference :30-47); towers consume [BERT ; shared-lstm];
    loss = task losses + lambda * adv CE (:86-94)."""

    def __init__(self, params: Dict):
        super().__init__(params)
        cfg = params.get("bert_config") or BertConfig(
            vocab_size=params.get("vocab_size", 21128))
        rnn = params.get("rnn_params", {})
        hidden = rnn.get("hidden_units_list", [128])[0]
        self.shared_lstm = BiLSTM(cfg.hidden_size, hidden,
                                  rnn.get("cell_activation", "relu"),
                                  rnn.get("keep_prob_list", [0.8])[0])
        self.discriminator = nn.Linear(2 * hidden, 2)
        self.lam = params.get("lambda", 0.05)
        self.shrink = params.get("shrink_gradient_reverse", 0.01)
        # towers consume shared-lstm output concat BERT sequence
        act = rnn.get("cell_activation", "relu")
        keep = rnn.get("keep_prob_list", [0.8])[0]
        in_dim = cfg.hidden_size + 2 * hidden
        self.tower0 = TaskTower(in_dim, hidden, self.label_sizes[0], act, keep)
        in1 = in_dim + (2 * hidden if self.asymmetry else 0)
        self.tower1 = TaskTower(in1, hidden, self.label_sizes[1], act, keep)

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        seq = self._shared(features)
        lens = _lens(features)
        task_ids = features.get("task_ids")
        if task_ids is None:
            task_ids = torch.zeros_like(features["token_ids"])
        sample_task = task_ids[:, 0]
        B = seq.shape[0]

        shared = self.shared_lstm(seq, lens)                   # [B,L,2h]
        # adversarial head: mask pads to -inf, max-pool over L (:34-40)
        neg = torch.finfo(shared.dtype).min
        pooled = shared.masked_fill(
            features["mask"][:, :, None] == 0, neg).max(dim=1).values
        adv_logits = self.discriminator(flip_gradient(pooled, self.shrink))

        tower_in = torch.cat([seq, shared], dim=-1)
        h0, logit0 = self.tower0(tower_in, lens)
        in1 = torch.cat([tower_in, h0], dim=-1) if self.asymmetry else tower_in
        _, logit1 = self.tower1(in1, lens)

        loss = None
        if "label_ids" in features:
            task_loss = self._mtl_loss(features, logit0, logit1, sample_task, B)
            adv = nn.functional.cross_entropy(adv_logits.float(), sample_task)
            loss = task_loss + self.lam * adv
        pred = None
        if compute_pred:
            p0 = self.tower0.crf.decode(logit0, features["mask"])
            p1 = self.tower1.crf.decode(logit1, features["mask"])
            pred = torch.where(sample_task[:, None] == 0, p0, p1)
        return ModelOutput(loss, pred, task_ids=task_ids)
