"""Single-task model graphs 1-12 (reference model/*.py).

Each class mirrors one reference module's build_graph; the registry in
models/__init__.py binds the same model names + per-model param dicts.
"""
from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from .. import ops
from .base import ModelOutput, NerModel
from .bert import BertConfig, BertEmbeddingOnly, BertModel
from .layers import CRF, BiLSTM, MultiKernelCNN, TokenEmbedding


def _lens(features):
    return features["mask"].long().sum(1)


class BilstmCrf(NerModel):
    """giga char emb -> dropout -> BiLSTM -> dropout -> dense -> CRF
    (reference model/bilstm_crf.py:8-62). Subclasses widen the embedding
    with a word-enhance feature before the BiLSTM."""

    def __init__(self, params: Dict):
        super().__init__(params)
        dim = params.get("embedding_dim", 50)
        self.embedding = TokenEmbedding(params["vocab_size"], dim,
                                        params.get("embedding"))
        self.emb_dropout = nn.Dropout(params.get("embedding_dropout", 0.2))
        extra = self._build_enhance(params)
        rnn = params.get("rnn_params", {})
        self.bilstm = BiLSTM(dim + extra, rnn.get("hidden_units_list", [128])[0],
                             rnn.get("cell_activation", "tanh"),
                             rnn.get("keep_prob_list", [0.8])[0])
        self.logits = nn.Linear(2 * self.bilstm.hidden_size, params["label_size"])
        self.crf = CRF(params["label_size"])

    def _build_enhance(self, params) -> int:
        return 0

    def _enhance(self, features, emb):
        return emb

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        lens = _lens(features)
        emb = self.emb_dropout(self.embedding(features["token_ids"]))
        emb = self._enhance(features, emb)
        hidden = self.bilstm(emb, lens)
        logits = self.logits(hidden)
        loss = None
        if "label_ids" in features:
            loss = self.crf.neg_log_likelihood(
                logits, features["label_ids"], features["mask"]) \
                / features["token_ids"].shape[0]
        pred = self.crf.decode(logits, features["mask"]) if compute_pred else None
        return ModelOutput(loss, pred, logits=logits)


class BilstmCrfSoftword(BilstmCrf):
    """+ learned softword BMES embedding concat (model/bilstm_crf_softword.py)."""

    def _build_enhance(self, params) -> int:
        dim = params.get("embedding_dim", 50)
        self.soft_emb = nn.Embedding(5, dim)
        return dim

    def _enhance(self, features, emb):
        return torch.cat([emb, self.soft_emb(features["softword_ids"])], -1)


class BilstmCrfExSoftword(BilstmCrf):
    """+ multi-hot(5) x learned [5,dim] matrix (model/bilstm_crf_ex_softword.py)."""

    def _build_enhance(self, params) -> int:
        dim = params.get("embedding_dim", 50)
        self.soft_proj = nn.Linear(5, dim, bias=False)
        return dim

    def _enhance(self, features, emb):
        return torch.cat([emb, self.soft_proj(
            features["ex_softword_ids"].to(emb.dtype))], -1)


class BilstmCrfSoftlexicon(BilstmCrf):
    """+ softlexicon fused word embedding 4x50 concat -> BiLSTM(200)
    (model/bilstm_crf_softlexicon.py:13-84; the fuse is kernel K2)."""

    def _build_enhance(self, params) -> int:
        self.word_emb = TokenEmbedding(params["word_vocab_size"],
                                       params.get("word_dim", 50),
                                       params.get("word_embedding"))
        return 4 * params.get("word_dim", 50)

    def _enhance(self, features, emb):
        # table stays fp32 (HIP kernel contract); fuse output cast to the
        # compute dtype afterwards
        fused = ops.softlexicon_fuse(self.word_emb.emb.weight,
                                     features["softlexicon_ids"],
                                     features["softlexicon_weights"])
        return torch.cat([emb, fused.to(emb.dtype)], -1)


class BilstmCrfBichar(BilstmCrf):
    """+ pretrained bichar embedding concat, frozen (model/bilstm_crf_bichar.py)."""

    def _build_enhance(self, params) -> int:
        dim = params.get("bichar_dim", 50)
        self.bichar_emb = TokenEmbedding(params.get("bichar_vocab_size", 50000),
                                         dim, params.get("bichar_embedding"),
                                         freeze=True)
        return dim

    def _enhance(self, features, emb):
        return torch.cat([emb, self.bichar_emb(features["bichar_ids"])], -1)


# ----------------------------------------------------------- BERT family
class BertBase(NerModel):
    def __init__(self, params: Dict):
        super().__init__(params)
        cfg = params.get("bert_config") or BertConfig(
            vocab_size=params.get("vocab_size", 21128))
        self.bert = BertModel(cfg)
        self.bert_dropout = nn.Dropout(params.get("embedding_dropout", 0.2))
        self.hidden = cfg.hidden_size

    def encode(self, features):
        seq = self.bert(features["token_ids"], features["mask"])
        return self.bert_dropout(seq)


class BertCe(BertBase):
    """BERT -> dense -> masked softmax CE; argmax decode (model/bert_ce.py)."""

    def __init__(self, params):
        super().__init__(params)
        self.logits = nn.Linear(self.hidden, params["label_size"])

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        logits = self.logits(self.encode(features))
        loss = None
        if "label_ids" in features:
            loss = self._loss(logits, features)
        pred = logits.argmax(-1) * features["mask"] if compute_pred else None
        return ModelOutput(loss, pred, logits=logits)

    def _loss(self, logits, features):
        return ops.masked_cross_entropy(logits, features["label_ids"],
                                        features["mask"])


class BertDice(BertCe):
    """BERT -> dense -> Dice/DSC loss (model/bert_dice.py, tools/loss.py:19-46)."""

    def _loss(self, logits, features):
        tag2idx = self.params.get("tag2idx", {})
        skip = tuple(tag2idx[t] for t in ("O", "[PAD]", "[CLS]", "[SEP]")
                     if t in tag2idx) or (0, 1)
        return ops.dice_loss(logits, features["label_ids"], features["mask"],
                             skip, self.params.get("alpha", 0.1),
                             self.params.get("gamma", 1.0))


class BertCrf(BertBase):
    """BERT -> dense -> CRF (model/bert_crf.py)."""

    def __init__(self, params):
        super().__init__(params)
        self.logits = nn.Linear(self.hidden, params["label_size"])
        self.crf = CRF(params["label_size"])

    def encode_hidden(self, features):
        return self.encode(features)

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        logits = self.logits(self.encode_hidden(features))
        loss = None
        if "label_ids" in features:
            loss = self.crf.neg_log_likelihood(
                logits, features["label_ids"], features["mask"]) \
                / features["token_ids"].shape[0]
        pred = self.crf.decode(logits, features["mask"]) if compute_pred else None
        return ModelOutput(loss, pred, logits=logits)


class BertBilstmCrf(BertCrf):
    """BERT -> BiLSTM(128, relu, keep 0.8) -> dense -> CRF
    (model/bert_bilstm_crf.py:8-48); the flagship BASELINE config."""

    def __init__(self, params):
        super().__init__(params)
        rnn = params.get("rnn_params", {})
        self.bilstm = BiLSTM(self.hidden, rnn.get("hidden_units_list", [128])[0],
                             rnn.get("cell_activation", "relu"),
                             rnn.get("keep_prob_list", [0.8])[0])
        self.logits = nn.Linear(2 * self.bilstm.hidden_size, params["label_size"])

    def encode_hidden(self, features):
        return self.bilstm(self.encode(features), _lens(features))


class BertCnnCrf(BertCrf):
    """BERT -> multi-kernel conv1d concat -> dense -> CRF (model/bert_cnn_crf.py)."""

    def __init__(self, params):
        super().__init__(params)
        cnn = params.get("cnn_params", {})
        self.cnn = MultiKernelCNN(self.hidden, cnn.get("filters", 128),
                                  cnn.get("kernel_sizes", (2, 3, 4)),
                                  cnn.get("keep_prob", 0.8))
        self.logits = nn.Linear(self.cnn.output_size, params["label_size"])

    def encode_hidden(self, features):
        return self.cnn(self.encode(features))


class BertBilstmCrfBigram(NerModel):
    """BERT token-embedding-only (use_bert False default) -> shifted-concat
    bigram emb -> BiLSTM -> CRF (model/bert_bilstm_crf_bigram.py:8-59)."""

    def __init__(self, params: Dict):
        super().__init__(params)
        cfg = params.get("bert_config") or BertConfig(
            vocab_size=params.get("vocab_size", 21128))
        self.use_bert = params.get("use_bert", False)
        self.encoder = (BertModel(cfg) if self.use_bert
                        else BertEmbeddingOnly(cfg))
        self.dropout = nn.Dropout(params.get("embedding_dropout", 0.2))
        H = cfg.hidden_size
        rnn = params.get("rnn_params", {})
        self.bilstm = BiLSTM(2 * H, rnn.get("hidden_units_list", [128])[0],
                             rnn.get("cell_activation", "tanh"),
                             rnn.get("keep_prob_list", [0.8])[0])
        self.logits = nn.Linear(2 * self.bilstm.hidden_size, params["label_size"])
        self.crf = CRF(params["label_size"])

    def forward(self, features, compute_pred: bool = False) -> ModelOutput:
        if self.use_bert:
            emb = self.encoder(features["token_ids"], features["mask"])
        else:
            emb = self.encoder(features["token_ids"])
        emb = self.dropout(emb)
        # bigram: concat emb_t with emb_{t+1} (shift left, zero pad tail)
        nxt = torch.cat([emb[:, 1:], torch.zeros_like(emb[:, :1])], dim=1)
        bigram = torch.cat([emb, nxt], dim=-1)
        hidden = self.bilstm(bigram, _lens(features))
        logits = self.logits(hidden)
        loss = None
        if "label_ids" in features:
            loss = self.crf.neg_log_likelihood(
                logits, features["label_ids"], features["mask"]) \
                / features["token_ids"].shape[0]
        pred = self.crf.decode(logits, features["mask"]) if compute_pred else None
        return ModelOutput(loss, pred, logits=logits)


class BertBilstmCrfSoftlexicon(BertCrf):
    """BERT(768) + softlexicon fused emb concat -> BiLSTM(200) -> CRF
    (model/bert_bilstm_crf_softlexicon.py:14-85)."""

    def __init__(self, params):
        super().__init__(params)
        wdim = params.get("word_dim", 50)
        self.word_emb = TokenEmbedding(params.get("word_vocab_size", 5003),
                                       wdim, params.get("word_embedding"))
        rnn = params.get("rnn_params", {})
        self.bilstm = BiLSTM(self.hidden + 4 * wdim,
                             rnn.get("hidden_units_list", [200])[0],
                             rnn.get("cell_activation", "tanh"),
                             rnn.get("keep_prob_list", [0.8])[0])
        self.logits = nn.Linear(2 * self.bilstm.hidden_size, params["label_size"])

    def encode_hidden(self, features):
        seq = self.encode(features)
        fused = ops.softlexicon_fuse(self.word_emb.emb.weight,
                                     features["softlexicon_ids"],
                                     features["softlexicon_weights"])
        return self.bilstm(torch.cat([seq, fused.to(seq.dtype)], -1),
                           _lens(features))
