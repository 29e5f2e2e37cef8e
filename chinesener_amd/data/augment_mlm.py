"""BERT-MLM paraphrase augmentation (reference
data/people_daily_augment/augment_mlm.py:25-184): mask a fraction of
NON-entity positions, run the MLM head, refill from the top-k
prediction — entity spans and their tags are preserved."""
from __future__ import annotations

import random
from typing import List, Optional, Sequence, Tuple

import torch

from ..models.bert import BertConfig, BertMlmHead, BertModel
from .tokenizer import Vocab, WordpieceTokenizer


class MlmSR:
    """Mask-and-refill synonym replacement via an MLM head.

    The reference builds a full estimator + FastPredict stream over the
    pretrained checkpoint; here the model is injectable (random-init in
    tests, trained weights in production)."""

    def __init__(self, bert: Optional[BertModel] = None,
                 tokenizer: Optional[WordpieceTokenizer] = None,
                 mask_prob: float = 0.15, topk: int = 5, seed: int = 1234,
                 max_seq_len: int = 512, device: str = "cpu"):
        self.tokenizer = tokenizer or WordpieceTokenizer(Vocab.synthetic())
        if bert is None:
            cfg = BertConfig(vocab_size=len(self.tokenizer.vocab),
                             hidden_size=128, num_hidden_layers=2,
                             num_attention_heads=4, intermediate_size=256)
            bert = BertModel(cfg)
        self.bert = bert.to(device).eval()
        self.head = BertMlmHead(self.bert).to(device).eval()
        self.mask_prob = mask_prob
        self.topk = topk
        self.max_seq_len = max_seq_len
        self.rng = random.Random(seed)
        self.device = device
        v = self.tokenizer.vocab
        self.mask_id = v.stoi.get("[MASK]", v.stoi.get("[UNK]", 1))
        self.cls_id = v.stoi["[CLS]"]
        self.sep_id = v.stoi["[SEP]"]

    @torch.no_grad()
    def __call__(self, sentence: Sequence[str], tags: Sequence[str]
                 ) -> Tuple[List[str], List[str]]:
        chars = list(sentence)[: self.max_seq_len - 2]
        tags = list(tags)[: len(chars)]
        ids = self.tokenizer.convert_tokens_to_ids(chars)
        maskable = [i for i, t in enumerate(tags) if t == "O"]
        n_mask = max(1, int(len(maskable) * self.mask_prob)) if maskable else 0
        if not n_mask:
            return chars, tags
        chosen = sorted(self.rng.sample(maskable, min(n_mask, len(maskable))))
        inp = [self.cls_id] + list(ids) + [self.sep_id]
        for i in chosen:
            inp[i + 1] = self.mask_id
        token_ids = torch.tensor([inp], device=self.device)
        attn = torch.ones_like(token_ids)
        logits = self.head(self.bert(token_ids, attn))
        out = list(chars)
        itos = self.tokenizer.vocab.itos
        for i in chosen:
            top = torch.topk(logits[0, i + 1], self.topk).indices.tolist()
            cand = [itos[j] for j in top
                    if j < len(itos) and not itos[j].startswith("[")]
            if cand:
                out[i] = self.rng.choice(cand)
        return out, tags


def augment_mlm(sentences: Sequence[Sequence[str]],
                tag_seqs: Sequence[Sequence[str]],
                mlm: Optional[MlmSR] = None,
                ) -> Tuple[List[List[str]], List[List[str]]]:
    mlm = mlm or MlmSR()
    out_s, out_t = [], []
    for sent, tags in zip(sentences, tag_seqs):
        s, t = mlm(sent, tags)
        out_s.append(s)
        out_t.append(t)
    return out_s, out_t
