"""Data augmentation (reference data/people_daily_augment/augmentation.py
:9-236 + build_ner_dict.py:22-63): chunk a tagged sentence by entity
spans, then

* ``EntityReplace``  — swap an entity surface for another of the same
  type sampled from a corpus-built entity dictionary;
* ``SynonymReplace`` — replace non-entity words with embedding
  nearest-neighbours (the reference uses gensim top-5 NN; gensim is not
  in this image, so the NN search runs on an embedding matrix directly);
* ``SentenceShuffle`` — swap comma-separated clauses;
* ``augment()``      — driver applying a random subset per sentence and
  dumping ``train_augment.pkl``.

MLM-based paraphrase augmentation lives in ``augment_mlm.py``.
"""
from __future__ import annotations

import os
import pickle
import random
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..eval.entity_eval import extract_spans
from .datasets import load_data

Chunk = Tuple[str, Optional[str]]  # (surface, entity type or None)


def chunk_by_tag(sentence: Sequence[str], tags: Sequence[str]) -> List[Chunk]:
    """Split a char sequence into entity/non-entity chunks (reference
    AugHandler chunkers :24-64)."""
    text = "".join(sentence)
    spans = extract_spans(tags)
    chunks: List[Chunk] = []
    pos = 0
    for typ, start, end in spans:
        if start > pos:
            chunks.append((text[pos:start], None))
        chunks.append((text[start:end], typ))
        pos = end
    if pos < len(text):
        chunks.append((text[pos:], None))
    return chunks


def chunks_to_bio(chunks: List[Chunk]) -> Tuple[List[str], List[str]]:
    sent, tags = [], []
    for surface, typ in chunks:
        for i, ch in enumerate(surface):
            sent.append(ch)
            if typ is None:
                tags.append("O")
            else:
                tags.append(("B-" if i == 0 else "I-") + typ)
    return sent, tags


def build_entity_dict(data_names: Sequence[str], data_dir: str,
                      split: str = "train") -> Dict[str, List[str]]:
    """Entity surfaces per type over one or more corpora (reference
    build_ner_dict.py:22-63)."""
    out: Dict[str, set] = {}
    for name in data_names:
        sentences, tags = load_data(name, os.path.join(data_dir, name), split)
        for sent, tag in zip(sentences, tags):
            text = "".join(sent)
            for typ, s, e in extract_spans(tag):
                out.setdefault(typ, set()).add(text[s:e])
    return {t: sorted(v) for t, v in out.items()}


class EntityReplace:
    """Swap each entity with probability p for a same-type dict sample
    (reference EntityReplace :67-104)."""

    def __init__(self, entity_dict: Dict[str, List[str]], p: float = 0.5,
                 seed: int = 1234):
        self.entity_dict = entity_dict
        self.p = p
        self.rng = random.Random(seed)

    def __call__(self, chunks: List[Chunk]) -> List[Chunk]:
        out = []
        for surface, typ in chunks:
            cands = self.entity_dict.get(typ) if typ else None
            if typ and cands and self.rng.random() < self.p:
                out.append((self.rng.choice(cands), typ))
            else:
                out.append((surface, typ))
        return out


class SynonymReplace:
    """Replace non-entity chars with an embedding nearest-neighbour
    (reference SynomReplace :107-147 uses gensim most_similar top-5)."""

    def __init__(self, vocab: Sequence[str], embeddings: np.ndarray,
                 topn: int = 5, p: float = 0.2, seed: int = 1234):
        self.vocab = list(vocab)
        self.stoi = {w: i for i, w in enumerate(self.vocab)}
        emb = np.asarray(embeddings, dtype=np.float32)
        norm = np.linalg.norm(emb, axis=1, keepdims=True)
        self.emb = emb / np.clip(norm, 1e-8, None)
        self.topn = topn
        self.p = p
        self.rng = random.Random(seed)

    def most_similar(self, word: str) -> List[str]:
        i = self.stoi.get(word)
        if i is None:
            return []
        sims = self.emb @ self.emb[i]
        sims[i] = -1.0
        top = np.argpartition(-sims, self.topn)[:self.topn]
        return [self.vocab[j] for j in top[np.argsort(-sims[top])]]

    def __call__(self, chunks: List[Chunk]) -> List[Chunk]:
        out = []
        for surface, typ in chunks:
            if typ is not None:
                out.append((surface, typ))
                continue
            new = []
            for ch in surface:
                cands = (self.most_similar(ch)
                         if self.rng.random() < self.p else [])
                new.append(self.rng.choice(cands) if cands else ch)
            out.append(("".join(new), None))
        return out


class SentenceShuffle:
    """Swap comma-separated clauses (reference SentenceShuffle :150-180)."""

    SEPS = "，,；;"

    def __init__(self, seed: int = 1234):
        self.rng = random.Random(seed)

    def __call__(self, sent: Sequence[str], tags: Sequence[str]
                 ) -> Tuple[List[str], List[str]]:
        # split on separators keeping (chars, tags) clause pairs
        clauses, cur_s, cur_t = [], [], []
        for ch, tg in zip(sent, tags):
            cur_s.append(ch)
            cur_t.append(tg)
            if ch in self.SEPS:
                clauses.append((cur_s, cur_t))
                cur_s, cur_t = [], []
        if cur_s:
            clauses.append((cur_s, cur_t))
        if len(clauses) < 2:
            return list(sent), list(tags)
        body = clauses[:-1] if sent[-1] not in self.SEPS else clauses
        self.rng.shuffle(body)
        ordered = body + ([clauses[-1]] if body is not clauses else [])
        out_s, out_t = [], []
        for cs, ct in ordered:
            out_s.extend(cs)
            out_t.extend(ct)
        return out_s, out_t


def augment(data: str, data_dir: str, methods: Optional[Sequence[str]] = None,
            n_aug_per_sentence: int = 1, seed: int = 1234,
            entity_dict: Optional[Dict[str, List[str]]] = None,
            out_name: str = "train_augment.pkl") -> str:
    """Apply the selected augmenters to the train split and dump
    (sentences, tags) to {data_dir}/{data}/train_augment.pkl (reference
    augment() :183-236)."""
    methods = list(methods or ["entity_replace", "sentence_shuffle"])
    corpus_dir = os.path.join(data_dir, data)
    sentences, tags = load_data(data, corpus_dir, "train")
    if entity_dict is None and "entity_replace" in methods:
        entity_dict = build_entity_dict([data], data_dir)
    rng = random.Random(seed)
    ent = EntityReplace(entity_dict or {}, seed=seed)
    shuf = SentenceShuffle(seed=seed)
    aug_sents, aug_tags = [], []
    for sent, tag in zip(sentences, tags):
        for k in range(n_aug_per_sentence):
            method = rng.choice(methods)
            if method == "entity_replace":
                s, t = chunks_to_bio(ent(chunk_by_tag(sent, tag)))
            elif method == "sentence_shuffle":
                s, t = shuf(sent, tag)
            else:
                raise ValueError(f"unknown augment method {method}")
            if s != list(sent):
                aug_sents.append(s)
                aug_tags.append(t)
    os.makedirs(corpus_dir, exist_ok=True)
    path = os.path.join(corpus_dir, out_name)
    with open(path, "wb") as f:
        pickle.dump({"sentences": aug_sents, "tags": aug_tags}, f)
    return path
