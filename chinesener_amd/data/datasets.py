"""Corpus adapters + synthetic corpus generator.

Each adapter mirrors a reference per-dataset preprocess.py: a
``load_data(data_dir, file_name) -> (sentences, tags)`` loader plus
TAG2IDX / MAPPING / MAX_SEQ_LEN constants (e.g. reference
data/msra/preprocess.py:7-48, data/people_daily/preprocess.py:28-63,
data/msr/preprocess.py:7-66, data/weibo/preprocess.py:28-61,
data/cluener/preprocess.py:20-66).

Because this environment has no network (no real corpora), every dataset
also has a deterministic synthetic generator producing corpus-shaped
data (same tag scheme, sentence-length distribution and sample counts
anchored to BASELINE.md) for tests and benchmarks.
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass
from typing import Dict, List, Tuple

import numpy as np

Sentences = List[str]
Tags = List[List[str]]

PAD_TAG, CLS_TAG, SEP_TAG = "[PAD]", "[CLS]", "[SEP]"


def _bio_tagset(types: List[str]) -> Dict[str, int]:
    tags = [PAD_TAG, "O"]
    for t in types:
        tags += [f"B-{t}", f"I-{t}"]
    tags += [CLS_TAG, SEP_TAG]
    return {t: i for i, t in enumerate(tags)}


@dataclass
class DatasetSpec:
    name: str
    tag2idx: Dict[str, int]
    max_seq_len: int
    entity_types: List[str]
    n_train: int
    n_valid: int
    n_test: int
    avg_len: int = 45
    scheme: str = "bio"     # 'bio' NER or 'bies' CWS

    @property
    def idx2tag(self) -> Dict[int, str]:
        return {v: k for k, v in self.tag2idx.items()}

    @property
    def label_size(self) -> int:
        return len(self.tag2idx)


# Sample counts anchor to the real corpora (BASELINE.md dataset table).
DATASETS: Dict[str, DatasetSpec] = {
    "msra": DatasetSpec("msra", _bio_tagset(["LOC", "PER", "ORG"]), 150,
                        ["LOC", "PER", "ORG"], 42000, 3000, 3442),
    "people_daily": DatasetSpec("people_daily", _bio_tagset(["LOC", "PER", "ORG"]),
                                150, ["LOC", "PER", "ORG"], 20865, 2318, 4636),
    "weibo": DatasetSpec("weibo", _bio_tagset(["PER", "LOC", "ORG", "GPE"]), 150,
                         ["PER", "LOC", "ORG", "GPE"], 1350, 270, 270, avg_len=30),
    "cluener": DatasetSpec("cluener", _bio_tagset(["LOC", "PER", "ORG"]), 150,
                           ["LOC", "PER", "ORG"], 10748, 1343, 1345),
    "msr": DatasetSpec("msr", {t: i for i, t in enumerate(
        [PAD_TAG, "B", "M", "E", "S", CLS_TAG, SEP_TAG])}, 150,
        [], 86918, 4000, 3985, scheme="bies"),
    # virtual dataset: train split replaced by the augmentation dump
    # (reference data/people_daily_augment/preprocess.py:8-20)
    "people_daily_augment": DatasetSpec(
        "people_daily_augment", _bio_tagset(["LOC", "PER", "ORG"]), 150,
        ["LOC", "PER", "ORG"], 20865, 2318, 4636),
}


def get_spec(name: str) -> DatasetSpec:
    if name not in DATASETS:
        raise KeyError(f"unknown dataset '{name}' (known: {sorted(DATASETS)})")
    return DATASETS[name]


# ---------------------------------------------------------------- loaders

def load_sentence_tag_dirs(data_dir: str, split: str) -> Tuple[Sentences, Tags]:
    """MSRA layout: {split}/sentences.txt + {split}/tags.txt, space-separated
    (reference data/msra/preprocess.py:7-35)."""
    with open(os.path.join(data_dir, split, "sentences.txt"), encoding="utf-8") as f:
        sentences = ["".join(line.split()) for line in f if line.strip()]
    with open(os.path.join(data_dir, split, "tags.txt"), encoding="utf-8") as f:
        tags = [line.split() for line in f if line.strip()]
    return sentences, tags


def load_conll(data_dir: str, file_name: str) -> Tuple[Sentences, Tags]:
    """CoNLL-ish 'char tag' lines, blank line between sentences
    (reference data/people_daily/preprocess.py:28-63)."""
    sentences, tags = [], []
    chars: List[str] = []
    labels: List[str] = []
    with open(os.path.join(data_dir, file_name), encoding="utf-8") as f:
        for line in f:
            line = line.rstrip("\n")
            if not line:
                if chars:
                    sentences.append("".join(chars))
                    tags.append(labels)
                    chars, labels = [], []
                continue
            parts = line.split()
            if len(parts) >= 2:
                chars.append(parts[0])
                labels.append(parts[-1])
    if chars:
        sentences.append("".join(chars))
        tags.append(labels)
    return sentences, tags


def load_cluener_json(data_dir: str, file_name: str,
                      mapping: Dict[str, str] | None = None) -> Tuple[Sentences, Tags]:
    """CLUENER jsonl {text, label:{type:{surface:[[s,e],...]}}} -> BIO
    (reference data/cluener/preprocess.py:20-66)."""
    mapping = mapping or {"address": "LOC", "name": "PER", "company": "ORG",
                          "government": "ORG", "organization": "ORG"}
    sentences, tags = [], []
    with open(os.path.join(data_dir, file_name), encoding="utf-8") as f:
        for line in f:
            if not line.strip():
                continue
            rec = json.loads(line)
            text = rec["text"]
            labels = ["O"] * len(text)
            for etype, surf2spans in rec.get("label", {}).items():
                mapped = mapping.get(etype)
                if mapped is None:
                    continue
                for spans in surf2spans.values():
                    for s, e in spans:
                        labels[s] = f"B-{mapped}"
                        for i in range(s + 1, e + 1):
                            labels[i] = f"I-{mapped}"
            sentences.append(text)
            tags.append(labels)
    return sentences, tags


def load_data(name: str, data_dir: str, split: str) -> Tuple[Sentences, Tags]:
    """Dispatch on dataset layout; falls back to synthetic when files absent."""
    spec = get_spec(name)
    try:
        if name == "people_daily_augment":
            # train comes from train_augment.pkl (augment.py dump);
            # other splits fall through to the base corpus
            base_dir = data_dir.replace("_augment", "")
            if split == "train":
                import pickle
                with open(os.path.join(data_dir, "train_augment.pkl"),
                          "rb") as f:
                    blob = pickle.load(f)
                if isinstance(blob, dict):
                    return blob["sentences"], blob["tags"]
                return blob      # (sentences, tags) tuple form
            return load_conll(base_dir, f"example.{split}")
        if name == "msra":
            return load_sentence_tag_dirs(data_dir, split)
        if name == "cluener":
            return load_cluener_json(data_dir, f"{split}.json")
        return load_conll(data_dir, f"example.{split}" if name == "people_daily"
                          else f"{split}.txt")
    except FileNotFoundError:
        return synthetic_corpus(spec, split)


# ------------------------------------------------------- synthetic corpus

_SYNTH_CACHE: Dict[Tuple[str, str], Tuple[Sentences, Tags]] = {}


def synthetic_corpus(spec: DatasetSpec, split: str = "train",
                     n: int | None = None, seed: int = 1234
                     ) -> Tuple[Sentences, Tags]:
    """Deterministic corpus-shaped random data (CJK chars, BIO/BIES tags)."""
    key = (spec.name, split)
    if n is None and key in _SYNTH_CACHE:
        return _SYNTH_CACHE[key]
    # Train-size cap: CPU tests keep the default small; the shipped
    # data/ caches are built with CHINESENER_SYNTH_TRAIN=8000 so the
    # epoch-derived LR schedule is long enough for from-scratch (random
    # init) convergence — see profiles/convergence_r01.md.
    cap = int(os.environ.get("CHINESENER_SYNTH_TRAIN", "2000"))
    count = n if n is not None else {
        "train": min(spec.n_train, cap),
        "valid": min(spec.n_valid, 200),
        "test": min(spec.n_test, 200),
        "predict": min(spec.n_test, 200),
    }.get(split, 200)
    rng = np.random.default_rng(seed + hash(key) % 10000)
    chars = [chr(0x4E00 + i) for i in range(3000)]
    # learnable structure: each entity type draws its characters from a
    # dedicated disjoint char range, so char->tag is actually learnable
    # (convergence smokes train on this and check entity F1 rises)
    type_chars = {t: [chr(0x4E00 + 3000 + 200 * i + j) for j in range(200)]
                  for i, t in enumerate(spec.entity_types)}
    sentences, tags = [], []
    for _ in range(count):
        ln = int(np.clip(rng.poisson(spec.avg_len), 4, spec.max_seq_len - 2))
        sent = "".join(rng.choice(chars, size=ln))
        if spec.scheme == "bies":
            labels: List[str] = []
            i = 0
            while i < ln:
                w = int(rng.integers(1, 5))
                w = min(w, ln - i)
                if w == 1:
                    labels.append("S")
                else:
                    labels += ["B"] + ["M"] * (w - 2) + ["E"]
                i += w
        else:
            labels = ["O"] * ln
            n_ent = int(rng.integers(0, max(2, ln // 15) + 1))
            for _ in range(n_ent):
                etype = rng.choice(spec.entity_types)
                elen = int(rng.integers(2, 5))
                start = int(rng.integers(0, max(1, ln - elen)))
                if any(labels[j] != "O" for j in range(start, start + elen)):
                    continue
                labels[start] = f"B-{etype}"
                for j in range(start + 1, start + elen):
                    labels[j] = f"I-{etype}"
                # rewrite the span's chars from the type's char range
                sl = list(sent)
                for j in range(start, start + elen):
                    sl[j] = type_chars[str(etype)][int(rng.integers(0, 200))]
                sent = "".join(sl)
        sentences.append(sent)
        tags.append(labels)
    if n is None:
        _SYNTH_CACHE[key] = (sentences, tags)
    return sentences, tags
