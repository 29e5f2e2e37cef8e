"""L1 feature/preprocess layer: sentence -> padded feature dict, cached npz.

Parity with the reference's data/base_preprocess.py: ``get_instance``
(:74-91) returns a BasicProc / SoftwordProc / ExSoftwordProc /
SoftLexiconProc / BicharProc; each has ``init_data`` + ``dump_cache``
(tfrecord -> npz here) and the online-inference-shared
``build_seq_feature`` (:176-187). The cache carries
``data_params`` (n_sample, max_seq_len, label_size, tag2idx, idx2tag and
embedding matrices) like the reference's data_params.pkl (:212-253).
"""
from __future__ import annotations

import os
import pickle
from typing import Dict, List, Optional, Sequence

import numpy as np

from . import word_enhance as we
from .datasets import DatasetSpec, get_spec, load_data, synthetic_corpus
from .tokenizer import (CharTokenizer, Vocab, WordpieceTokenizer, get_tokenizer,
                        tokenizer_type_from_model)
from .word_enhance import Lexicon

# longest suffix first: '_ex_softword' must win over '_softword'
MODEL_SUFFIX2ENHANCE = {"_ex_softword": "ex_softword", "_softword": "softword",
                        "_softlexicon": "softlexicon", "_bichar": "bichar"}


def extract_prefix_surfix(model_name: str):
    """Split model name into (word_enhance, tokenizer_type) by the reference
    naming convention (base_preprocess.py:22-33): 'bert' prefix -> wordpiece
    tokenizer; '_softword|_ex_softword|_softlexicon|_bichar' suffix -> feature.
    """
    enhance = None
    for suffix, name in MODEL_SUFFIX2ENHANCE.items():
        if model_name.endswith(suffix):
            enhance = name
            break
    return enhance, tokenizer_type_from_model(model_name)


class BasicProc:
    """Tokenize -> [CLS]...[SEP] (bert) / bare (char) -> pad to max_seq_len."""

    word_enhance: Optional[str] = None

    def __init__(self, tokenizer, max_seq_len: int, tag2idx: Dict[str, int],
                 lexicon: Optional[Lexicon] = None):
        self.tokenizer = tokenizer
        self.max_seq_len = max_seq_len
        self.tag2idx = tag2idx
        self.idx2tag = {v: k for k, v in tag2idx.items()}
        self.lexicon = lexicon
        self.is_bert = isinstance(tokenizer, WordpieceTokenizer)

    # ---- shared online/offline feature builder (reference :176-187) ----
    def build_seq_feature(self, sentence: str,
                          labels: Optional[Sequence[str]] = None) -> Dict[str, np.ndarray]:
        L = self.max_seq_len
        body = L - 2 if self.is_bert else L
        tokens = self.tokenizer.tokenize(sentence)[:body]
        n = len(tokens)
        ids = self.tokenizer.convert_tokens_to_ids(tokens)
        if labels is not None:
            labels = list(labels)[:body]
            assert len(labels) == n, (
                f"token/label length mismatch {n} vs {len(labels)}")
            lab_ids = [self.tag2idx[t] for t in labels]
        else:
            lab_ids = [self.tag2idx.get("O", 0)] * n

        pad_tag = self.tag2idx["[PAD]"]
        if self.is_bert:
            tok = ([self.tokenizer.cls_id] + ids + [self.tokenizer.sep_id])
            lab = ([self.tag2idx["[CLS]"]] + lab_ids + [self.tag2idx["[SEP]"]])
            seq_len = n + 2
        else:
            tok, lab, seq_len = ids, lab_ids, n
        pad_id = self.tokenizer.pad_id
        tok = tok + [pad_id] * (L - len(tok))
        lab = lab + [pad_tag] * (L - len(lab))
        mask = [1] * seq_len + [0] * (L - seq_len)
        feat = {
            "token_ids": np.asarray(tok, dtype=np.int32),
            "label_ids": np.asarray(lab, dtype=np.int32),
            "mask": np.asarray(mask, dtype=np.int32),
            "seq_len": np.int32(seq_len),
        }
        self._add_enhance(feat, sentence[:body])
        return feat

    def _add_enhance(self, feat: Dict[str, np.ndarray], sentence: str) -> None:
        pass

    def _shift_pad(self, rows: np.ndarray, fill=0) -> np.ndarray:
        """Align per-raw-char features to the padded token frame: insert a
        CLS row for bert, pad to max_seq_len (reference alignment rules,
        word_enhance.py:89-160 — per-char wordpieces make this a shift)."""
        L = self.max_seq_len
        tail_shape = rows.shape[1:]
        out = np.full((L,) + tail_shape, fill, dtype=rows.dtype)
        off = 1 if self.is_bert else 0
        n = min(len(rows), L - off - (1 if self.is_bert else 0))
        out[off:off + n] = rows[:n]
        return out

    # ----------------------------- offline: corpus -> cached features ----
    def init_data(self, name: str, data_dir: str, split: str):
        sentences, tags = load_data(name, data_dir, split)
        feats, raws, dropped = [], [], 0
        for sent, t in zip(sentences, tags):
            try:
                feats.append(self.build_seq_feature(sent, t))
                raws.append(sent)
            except (AssertionError, KeyError):
                dropped += 1   # skip-and-count like reference :243-246
        return feats, raws, dropped

    def stack(self, feats: List[Dict[str, np.ndarray]]) -> Dict[str, np.ndarray]:
        return {k: np.stack([f[k] for f in feats]) for k in feats[0]}


class SoftwordProc(BasicProc):
    word_enhance = "softword"

    def _add_enhance(self, feat, sentence):
        ids = np.asarray(we.build_softword(sentence, self.lexicon), dtype=np.int32)
        feat["softword_ids"] = self._shift_pad(ids)


class ExSoftwordProc(BasicProc):
    word_enhance = "ex_softword"

    def _add_enhance(self, feat, sentence):
        hot = np.asarray(we.build_ex_softword(sentence, self.lexicon), dtype=np.float32)
        feat["ex_softword_ids"] = self._shift_pad(hot)


class SoftLexiconProc(BasicProc):
    word_enhance = "softlexicon"

    def _add_enhance(self, feat, sentence):
        ids, wts = we.build_soft_lexicon(sentence, self.lexicon)
        feat["softlexicon_ids"] = self._shift_pad(ids)
        feat["softlexicon_weights"] = self._shift_pad(wts)


class BicharProc(BasicProc):
    word_enhance = "bichar"

    def _add_enhance(self, feat, sentence):
        bichars = we.build_bichar(list(sentence))
        ids = np.asarray([we.bichar_vocab_id(b) for b in bichars], dtype=np.int32)
        feat["bichar_ids"] = self._shift_pad(ids)


_PROCS = {None: BasicProc, "softword": SoftwordProc, "ex_softword": ExSoftwordProc,
          "softlexicon": SoftLexiconProc, "bichar": BicharProc}


def get_instance(tokenizer_type: str, max_seq_len: int, tag2idx: Dict[str, int],
                 word_enhance: Optional[str] = None,
                 vocab: Optional[Vocab] = None,
                 lexicon: Optional[Lexicon] = None) -> BasicProc:
    """Factory mirroring reference get_instance (base_preprocess.py:74-91)."""
    tokenizer = get_tokenizer(tokenizer_type, vocab=vocab)
    if word_enhance is not None and lexicon is None:
        lexicon = Lexicon.synthetic(tokenizer.vocab.itos)
    cls = _PROCS[word_enhance]
    return cls(tokenizer, max_seq_len, tag2idx, lexicon)


# ------------------------------------------------------------- cache API

def cache_name(tokenizer_type: str, split: str, word_enhance: Optional[str]) -> str:
    """{tokenizer}_{split}[_{word_enhance}] like the reference tfrecord names."""
    parts = [tokenizer_type, split] + ([word_enhance] if word_enhance else [])
    return "_".join(parts)


def build_cache(name: str, data_dir: str, model_name: str, splits=("train", "valid", "test"),
                spec: Optional[DatasetSpec] = None, force: bool = False) -> Dict:
    """Preprocess corpus splits to {cache_name}.npz + data_params.pkl."""
    spec = spec or get_spec(name)
    enhance, tok_type = extract_prefix_surfix(model_name)
    proc = get_instance(tok_type, spec.max_seq_len, spec.tag2idx, enhance)
    os.makedirs(data_dir, exist_ok=True)

    n_train = 0
    for split in splits:
        path = os.path.join(data_dir, cache_name(tok_type, split, enhance) + ".npz")
        if os.path.exists(path) and not force:
            if split == "train":
                n_train = int(np.load(path)["token_ids"].shape[0])
            continue
        feats, raws, dropped = proc.init_data(name, data_dir, split)
        arrays = proc.stack(feats)
        arrays["raw"] = np.asarray(raws, dtype=object)
        np.savez_compressed(path, **arrays)
        if split == "train":
            n_train = len(feats)

    params = {
        "n_sample": n_train,
        "max_seq_len": spec.max_seq_len,
        "label_size": spec.label_size,
        "tag2idx": spec.tag2idx,
        "idx2tag": spec.idx2tag,
        "word_enhance": enhance,
        "tokenizer_type": tok_type,
        "vocab_size": len(proc.tokenizer.vocab),
    }
    if tok_type != "bert":
        # char embedding matrix travels inside params like the reference's
        # giga matrix (base_preprocess.py:223-226); synthetic here.
        rng = np.random.default_rng(1234)
        params["embedding"] = rng.standard_normal(
            (len(proc.tokenizer.vocab), 50)).astype(np.float32) * 0.1
        params["embedding_dim"] = 50
    if enhance == "softlexicon":
        params["word_embedding"] = proc.lexicon.embedding
        params["word_vocab_size"] = len(proc.lexicon)
        params["word_dim"] = proc.lexicon.dim
    if enhance == "bichar":
        params["bichar_vocab_size"] = 50000
        params["bichar_dim"] = 50
    pkl = os.path.join(data_dir, f"{cache_name(tok_type, 'data', enhance)}_params.pkl")
    with open(pkl, "wb") as f:
        pickle.dump(params, f)
    return params
