"""Pretrained embedding file utilities (reference
pretrain_model/glove_2_wv.py:10-21 + pretrain_model/lattice/preprocess.py
:10-35, without gensim): load glove/word2vec text formats into
(vocab, matrix) and merge char+word tables for lattice-style models."""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np


def load_text_embeddings(path: str, max_words: Optional[int] = None,
                         encoding: str = "utf-8"
                         ) -> Tuple[List[str], np.ndarray]:
    """Reads glove (no header) or word2vec (count dim header) text files."""
    vocab: List[str] = []
    rows: List[np.ndarray] = []
    dim = None
    with open(path, encoding=encoding, errors="ignore") as f:
        first = f.readline().rstrip("\n")
        parts = first.split(" ")
        if len(parts) == 2 and all(p.isdigit() for p in parts):
            pass  # word2vec header — skip
        elif len(parts) > 2:
            vocab.append(parts[0])
            rows.append(np.asarray(parts[1:], dtype=np.float32))
            dim = len(parts) - 1
        for line in f:
            parts = line.rstrip("\n").split(" ")
            if len(parts) < 3:
                continue
            vec = np.asarray(parts[1:], dtype=np.float32)
            if dim is None:
                dim = len(vec)
            if len(vec) != dim:
                continue
            vocab.append(parts[0])
            rows.append(vec)
            if max_words and len(vocab) >= max_words:
                break
    return vocab, np.stack(rows) if rows else np.zeros((0, dim or 0), np.float32)


def normalize_rows(matrix: np.ndarray) -> np.ndarray:
    """L2-normalize (reference VocabModel normalized embedding matrix,
    data/word_enhance.py:36-81)."""
    norm = np.linalg.norm(matrix, axis=1, keepdims=True)
    return matrix / np.clip(norm, 1e-8, None)


def add_special_tokens(vocab: List[str], matrix: np.ndarray,
                       specials: Tuple[str, ...] = ("<None>", "<PAD>", "<eos>"),
                       seed: int = 1234) -> Tuple[List[str], np.ndarray]:
    """Prepend special tokens with small random rows (reference addon
    tokens, data/word_enhance.py:44-58)."""
    rng = np.random.default_rng(seed)
    dim = matrix.shape[1]
    extra = rng.normal(scale=0.01, size=(len(specials), dim)).astype(np.float32)
    extra[list(specials).index("<PAD>") if "<PAD>" in specials else 1] = 0.0
    return list(specials) + vocab, np.concatenate([extra, matrix], axis=0)


def combine_embeddings(char_vocab: List[str], char_mat: np.ndarray,
                       word_vocab: List[str], word_mat: np.ndarray
                       ) -> Tuple[List[str], np.ndarray]:
    """Merge char + word tables, chars first, words that are not single
    chars appended (reference combine_w2v for lattice,
    pretrain_model/lattice/preprocess.py:10-35)."""
    assert char_mat.shape[1] == word_mat.shape[1], "dim mismatch"
    seen: Dict[str, int] = {w: i for i, w in enumerate(char_vocab)}
    out_vocab = list(char_vocab)
    rows = [char_mat]
    keep = [i for i, w in enumerate(word_vocab) if w not in seen]
    out_vocab.extend(word_vocab[i] for i in keep)
    rows.append(word_mat[keep])
    return out_vocab, np.concatenate(rows, axis=0)
