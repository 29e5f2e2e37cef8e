"""Word-enhance lexicon features: softword / ex_softword / softlexicon / bichar.

Re-implements the four feature families of the reference's
data/word_enhance.py (Soft2Idx :18-24, build_softword :235-259,
build_ex_softword :262-299, build_soft_lexicon :302-337,
postproc_soft_lexicon :163-205) without the jieba/gensim dependencies.

Segmentation for softword: the reference segments with ``jieba.cut``
(word_enhance.py:244). jieba's core algorithm is a prefix-dictionary DAG
with a max-log-probability route DP over word frequencies, plus an HMM
pass over leftover single-char runs. ``Lexicon.max_prob_segment``
re-implements the DAG + max-probability route exactly (= jieba
``cut(HMM=False)`` semantics) over this lexicon's frequencies; the HMM
OOV pass is intentionally absent (it requires jieba's trained emission
tables, unavailable offline — the delta is measured in
profiles/segmentation_delta_r02.md). Forward maximum matching remains
available as ``segmenter="maxmatch"``. Lexicon vectors come from an
``npz`` file or a deterministic synthetic builder (no network).
"""
from __future__ import annotations

import hashlib
from dataclasses import dataclass, field
from typing import Dict, List, Sequence, Tuple

import numpy as np

from .trie import Trie

# BMES(+None) coding, reference Soft2Idx (word_enhance.py:18-24).
SOFT2IDX: Dict[str, int] = {"None": 0, "B": 1, "M": 2, "E": 3, "S": 4}
NUM_SOFT = len(SOFT2IDX)           # 5
MAX_WORD_LEN = 10                  # reference MaxWordLen (word_enhance.py:262-299)
MAX_LEXICON_LEN = 10               # reference MaxLexiconLen (postproc :163-205)
NUM_ROLES = 4                      # B/M/E/S groups for softlexicon
WORD_ENHANCE_METHODS = ("softword", "ex_softword", "softlexicon", "bichar")

PAD_WORD, NONE_WORD, EOS_WORD = "<PAD>", "<None>", "<eos>"


@dataclass
class Lexicon:
    """Word lexicon with frequencies + embedding matrix.

    Mirrors the reference's VocabModel gensim wrapper (word_enhance.py:36-81):
    rows 0/1/2 are <PAD>/<None>/<eos> addon tokens, embeddings L2-normalized.
    """
    words: List[str]
    freq: np.ndarray                  # [n_words] float
    embedding: np.ndarray             # [n_words, dim] float32, L2-normalized
    word2idx: Dict[str, int] = field(init=False)
    trie: Trie = field(init=False)

    def __post_init__(self):
        self.word2idx = {w: i for i, w in enumerate(self.words)}
        self.trie = Trie(w for w in self.words if len(w) >= 2)
        self._logtotal = float(np.log(max(float(self.freq.sum()), 1.0)))

    def _word_logp(self, w: str) -> float:
        """log p(word) with jieba's OOV convention: unseen words count
        frequency 1 (jieba calc(): log(FREQ.get(word) or 1) - logtotal)."""
        i = self.word2idx.get(w)
        f = float(self.freq[i]) if i is not None else 1.0
        return float(np.log(max(f, 1.0))) - self._logtotal

    def max_prob_segment(self, text: str,
                         max_len: int = MAX_WORD_LEN
                         ) -> List[Tuple[int, int]]:
        """jieba ``cut(HMM=False)`` core: prefix-dict DAG + right-to-left
        max-log-probability route DP. Ties prefer the longer word (jieba
        max() over (score, end) tuples). Returns [(start, end)) spans."""
        n = len(text)
        route = [0.0] * (n + 1)
        nxt = [0] * n
        for i in range(n - 1, -1, -1):
            best_score, best_j = None, i + 1
            cands = self.trie.prefixes(text, i, max_len)
            cands.append(text[i])          # single char (in-dict or OOV)
            for w in cands:
                j = i + len(w)
                score = self._word_logp(w) + route[j]
                if (best_score is None or score > best_score
                        or (score == best_score and j > best_j)):
                    best_score, best_j = score, j
            route[i] = best_score
            nxt[i] = best_j
        spans: List[Tuple[int, int]] = []
        i = 0
        while i < n:
            spans.append((i, nxt[i]))
            i = nxt[i]
        return spans

    @property
    def pad_id(self) -> int:
        return 0

    @property
    def none_id(self) -> int:
        return 1

    @property
    def dim(self) -> int:
        return self.embedding.shape[1]

    def __len__(self) -> int:
        return len(self.words)

    @classmethod
    def from_npz(cls, path: str) -> "Lexicon":
        z = np.load(path, allow_pickle=True)
        return cls(list(z["words"]), z["freq"].astype(np.float64),
                   z["embedding"].astype(np.float32))

    @classmethod
    def synthetic(cls, vocab_chars: Sequence[str], n_words: int = 5000,
                  dim: int = 50, seed: int = 1234) -> "Lexicon":
        """Deterministic lexicon of 2-4 char words over the given char set."""
        rng = np.random.default_rng(seed)
        chars = [c for c in vocab_chars if len(c) == 1 and not c.startswith("[")]
        words = [PAD_WORD, NONE_WORD, EOS_WORD]
        seen = set(words)
        while len(words) < n_words + 3:
            ln = int(rng.integers(2, 5))
            w = "".join(rng.choice(chars, size=ln))
            if w not in seen:
                seen.add(w)
                words.append(w)
        freq = rng.zipf(1.5, size=len(words)).astype(np.float64)
        emb = rng.standard_normal((len(words), dim)).astype(np.float32)
        emb /= np.linalg.norm(emb, axis=1, keepdims=True) + 1e-8
        return cls(words, freq, emb)


def build_softword(sentence: str, lexicon: Lexicon,
                   segmenter: str = "maxprob") -> List[int]:
    """Per-char BMES segmentation ids (reference build_softword :235-259).

    segmenter="maxprob" (default) replicates jieba's DAG +
    max-probability route (the reference's jieba.cut semantics minus the
    HMM OOV pass); "maxmatch" keeps forward maximum matching. Single
    chars -> S.
    """
    if segmenter == "maxprob":
        spans = lexicon.max_prob_segment(sentence, MAX_WORD_LEN)
    else:
        spans = lexicon.trie.max_match_segment(sentence, MAX_WORD_LEN)
    ids: List[int] = [0] * len(sentence)
    for start, end in spans:
        if end - start == 1:
            ids[start] = SOFT2IDX["S"]
        else:
            ids[start] = SOFT2IDX["B"]
            for i in range(start + 1, end - 1):
                ids[i] = SOFT2IDX["M"]
            ids[end - 1] = SOFT2IDX["E"]
    return ids


def build_ex_softword(sentence: str, lexicon: Lexicon) -> List[List[int]]:
    """Per-char multi-hot [len, 5] over BMES of ALL matching lexicon words
    (reference build_ex_softword :262-299)."""
    n = len(sentence)
    multihot = [[0] * NUM_SOFT for _ in range(n)]
    for start in range(n):
        for w in lexicon.trie.prefixes(sentence, start, MAX_WORD_LEN):
            if len(w) < 2:
                continue
            end = start + len(w) - 1
            multihot[start][SOFT2IDX["B"]] = 1
            for i in range(start + 1, end):
                multihot[i][SOFT2IDX["M"]] = 1
            multihot[end][SOFT2IDX["E"]] = 1
    for i in range(n):
        if sum(multihot[i]) == 0:
            multihot[i][SOFT2IDX["None"]] = 1
    return multihot


def build_soft_lexicon(sentence: str, lexicon: Lexicon
                       ) -> Tuple[np.ndarray, np.ndarray]:
    """SoftLexicon ids+weights (reference build_soft_lexicon :302-337 +
    postproc_soft_lexicon :163-205).

    Returns ids [len, 4*10] int32, weights [len, 4*10] float32.
    Per char, per role B/M/E/S: the lexicon word-ids whose match places the
    char in that role, truncated to the MAX_LEXICON_LEN most frequent,
    padded with <PAD>; weights are word frequencies normalized across all
    40 slots of the char (reference normalizes across the 4 sets jointly).
    """
    n = len(sentence)
    roles: List[List[List[int]]] = [[[] for _ in range(NUM_ROLES)] for _ in range(n)]
    for start in range(n):
        for w in lexicon.trie.prefixes(sentence, start, MAX_WORD_LEN):
            if len(w) < 2:
                continue
            wid = lexicon.word2idx[w]
            end = start + len(w) - 1
            roles[start][0].append(wid)               # B
            for i in range(start + 1, end):
                roles[i][1].append(wid)               # M
            roles[end][2].append(wid)                 # E
    # role S: char itself if it is a lexicon word
    for i, ch in enumerate(sentence):
        wid = lexicon.word2idx.get(ch)
        if wid is not None:
            roles[i][3].append(wid)

    ids = np.zeros((n, NUM_ROLES * MAX_LEXICON_LEN), dtype=np.int32)
    weights = np.zeros((n, NUM_ROLES * MAX_LEXICON_LEN), dtype=np.float32)
    for i in range(n):
        for r in range(NUM_ROLES):
            cand = roles[i][r]
            if not cand:
                cand = [lexicon.none_id]
            cand = sorted(cand, key=lambda w: -lexicon.freq[w])[:MAX_LEXICON_LEN]
            base = r * MAX_LEXICON_LEN
            for k, wid in enumerate(cand):
                ids[i, base + k] = wid
                weights[i, base + k] = lexicon.freq[wid]
        total = weights[i].sum()
        if total > 0:
            weights[i] /= total
    return ids, weights


def build_bichar(tokens: List[str]) -> List[str]:
    """Bichar tokens: char_i + char_{i+1}, last pads with <eos>
    (reference BicharProc, base_preprocess.py:260-296)."""
    out = []
    for i in range(len(tokens)):
        nxt = tokens[i + 1] if i + 1 < len(tokens) else EOS_WORD
        out.append(tokens[i] + (nxt if nxt != EOS_WORD else EOS_WORD))
    return out


def bichar_vocab_id(bichar: str, vocab_size: int = 50000) -> int:
    """Stable hash id for bichar embeddings when no pretrained bichar table
    is available (ids 0/1 reserved for pad/unk)."""
    h = int(hashlib.md5(bichar.encode("utf-8")).hexdigest()[:8], 16)
    return 2 + h % (vocab_size - 2)
