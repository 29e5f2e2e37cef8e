"""Character trie for lexicon lookup.

Capability parity with the reference's data/trie.py:3-51, but wired into
the main word-enhance path (the reference left it unused and did O(L*W)
substring scans instead — data/word_enhance.py:262-299); the trie makes
ex-softword/softlexicon preprocessing O(L * max_word_len).
"""
from __future__ import annotations

from typing import Dict, Iterable, List, Tuple


class Trie:
    __slots__ = ("root",)

    def __init__(self, words: Iterable[str] = ()):
        self.root: Dict = {}
        for w in words:
            self.insert(w)

    def insert(self, word: str) -> None:
        node = self.root
        for ch in word:
            node = node.setdefault(ch, {})
        node[""] = True  # terminal marker

    def __contains__(self, word: str) -> bool:
        node = self.root
        for ch in word:
            node = node.get(ch)
            if node is None:
                return False
        return "" in node

    def prefixes(self, text: str, start: int, max_len: int = 10) -> List[str]:
        """All lexicon words beginning at text[start], longest capped at max_len."""
        node = self.root
        out: List[str] = []
        end = min(len(text), start + max_len)
        for i in range(start, end):
            node = node.get(text[i])
            if node is None:
                break
            if "" in node:
                out.append(text[start:i + 1])
        return out

    def max_match_segment(self, text: str, max_len: int = 10) -> List[Tuple[int, int]]:
        """Forward maximum-matching segmentation -> [(start, end)) spans.

        Replaces the reference's jieba dependency (data/word_enhance.py:235-259)
        with a self-contained lexicon segmenter; unmatched chars become
        single-char spans.
        """
        spans: List[Tuple[int, int]] = []
        i = 0
        n = len(text)
        while i < n:
            words = self.prefixes(text, i, max_len)
            if words:
                j = i + len(words[-1])
            else:
                j = i + 1
            spans.append((i, j))
            i = j
        return spans
