"""MRC dataset: every sentence expands into one sample per entity-type
query — [CLS] query [SEP] text — with segment ids 0/1, BIO labels over
the text region only, and dynamic padded batching (reference
mrc/dataset.py:27-203; Tag2Query :12-16; padded_batch train / batch(1)
predict)."""
from __future__ import annotations

from typing import Dict, Iterator, List, Optional, Sequence

import numpy as np
import torch

from ..data.datasets import get_spec, load_data
from ..data.tokenizer import Vocab, WordpieceTokenizer
from ..models.mrc import MRC_LABELS, TAG2QUERY


def build_single_feature(tokenizer, query: str, text: str,
                         tags: Optional[Sequence[str]] = None,
                         tag_type: str = "", max_seq_len: int = 170) -> Dict:
    """One [CLS]+query+[SEP]+text sample (reference mrc/dataset.py:160-196).
    Labels: B/I where a span of `tag_type` sits, over the text region."""
    q_toks = tokenizer.tokenize(query)
    cls_id = tokenizer.vocab.stoi["[CLS]"]
    sep_id = tokenizer.vocab.stoi["[SEP]"]
    budget = max_seq_len - len(q_toks) - 2
    text = text[:budget]
    t_toks = tokenizer.tokenize(text)
    ids = ([cls_id] + tokenizer.convert_tokens_to_ids(q_toks) + [sep_id]
           + tokenizer.convert_tokens_to_ids(t_toks))
    query_len = len(q_toks) + 2  # CLS + query + SEP
    text_len = len(t_toks)
    n = len(ids)
    segment_ids = [0] * query_len + [1] * text_len
    text_mask = [0] * query_len + [1] * text_len
    label_ids = [0] * n
    if tags is not None:
        for i, tag in enumerate(list(tags)[:text_len]):
            if tag == f"B-{tag_type}":
                label_ids[query_len + i] = MRC_LABELS["B"]
            elif tag == f"I-{tag_type}":
                label_ids[query_len + i] = MRC_LABELS["I"]
    return {"token_ids": np.asarray(ids, dtype=np.int64),
            "segment_ids": np.asarray(segment_ids, dtype=np.int64),
            "mask": np.ones(n, dtype=np.int64),
            "text_mask": np.asarray(text_mask, dtype=np.int64),
            "label_ids": np.asarray(label_ids, dtype=np.int64),
            "query_len": np.int64(query_len),
            "text_len": np.int64(text_len),
            "tag_type": tag_type}


def _pad_stack(feats: List[Dict], keys) -> Dict[str, torch.Tensor]:
    """Dynamic padded batch (the reference's padded_batch with no fixed L)."""
    max_len = max(len(f["token_ids"]) for f in feats)
    out: Dict[str, torch.Tensor] = {}
    for k in keys:
        if np.isscalar(feats[0][k]) or feats[0][k].ndim == 0:
            out[k] = torch.as_tensor(np.stack([f[k] for f in feats]))
        else:
            arr = np.zeros((len(feats), max_len), dtype=np.int64)
            for i, f in enumerate(feats):
                arr[i, :len(f[k])] = f[k]
            out[k] = torch.from_numpy(arr)
    return out


class MrcDataset:
    """Generator-style pipeline over (sentence × tag queries); works off a
    real corpus dir or the synthetic corpus (no network)."""

    def __init__(self, data_dir: str, data: str = "msra", batch_size: int = 32,
                 max_seq_len: int = 170, tokenizer=None,
                 tag2query: Optional[Dict[str, str]] = None):
        self.data_dir = data_dir
        self.data = data
        self.spec = get_spec(data)
        self.batch_size = batch_size
        self.max_seq_len = max_seq_len
        self.tokenizer = tokenizer or WordpieceTokenizer(Vocab.synthetic())
        self.tag2query = tag2query or {
            t: TAG2QUERY.get(t, f"找出{t}") for t in self.spec.entity_types}

    def build_features(self, split: str) -> List[Dict]:
        sentences, tags = load_data(self.data, self.data_dir, split)
        feats = []
        for sent, tag in zip(sentences, tags):
            text = "".join(sent)
            for tag_type, query in self.tag2query.items():
                feats.append(build_single_feature(
                    self.tokenizer, query, text, tag, tag_type,
                    self.max_seq_len))
        return feats

    @property
    def params(self) -> Dict:
        n = self.spec.n_train * len(self.tag2query)
        step = max(1, n // self.batch_size)
        return {"n_sample": n, "step_per_epoch": step,
                "label_size": 3, "max_seq_len": self.max_seq_len,
                "vocab_size": len(self.tokenizer.vocab)}

    def iter_batches(self, split: str = "train", shuffle: bool = True,
                     seed: int = 1234, batch_size: Optional[int] = None
                     ) -> Iterator[Dict[str, torch.Tensor]]:
        feats = self.build_features(split)
        order = np.arange(len(feats))
        if shuffle:
            np.random.default_rng(seed).shuffle(order)
        bs = batch_size or self.batch_size
        keys = ["token_ids", "segment_ids", "mask", "text_mask", "label_ids",
                "query_len", "text_len"]
        for i in range(0, len(order), bs):
            chunk = [feats[j] for j in order[i:i + bs]]
            yield _pad_stack(chunk, keys)
