"""Prediction post-processing: strip special tokens, ids→tags, BIO→entity
decode (reference tools/predict_utils.py:6-60 and tools/infer_utils.py:76-118
behaviour, re-implemented)."""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

SPECIAL_TAGS = ("[PAD]", "[CLS]", "[SEP]")


def fix_tokens(tokens: Sequence[str]) -> List[str]:
    """Clean wordpiece artifacts for display: strip '##' continuation
    prefixes and map '[UNK]' to a placeholder char (reference
    tools/infer_utils.py:102-118)."""
    out = []
    for tok in tokens:
        if tok.startswith("##"):
            tok = tok[2:]
        if tok == "[UNK]":
            tok = "?"
        out.append(tok)
    return out


def process_prediction(row: Dict, idx2tag: Dict[int, str],
                       tokens: Optional[Sequence[str]] = None) -> Dict:
    """One predict-pkl row (pred_ids, label_ids, mask arrays) → real-token
    tag sequences with [CLS]/[SEP]/[PAD] stripped (reference
    tools/predict_utils.py:39-60)."""
    pred_ids = list(row["pred_ids"])
    label_ids = list(row["label_ids"])
    mask = list(row.get("mask", [1] * len(pred_ids)))
    pred_tags, label_tags, kept_tokens = [], [], []
    for i, (p, y, m) in enumerate(zip(pred_ids, label_ids, mask)):
        if not m:
            continue
        ytag = idx2tag.get(int(y), "O")
        if ytag in SPECIAL_TAGS:
            continue
        ptag = idx2tag.get(int(p), "O")
        pred_tags.append("O" if ptag in SPECIAL_TAGS else ptag)
        label_tags.append(ytag)
        if tokens is not None and i < len(tokens):
            kept_tokens.append(tokens[i])
    out = {"pred_tags": pred_tags, "label_tags": label_tags}
    if tokens is not None:
        out["tokens"] = fix_tokens(kept_tokens)
    return out


def decode_prediction(tokens: Sequence[str], tags: Sequence[str]) -> Dict[str, List[str]]:
    """BIO tag sequence + tokens → {entity_type: [surface, ...]}; a B/I
    type mismatch inside one span marks the surface with '[ERROR]'
    (reference tools/predict_utils.py:6-36)."""
    entities: Dict[str, List[str]] = {}
    cur_type, cur_toks, cur_err = None, [], False

    def flush():
        nonlocal cur_type, cur_toks, cur_err
        if cur_type is not None and cur_toks:
            surface = "".join(cur_toks) + ("[ERROR]" if cur_err else "")
            entities.setdefault(cur_type, []).append(surface)
        cur_type, cur_toks, cur_err = None, [], False

    for tok, tag in zip(fix_tokens(tokens), tags):
        if tag.startswith("B-"):
            flush()
            cur_type, cur_toks = tag[2:], [tok]
        elif tag.startswith("I-"):
            if cur_type is None:
                cur_type, cur_toks = tag[2:], [tok]
            else:
                if tag[2:] != cur_type:
                    cur_err = True
                cur_toks.append(tok)
        else:
            flush()
    flush()
    return entities


def extract_entity(tokens: Sequence[str], tags: Sequence[str]) -> Dict[str, set]:
    """Like decode_prediction but deduplicated per type (reference
    tools/infer_utils.py:76-99 returns {type: {surfaces}})."""
    ents = decode_prediction(tokens, tags)
    return {t: set(v) for t, v in ents.items()}


def bio_extract_entity(text: str, tags: Sequence[str]) -> List[str]:
    """Span surfaces from a BIO sequence over raw text characters
    (reference mrc/evaluation.py:8-24)."""
    spans, start = [], None
    for i, tag in enumerate(list(tags) + ["O"]):
        inside = i < len(text) and tag != "O"
        if tag.startswith("B-") or (inside and start is None):
            if start is not None:
                spans.append(text[start:i])
            start = i
        elif not inside and start is not None:
            spans.append(text[start:i])
            start = None
    return spans
