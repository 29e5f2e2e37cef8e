"""BIO corpus → MRC jsonl converter (reference mrc/convert2mrc.py:20-88):
each sentence becomes {"title": text, "label": [{span, tag, start_pos,
end_pos}]} per split."""
from __future__ import annotations

import json
import os
from typing import Dict, List, Sequence, Tuple

from ..data.datasets import get_spec, load_data
from ..eval.entity_eval import extract_spans


def sentence_to_record(sentence: Sequence[str], tags: Sequence[str]) -> Dict:
    text = "".join(sentence)
    labels = []
    for typ, start, end in extract_spans(tags):
        labels.append({"span": text[start:end], "tag": typ,
                       "start_pos": start, "end_pos": end})
    return {"title": text, "label": labels}


def convert2mrc(data: str, data_dir: str,
                splits: Tuple[str, ...] = ("train", "valid", "test")
                ) -> List[str]:
    """Writes {split}_mrc.jsonl next to the source corpus; returns paths."""
    get_spec(data)  # validates the dataset name
    out_paths = []
    for split in splits:
        sentences, tags = load_data(data, data_dir, split)
        path = os.path.join(data_dir, f"{split}_mrc.jsonl")
        with open(path, "w", encoding="utf-8") as f:
            for sent, tag in zip(sentences, tags):
                f.write(json.dumps(sentence_to_record(sent, tag),
                                   ensure_ascii=False) + "\n")
        out_paths.append(path)
    return out_paths


def load_mrc(path: str) -> List[Dict]:
    with open(path, encoding="utf-8") as f:
        return [json.loads(line) for line in f if line.strip()]
