#!/usr/bin/env python3
"""Write serving warmup requests into the latest export's assets.extra/
(reference warmup.py:11-24 writes tf_serving_warmup_requests). The
server replays these at startup, which also fixes the hipGraph batch
buckets before the first real request."""
from __future__ import annotations

import argparse
import os
import pickle
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from chinesener_amd.config import EXPORT_DIR
from chinesener_amd.data.datasets import get_spec
from chinesener_amd.data.preprocess import extract_prefix_surfix, get_instance
from chinesener_amd.serve.export import latest_version_dir
from chinesener_amd.serve.server import WARMUP_FILE

WARMUP_SENTENCES = ["据悉是受到轻伤直接感受是疼痛", "北京大学的张三去了上海",
                    "中国建设银行今天发布了公告", "他在纽约时报工作过三年",
                    "春天的故事在深圳传唱"]


def build_warmup(model_name: str, data: str = "msra", n: int = 5,
                 export_root: str = EXPORT_DIR) -> str:
    spec = get_spec(data)
    enhance, tok_type = extract_prefix_surfix(model_name)
    proc = get_instance(tok_type, spec.max_seq_len, spec.tag2idx, enhance)
    is_mtl = "mtl" in model_name or "adv" in model_name
    requests = []
    for sent in WARMUP_SENTENCES[:n]:
        feat = proc.build_seq_feature(sent)
        feats = {k: v[None, ...] for k, v in feat.items()}
        if is_mtl:
            feats["task_ids"] = np.ones_like(feats["token_ids"])
        requests.append(feats)
    out_dir = latest_version_dir(model_name, export_root)
    path = os.path.join(out_dir, WARMUP_FILE)
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "wb") as f:
        pickle.dump(requests, f)
    print(f"wrote {len(requests)} warmup requests -> {path}")
    return path


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bert_bilstm_crf")
    ap.add_argument("--data", default="msra")
    ap.add_argument("--n", type=int, default=5)
    ap.add_argument("--export_root", default=EXPORT_DIR)
    args = ap.parse_args(argv)
    build_warmup(args.model, args.data, args.n, args.export_root)
    return 0


if __name__ == "__main__":
    sys.exit(main())
