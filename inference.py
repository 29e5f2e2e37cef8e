#!/usr/bin/env python3
"""Interactive gRPC inference client — reference inference.py parity:

  python chinesener_amd/serve/server.py --model bert_bilstm_crf   # terminal 1
  python inference.py --model bert_bilstm_crf                     # terminal 2
  输入: 北京大学的张三去了上海 → {'LOC': {'上海'}, 'ORG': {'北京大学'}, 'PER': {'张三'}}

InferHelper builds features online with the SAME L1 proc used at
training preprocess time (reference inference.py:33-99,
base_preprocess.py:176-187: fake labels, task_ids=1 for mtl), sends a
Predict RPC with retry/backoff, and decodes pred_ids → entities."""
from __future__ import annotations

import argparse
import logging
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from chinesener_amd.data.datasets import get_spec
from chinesener_amd.data.preprocess import extract_prefix_surfix, get_instance
from chinesener_amd.eval import extract_entity
from chinesener_amd.serve.client import PredictionClient, timer
from chinesener_amd.serve import rpc


class InferHelper:
    def __init__(self, model_name: str, data: str = "msra",
                 host: str = "127.0.0.1", port: int = rpc.DEFAULT_PORT,
                 max_seq_len: int | None = None, version: int | None = None):
        self.model_name = model_name
        spec = get_spec(data)
        self.max_seq_len = max_seq_len or spec.max_seq_len
        self.idx2tag = spec.idx2tag
        enhance, tok_type = extract_prefix_surfix(model_name)
        self.proc = get_instance(tok_type, self.max_seq_len, spec.tag2idx,
                                 enhance)
        self.is_mtl = "mtl" in model_name or "adv" in model_name
        self.client = PredictionClient(host, port)
        self.version = version

    def make_feature(self, sentence: str) -> dict:
        feat = self.proc.build_seq_feature(sentence)  # fake 'O' labels inside
        feats = {k: v[None, ...] for k, v in feat.items()}
        if self.is_mtl:
            # NER task is id 1 in the mtl pair (reference inference.py:70-71)
            feats["task_ids"] = np.ones_like(feats["token_ids"])
        return feats

    @timer
    def infer(self, sentence: str):
        feats = self.make_feature(sentence)
        resp = self.client.predict(self.model_name, feats, self.version)
        pred_ids = resp["outputs"]["pred_ids"][0]
        tokens = self.proc.tokenizer.tokenize(sentence)
        offset = 1 if self.proc.is_bert else 0  # skip [CLS]
        tags = [self.idx2tag.get(int(i), "O")
                for i in pred_ids[offset:offset + len(tokens)]]
        return extract_entity(tokens, tags)


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bert_bilstm_crf")
    ap.add_argument("--data", default="msra")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=rpc.DEFAULT_PORT)
    ap.add_argument("--text", default=None,
                    help="one-shot inference instead of the REPL")
    args = ap.parse_args(argv)
    logging.basicConfig(level=logging.INFO)
    helper = InferHelper(args.model, args.data, args.host, args.port)
    if args.text:
        print(helper.infer(args.text))
        return 0
    while True:  # reference inference.py:110-115 interactive loop
        try:
            text = input("输入文本: ").strip()
        except (EOFError, KeyboardInterrupt):
            break
        if not text or text in ("q", "quit", "exit"):
            break
        print(helper.infer(text))
    return 0


if __name__ == "__main__":
    sys.exit(main())
