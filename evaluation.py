#!/usr/bin/env python3
"""Offline evaluation CLI — parity with the reference's evaluation.py:

  python evaluation.py --model_name bert_bilstm_crf --data msra
  python evaluation.py --model_name bilstm_crf,bert_bilstm_crf --data msra --topn 5

Loads ./data/{data}/{model}_predict.pkl dumped by main.py (reference
main.py:52-55 / evaluation.py:29-36), strips special tokens, prints the
entity-level strict span report + token-level tag report (reference
evaluation.py:38-55), and for multiple models a weighted-avg-F1
comparison table sorted by F1 (reference MultiEval :97-111)."""
from __future__ import annotations

import argparse
import os
import pickle
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from chinesener_amd.config import DATA_DIR
from chinesener_amd.data.datasets import get_spec
from chinesener_amd.eval import (entity_report, process_prediction,
                                 report_to_text, tag_report)


class SingleEval:
    def __init__(self, model_name: str, data: str, data_dir: str = DATA_DIR):
        self.model_name = model_name
        self.data = data
        path = os.path.join(data_dir, data, f"{model_name}_predict.pkl")
        if not os.path.exists(path):
            raise FileNotFoundError(
                f"{path} not found — run main.py --model_name {model_name} "
                f"--data {data} first (it dumps test predictions)")
        with open(path, "rb") as f:
            self.rows = pickle.load(f)
        self.idx2tag = get_spec(data).idx2tag
        processed = [process_prediction(r, self.idx2tag) for r in self.rows]
        self.y_true = [p["label_tags"] for p in processed]
        self.y_pred = [p["pred_tags"] for p in processed]

    def gen_report(self, topn: int = 0) -> dict:
        ent = entity_report(self.y_true, self.y_pred)
        print(report_to_text(
            ent, f"\n=== {self.model_name} @ {self.data}: entity-level (strict span) ==="))
        print(report_to_text(
            tag_report(self.y_true, self.y_pred),
            f"\n=== {self.model_name} @ {self.data}: tag-level ==="))
        if topn:
            print(f"\n--- top {topn} mismatched samples ---")
            shown = 0
            for row, t, p in zip(self.rows, self.y_true, self.y_pred):
                if t != p and shown < topn:
                    if "raw" in row:
                        print("text:", row["raw"])
                    print("true:", " ".join(t))
                    print("pred:", " ".join(p))
                    shown += 1
        return ent


class MultiEval:
    def __init__(self, model_names, data, data_dir: str = DATA_DIR):
        self.evals = [SingleEval(m, data, data_dir) for m in model_names]

    def gen_report(self, topn: int = 0):
        import pandas as pd
        rows = []
        for ev in self.evals:
            ent = ev.gen_report(topn)
            avg = ent.get("weighted avg", ent.get("micro avg",
                          {"precision": 0, "recall": 0, "f1": 0}))
            rows.append({"model": ev.model_name,
                         "precision": round(avg["precision"], 4),
                         "recall": round(avg["recall"], 4),
                         "f1": round(avg["f1"], 4)})
        table = pd.DataFrame(rows).sort_values("f1", ascending=False)
        print("\n=== model comparison (weighted-avg entity F1) ===")
        print(table.to_string(index=False))
        return table


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model_name", required=True,
                    help="one name, or comma-separated list for comparison")
    ap.add_argument("--data", required=True)
    ap.add_argument("--topn", type=int, default=0)
    ap.add_argument("--data_dir", default=DATA_DIR)
    args = ap.parse_args(argv)
    models = args.model_name.split(",")
    if len(models) == 1:
        SingleEval(models[0], args.data, args.data_dir).gen_report(args.topn)
    else:
        MultiEval(models, args.data, args.data_dir).gen_report(args.topn)
    return 0


if __name__ == "__main__":
    sys.exit(main())
