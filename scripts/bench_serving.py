#!/usr/bin/env python3
"""Serving latency bench: hipGraph-captured BERT-BiLSTM-CRF inference
(BASELINE.json L6 path) vs eager, batch=1 and batch=8, p50/p99 over N
requests. Run on an MI355X box; prints one JSON line."""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=200)
    ap.add_argument("--layers", type=int, default=12)
    ap.add_argument("--seq_len", type=int, default=150)
    args = ap.parse_args()
    assert torch.cuda.is_available()

    from chinesener_amd.config import resolve_params
    from chinesener_amd.models import build_model, model_params
    from chinesener_amd.models.bert import BertConfig
    from chinesener_amd.serve.engine import InferenceEngine
    from chinesener_amd.serve.export import export_model
    from chinesener_amd.ops.tunable import load_tuned_gemm_table

    load_tuned_gemm_table()
    name = "bert_bilstm_crf"
    cfg = BertConfig(vocab_size=21128, num_hidden_layers=args.layers)
    params = resolve_params(model_params(name), {
        "vocab_size": 21128, "label_size": 10, "bert_config": cfg,
        "max_seq_len": args.seq_len, "model_name": name,
        "rnn_params": {"hidden_units_list": [128], "cell_activation": "relu",
                       "keep_prob_list": [1.0]}})
    model = build_model(name, params)
    export_model(model, name, params, export_root="/tmp/serving_bench")

    results = {}
    for graph in (True, False):
        eng = InferenceEngine(name, "/tmp/serving_bench", batch_sizes=(1, 8),
                              max_seq_len=args.seq_len, use_graph=graph)
        eng.warmup()
        rng = np.random.default_rng(0)
        for batch in (1, 8):
            feats = {"token_ids": rng.integers(1, 21128, (batch, args.seq_len)),
                     "segment_ids": np.zeros((batch, args.seq_len), np.int64),
                     "mask": np.ones((batch, args.seq_len), np.int64)}
            for _ in range(20):
                eng.predict(feats)
            lat = []
            for _ in range(args.requests):
                t0 = time.perf_counter()
                eng.predict(feats)
                lat.append((time.perf_counter() - t0) * 1000)
            lat.sort()
            key = f"{'graph' if graph else 'eager'}_b{batch}"
            results[key] = {
                "p50_ms": round(lat[len(lat) // 2], 3),
                "p99_ms": round(lat[int(len(lat) * 0.99)], 3),
                "mean_ms": round(sum(lat) / len(lat), 3)}
        del eng
        torch.cuda.empty_cache()

    # concurrency: 8 clients of batch-1 requests, locked engine vs
    # micro-batching queue (VERDICT weak #6 / serving-concurrency fix)
    import threading
    from chinesener_amd.serve.engine import MicroBatcher
    rng = np.random.default_rng(0)
    feats1 = {"token_ids": rng.integers(1, 21128, (1, args.seq_len)),
              "segment_ids": np.zeros((1, args.seq_len), np.int64),
              "mask": np.ones((1, args.seq_len), np.int64)}
    for mode in ("locked", "microbatch"):
        eng = InferenceEngine(name, "/tmp/serving_bench", batch_sizes=(1, 8),
                              max_seq_len=args.seq_len, use_graph=True)
        eng.warmup()
        target = MicroBatcher(eng, window_ms=0.3) if mode == "microbatch" else eng
        for _ in range(20):
            target.predict(feats1)
        lat_all = []
        lock = threading.Lock()
        n_clients, per_client = 8, max(25, args.requests // 8)

        def client():
            mine = []
            for _ in range(per_client):
                t0 = time.perf_counter()
                target.predict(feats1)
                mine.append((time.perf_counter() - t0) * 1000)
            with lock:
                lat_all.extend(mine)

        threads = [threading.Thread(target=client) for _ in range(n_clients)]
        t0 = time.perf_counter()
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        wall = time.perf_counter() - t0
        lat_all.sort()
        results[f"concurrent8_{mode}"] = {
            "p50_ms": round(lat_all[len(lat_all) // 2], 3),
            "p99_ms": round(lat_all[int(len(lat_all) * 0.99)], 3),
            "throughput_rps": round(n_clients * per_client / wall, 1)}
        if mode == "microbatch":
            target.close()
        del eng
        torch.cuda.empty_cache()

    print(json.dumps({
        "metric": "serving latency, bert_bilstm_crf PREDICT (BERT-12L + BiLSTM + Viterbi)",
        "seq_len": args.seq_len, "requests": args.requests,
        "results": results,
        "speedup_b1": round(results["eager_b1"]["p50_ms"]
                            / results["graph_b1"]["p50_ms"], 2)}))


if __name__ == "__main__":
    main()
