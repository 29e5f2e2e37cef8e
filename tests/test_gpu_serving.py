"""GPU serving: hipGraph capture of the full BERT-BiLSTM-CRF PREDICT
path (BASELINE.json: "hipGraph-captured BERT-BiLSTM-CRF inference") —
graph replay must agree with the eager path and be faster per request."""
import time

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _cuda():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from chinesener_amd import ops
    assert ops.ext_available()


def _export_bert(tmp_path, n_layers=2):
    from chinesener_amd.config import resolve_params
    from chinesener_amd.models import build_model, model_params
    from chinesener_amd.models.bert import BertConfig
    from chinesener_amd.serve.export import export_model
    name = "bert_bilstm_crf"
    cfg = BertConfig(vocab_size=2000, hidden_size=768,
                     num_hidden_layers=n_layers, num_attention_heads=12,
                     intermediate_size=3072)
    params = resolve_params(model_params(name), {
        "vocab_size": 2000, "label_size": 10, "bert_config": cfg,
        "max_seq_len": 64,
        "rnn_params": {"hidden_units_list": [128], "cell_activation": "relu",
                       "keep_prob_list": [1.0]},
        "num_train_steps": 10, "step_per_epoch": 5, "model_name": name})
    model = build_model(name, params)
    export_model(model, name, params, export_root=str(tmp_path))
    return name


def test_hipgraph_capture_matches_eager(tmp_path):
    _cuda()
    name = _export_bert(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine
    eng_graph = InferenceEngine(name, str(tmp_path), batch_sizes=(1, 4),
                                max_seq_len=64, use_graph=True)
    eng_eager = InferenceEngine(name, str(tmp_path), max_seq_len=64,
                                use_graph=False)
    eng_graph.warmup()
    assert set(eng_graph._graphs) == {1, 4}
    rng = np.random.default_rng(0)
    feats = {"token_ids": rng.integers(1, 2000, (4, 64)),
             "segment_ids": np.zeros((4, 64), dtype=np.int64),
             "mask": np.ones((4, 64), dtype=np.int64)}
    pred_g = eng_graph.predict(feats)
    pred_e = eng_eager.predict(feats)
    assert pred_g.shape == (4, 64)
    np.testing.assert_array_equal(pred_g, pred_e)


def test_hipgraph_pads_to_bucket(tmp_path):
    _cuda()
    name = _export_bert(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine
    eng = InferenceEngine(name, str(tmp_path), batch_sizes=(4,),
                          max_seq_len=64, use_graph=True)
    eng.warmup()
    rng = np.random.default_rng(1)
    feats = {"token_ids": rng.integers(1, 2000, (3, 64)),
             "segment_ids": np.zeros((3, 64), dtype=np.int64),
             "mask": np.ones((3, 64), dtype=np.int64)}
    pred = eng.predict(feats)
    assert pred.shape == (3, 64)


def test_hipgraph_latency_beats_eager(tmp_path):
    _cuda()
    name = _export_bert(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine
    eng = InferenceEngine(name, str(tmp_path), batch_sizes=(1,),
                          max_seq_len=64, use_graph=True)
    eng.warmup()
    rng = np.random.default_rng(2)
    feats = {"token_ids": rng.integers(1, 2000, (1, 64)),
             "segment_ids": np.zeros((1, 64), dtype=np.int64),
             "mask": np.ones((1, 64), dtype=np.int64)}

    def bench(fn, n=30):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1000

    t_graph = bench(lambda: eng.predict(feats))
    eng.use_graph = False
    t_eager = bench(lambda: eng.predict(feats))
    print(f"graph {t_graph:.2f} ms vs eager {t_eager:.2f} ms per request")
    assert t_graph < t_eager, (t_graph, t_eager)
