"""L6 serving: export → engine → gRPC server → client → entity decode,
all on CPU (hipGraph capture itself is covered by the gpu-marked test in
test_gpu_serving.py)."""
import os
import pickle
import socket

import numpy as np
import pytest
import torch

from chinesener_amd.models import build_model, model_params
from chinesener_amd.config import resolve_params
from chinesener_amd.serve.export import export_model, load_exported


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _export_tiny(tmp_path, name="bilstm_crf", vocab_size=200):
    if vocab_size is None:
        # match the synthetic char vocab the online proc tokenizes with
        from chinesener_amd.data.tokenizer import Vocab
        vocab_size = len(Vocab.synthetic())
    params = resolve_params(model_params(name), {
        "vocab_size": vocab_size, "label_size": 7, "embedding_dim": 16,
        "max_seq_len": 32,
        "rnn_params": {"hidden_units_list": [8], "cell_activation": "tanh",
                       "keep_prob_list": [1.0]},
        "num_train_steps": 10, "step_per_epoch": 5, "model_name": name})
    model = build_model(name, params)
    out = export_model(model, name, params, export_root=str(tmp_path))
    return model, params, out


def test_export_load_roundtrip(tmp_path):
    model, params, out_dir = _export_tiny(tmp_path)
    assert os.path.exists(os.path.join(out_dir, "model.pt"))
    loaded, lparams = load_exported("bilstm_crf", str(tmp_path))
    for (k1, v1), (k2, v2) in zip(model.state_dict().items(),
                                  loaded.state_dict().items()):
        assert k1 == k2 and torch.equal(v1, v2)
    assert lparams["label_size"] == 7


def test_engine_eager_predict(tmp_path):
    _export_tiny(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine
    eng = InferenceEngine("bilstm_crf", str(tmp_path), use_graph=False,
                          max_seq_len=32)
    feats = {"token_ids": np.random.randint(1, 200, (2, 32)),
             "mask": np.ones((2, 32), dtype=np.int64),
             "seq_len": np.array([32, 32])}
    pred = eng.predict(feats)
    assert pred.shape == (2, 32)
    assert (pred >= 0).all() and (pred < 7).all()


def test_grpc_server_client_roundtrip(tmp_path):
    _export_tiny(tmp_path)
    port = _free_port()
    from chinesener_amd.serve.server import serve
    from chinesener_amd.serve.client import PredictionClient
    server = serve(["bilstm_crf"], str(tmp_path), port=port, wait=False,
                   use_graph=False)
    try:
        client = PredictionClient(port=port)
        feats = {"token_ids": np.random.randint(1, 200, (1, 32)),
                 "mask": np.ones((1, 32), dtype=np.int64)}
        resp = client.predict("bilstm_crf", feats)
        assert resp["outputs"]["pred_ids"].shape == (1, 32)
        assert "latency_ms" in resp
        client.close()
    finally:
        server.stop(0)


def test_grpc_unknown_model_not_found(tmp_path):
    import grpc
    _export_tiny(tmp_path)
    port = _free_port()
    from chinesener_amd.serve.server import serve
    from chinesener_amd.serve.client import PredictionClient
    server = serve(["bilstm_crf"], str(tmp_path), port=port, wait=False,
                   use_graph=False)
    try:
        client = PredictionClient(port=port)
        with pytest.raises(grpc.RpcError) as ei:
            client.predict("nope", {"token_ids": np.ones((1, 4), dtype=np.int64),
                                    "mask": np.ones((1, 4), dtype=np.int64)})
        assert ei.value.code() == grpc.StatusCode.NOT_FOUND
        client.close()
    finally:
        server.stop(0)


def test_warmup_file_and_replay(tmp_path, monkeypatch):
    _export_tiny(tmp_path, vocab_size=None)
    import warmup as warmup_mod
    path = warmup_mod.build_warmup("bilstm_crf", "msra", n=2,
                                   export_root=str(tmp_path))
    with open(path, "rb") as f:
        reqs = pickle.load(f)
    assert len(reqs) == 2
    # server loads + replays them at startup without error
    port = _free_port()
    from chinesener_amd.serve.server import serve
    server = serve(["bilstm_crf"], str(tmp_path), port=port, wait=False,
                   use_graph=False)
    server.stop(0)


def test_infer_helper_end_to_end(tmp_path):
    """Full client loop: sentence → features → RPC → entities."""
    _export_tiny(tmp_path, vocab_size=None)
    port = _free_port()
    from chinesener_amd.serve.server import serve
    server = serve(["bilstm_crf"], str(tmp_path), port=port, wait=False,
                   use_graph=False)
    try:
        import inference
        helper = inference.InferHelper("bilstm_crf", "msra", port=port,
                                       max_seq_len=32)
        ents = helper.infer("北京大学的张三去了上海")
        assert isinstance(ents, dict)  # may be empty: random weights
    finally:
        server.stop(0)


def test_rpc_pack_unpack():
    from chinesener_amd.serve import rpc
    a = np.random.randn(3, 5).astype(np.float32)
    b = np.arange(6, dtype=np.int64).reshape(2, 3)
    msg = rpc.make_predict_request("m", {"a": a, "b": b}, version=3)
    out = rpc.loads(rpc.dumps(msg))
    np.testing.assert_array_equal(out["inputs"]["a"], a)
    np.testing.assert_array_equal(out["inputs"]["b"], b)
    assert out["model_spec"]["version"] == 3


def test_retry_backoff_budget():
    import grpc
    from chinesener_amd.serve.client import grpc_retry

    class FakeError(grpc.RpcError):
        def code(self):
            return grpc.StatusCode.UNAVAILABLE

    calls = []

    @grpc_retry
    def flaky():
        calls.append(1)
        if len(calls) < 3:
            raise FakeError()
        return "ok"

    assert flaky() == "ok"
    assert len(calls) == 3

    calls.clear()

    @grpc_retry
    def always_down():
        calls.append(1)
        raise FakeError()

    with pytest.raises(grpc.RpcError):
        always_down()
    assert len(calls) == 4  # 1 initial + 3 retries (UNAVAILABLE budget)


def test_grpc_concurrent_requests(tmp_path):
    """Thread-pooled server + per-engine lock: concurrent requests each
    get a response consistent with their own input (the hipGraph static
    buffers are shared state; engine.predict serializes)."""
    import threading
    _export_tiny(tmp_path)
    port = _free_port()
    from chinesener_amd.serve.server import serve
    from chinesener_amd.serve.client import PredictionClient
    server = serve(["bilstm_crf"], str(tmp_path), port=port, wait=False,
                   use_graph=False, max_workers=4)
    errs = []

    def worker(seed):
        try:
            client = PredictionClient(port=port)
            rng = np.random.default_rng(seed)
            for _ in range(5):
                b = int(rng.integers(1, 4))
                feats = {"token_ids": rng.integers(1, 200, (b, 32)),
                         "mask": np.ones((b, 32), dtype=np.int64)}
                resp = client.predict("bilstm_crf", feats)
                assert resp["outputs"]["pred_ids"].shape == (b, 32)
            client.close()
        except Exception as e:   # surfaced to the main thread
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(4)]
    try:
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=120)
    finally:
        server.stop(0)
    assert not errs, errs


def test_micro_batcher_coalesces_and_matches(tmp_path):
    """MicroBatcher returns per-request rows identical to direct engine
    predicts, and actually coalesces concurrent requests into fewer
    engine calls (serving-concurrency fix: one replay for k clients)."""
    import threading
    _export_tiny(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine, MicroBatcher
    eng = InferenceEngine("bilstm_crf", str(tmp_path), use_graph=False,
                          max_seq_len=32, batch_sizes=(1, 4, 8))
    batcher = MicroBatcher(eng, window_ms=15.0)
    rng = np.random.default_rng(7)
    feats = [{"token_ids": rng.integers(1, 200, (1, 32)),
              "mask": np.ones((1, 32), dtype=np.int64)} for _ in range(8)]
    expected = [eng.predict(f) for f in feats]
    n0 = eng.n_requests

    results = [None] * 8
    barrier = threading.Barrier(8)

    def worker(i):
        barrier.wait()
        results[i] = batcher.predict(feats[i])

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    for got, want in zip(results, expected):
        assert got is not None and (got == want).all()
    coalesced_calls = eng.n_requests - n0
    assert coalesced_calls < 8, f"no coalescing: {coalesced_calls} calls"
    batcher.close()


def test_micro_batcher_mixed_keysets(tmp_path):
    """Requests with different feature keys/shapes must not be merged;
    both still complete correctly."""
    _export_tiny(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine, MicroBatcher
    eng = InferenceEngine("bilstm_crf", str(tmp_path), use_graph=False,
                          max_seq_len=32)
    batcher = MicroBatcher(eng, window_ms=5.0)
    rng = np.random.default_rng(3)
    a = {"token_ids": rng.integers(1, 200, (2, 32)),
         "mask": np.ones((2, 32), dtype=np.int64)}
    b = {"token_ids": rng.integers(1, 200, (1, 32)),
         "mask": np.ones((1, 32), dtype=np.int64),
         "seq_len": np.array([32])}
    ra, rb = batcher.predict(a), batcher.predict(b)
    assert ra.shape == (2, 32) and rb.shape == (1, 32)
    assert (ra == eng.predict(a)).all() and (rb == eng.predict(b)).all()
    batcher.close()


def test_micro_batcher_ragged_stress(tmp_path):
    """Many concurrent clients with MIXED batch sizes: every request
    gets its own rows back bit-exactly; errors in one group don't leak
    into others."""
    import threading
    _export_tiny(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine, MicroBatcher
    eng = InferenceEngine("bilstm_crf", str(tmp_path), use_graph=False,
                          max_seq_len=32, batch_sizes=(1, 4, 8))
    batcher = MicroBatcher(eng, window_ms=3.0)
    rng = np.random.default_rng(11)
    reqs = [{"token_ids": rng.integers(1, 200, (int(rng.integers(1, 5)), 32)),
             "mask": np.ones((1, 32), dtype=np.int64).repeat(1, 0)}
            for _ in range(24)]
    for r in reqs:
        r["mask"] = np.ones((r["token_ids"].shape[0], 32), dtype=np.int64)
    expected = [eng.predict(r) for r in reqs]
    results = [None] * len(reqs)
    errs = []

    def worker(i):
        try:
            results[i] = batcher.predict(reqs[i])
        except Exception as e:
            errs.append((i, e))

    threads = [threading.Thread(target=worker, args=(i,))
               for i in range(len(reqs))]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errs, errs
    for got, want in zip(results, expected):
        assert got is not None and got.shape == want.shape
        assert (got == want).all()
    batcher.close()


def test_micro_batcher_oversize_request(tmp_path):
    """A request larger than max_batch must dispatch alone via the
    engine's eager path, not starve in the queue."""
    _export_tiny(tmp_path)
    from chinesener_amd.serve.engine import InferenceEngine, MicroBatcher
    eng = InferenceEngine("bilstm_crf", str(tmp_path), use_graph=False,
                          max_seq_len=32, batch_sizes=(1, 4))
    batcher = MicroBatcher(eng, window_ms=2.0)
    rng = np.random.default_rng(5)
    big = {"token_ids": rng.integers(1, 200, (9, 32)),
           "mask": np.ones((9, 32), dtype=np.int64)}
    out = batcher.predict(big)             # must not hang
    assert out.shape == (9, 32)
    assert (out == eng.predict(big)).all()
    batcher.close()
