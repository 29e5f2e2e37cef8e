"""Trainer loop, schedules, checkpoint/resume, CLI e2e (CPU)."""
import itertools
import os
import subprocess
import sys

import pytest
import torch

from chinesener_amd.config import RUN_CONFIG, resolve_params
from chinesener_amd.data.loader import NerDataset
from chinesener_amd.models import build_model, model_params, optimizer_family
from chinesener_amd.train.optimizers import (AdamWeightDecay, LrSchedule,
                                             build_optimizer, build_param_groups)
from chinesener_amd.train.trainer import Trainer

from conftest import make_tiny_batch, make_tiny_params

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_optimizer_family_dispatch():
    assert optimizer_family("bert_bilstm_crf") == "bert"
    assert optimizer_family("transformer_tener_crf_bichar") == "transformer"
    assert optimizer_family("bilstm_crf") == "custom"


def test_lr_schedule_shapes():
    s = LrSchedule("bert", 1e-4, num_train_steps=100, warmup_ratio=0.1)
    assert s.lr_at(5) < s.lr_at(10) - 1e-12          # warming up
    assert s.lr_at(10) == pytest.approx(1e-4)        # peak at warmup end
    assert s.lr_at(100) < 1e-5                        # decayed
    noam = LrSchedule("transformer", 1.0, num_train_steps=1000, warmup_steps=40)
    assert noam.lr_at(10) < noam.lr_at(40) and noam.lr_at(400) < noam.lr_at(40)
    exp = LrSchedule("custom", 1e-3, num_train_steps=100, step_per_epoch=10,
                     decay_rate=0.5)
    assert exp.lr_at(5) == pytest.approx(1e-3)
    assert exp.lr_at(25) == pytest.approx(2.5e-4)


def test_diff_lr_groups():
    params = make_tiny_params("bert_bilstm_crf")
    model = build_model("bert_bilstm_crf", params)
    groups = build_param_groups(model, 5e-5, 0.01,
                                {"crf": 500, "logit": 500, "bilstm": 100})
    scales = {g["lr_scale"] for g in groups}
    assert {1.0, 500.0, 100.0} <= scales
    # no-decay groups exist (LayerNorm/bias)
    assert any(g["weight_decay"] == 0.0 for g in groups)
    assert any(g["weight_decay"] > 0.0 for g in groups)


def test_diff_lr_warm_ramp():
    """The x500 crf/logit multiplier ramps 1 -> full over the schedule
    warmup window (all-O basin mitigation); after warmup it equals the
    reference's constant multiplier. Graph mode pins the full scale."""
    import torch
    w_crf = torch.nn.Parameter(torch.zeros(3))
    w_other = torch.nn.Parameter(torch.zeros(3))

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.crf_weight = w_crf
            self.other = w_other

    groups = build_param_groups(M(), 5e-5, 0.01, {"crf": 500})
    opt = AdamWeightDecay(groups, lr=5e-5)
    sched = LrSchedule("bert", 5e-5, num_train_steps=1000, warmup_ratio=0.1)
    assert sched.warmup == 100

    def crf_scale():
        return next(g["lr_scale"] for g in opt.param_groups
                    if g.get("lr_scale_base") == 500.0)

    sched.apply(opt, 1)
    assert abs(crf_scale() - (1.0 + 499.0 * 0.01)) < 1e-9
    sched.apply(opt, 50)
    assert abs(crf_scale() - (1.0 + 499.0 * 0.5)) < 1e-9
    sched.apply(opt, 100)
    assert crf_scale() == 500.0
    sched.apply(opt, 700)
    assert crf_scale() == 500.0     # post-warmup: exact reference scale
    # plain groups never ramp
    assert all(g["lr_scale"] == 1.0 for g in opt.param_groups
               if g.get("lr_scale_base", 1.0) == 1.0)


def test_max_steps_extends_schedule(tmp_path):
    """main.py: --max_steps beyond step_per_epoch*epochs must extend
    num_train_steps so the poly decay never reaches 0 mid-run."""
    from chinesener_amd.config import resolve_params
    params = resolve_params({}, {"step_per_epoch": 10, "num_train_steps": 100},
                            {"epoch_size": 10})
    max_steps = 500
    params["num_train_steps"] = max(params.get("num_train_steps", 0), max_steps)
    sched = LrSchedule("bert", 5e-5,
                       num_train_steps=params["num_train_steps"],
                       warmup_ratio=0.1)
    assert sched.lr_at(400) > 0


def test_adamw_decreases_loss():
    torch.manual_seed(0)
    w = torch.nn.Parameter(torch.randn(10))
    opt = AdamWeightDecay([{"params": [w], "lr_scale": 1.0,
                            "weight_decay": 0.0}], lr=0.1)
    target = torch.randn(10)
    first = None
    for _ in range(50):
        opt.zero_grad()
        loss = ((w - target) ** 2).sum()
        if first is None:
            first = float(loss)
        loss.backward()
        opt.step()
    assert float(loss) < first * 0.1


def test_trainer_steps_and_resume(tmp_path):
    torch.manual_seed(0)
    name = "bilstm_crf"
    params = make_tiny_params(name)
    params.update(num_train_steps=50, step_per_epoch=10, lr=1e-2)
    model = build_model(name, params)
    tr = Trainer(model, name, params, str(tmp_path / "ckpt"), device="cpu")
    batches = [make_tiny_batch(name, seed=i) for i in range(12)]
    tr.train(iter(batches), log_steps=1000, save_steps=5)
    assert tr.step == 12
    # resume picks up the saved step
    model2 = build_model(name, params)
    tr2 = Trainer(model2, name, params, str(tmp_path / "ckpt"), device="cpu")
    assert tr2.step >= 10
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        if tr2.step == tr.step:
            torch.testing.assert_close(p1, p2)


def test_trainer_loss_decreases_overfit():
    torch.manual_seed(0)
    name = "bilstm_crf"
    params = make_tiny_params(name)
    params.update(num_train_steps=200, step_per_epoch=50, lr=5e-2)
    model = build_model(name, params)
    import tempfile
    with tempfile.TemporaryDirectory() as d:
        tr = Trainer(model, name, params, d, device="cpu")
        batch = make_tiny_batch(name, batch_size=4, seed=0)
        losses = [tr.train_step(batch) for _ in range(60)]
    assert losses[-1] < losses[0] * 0.5, losses[::10]


def test_predict_dump_roundtrip(tmp_path):
    name = "bilstm_crf"
    params = make_tiny_params(name)
    model = build_model(name, params)
    tr = Trainer(model, name, params, str(tmp_path / "c"), device="cpu")
    rows = tr.predict([make_tiny_batch(name, seed=s) for s in range(2)])
    path = tr.dump_predictions(rows, str(tmp_path), "bilstm_crf")
    import pickle
    with open(path, "rb") as f:
        loaded = pickle.load(f)
    assert len(loaded) == 6 and "pred_ids" in loaded[0]


def test_main_cli_e2e(tmp_path):
    """Full CLI slice: train 4 steps of bilstm_crf on synthetic people_daily
    (BASELINE config 1 plumbing)."""
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "main.py"),
         "--model_name", "bilstm_crf", "--data", "people_daily",
         "--epochs", "1", "--batch_size", "8", "--max_steps", "4",
         "--data_dir", str(tmp_path / "data"),
         "--ckpt_root", str(tmp_path / "ckpt")],
        capture_output=True, text=True, env=env, timeout=600,
        cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-3000:]
    assert os.path.exists(tmp_path / "ckpt" / "ner_people_daily_bilstm_crf")
    assert os.path.exists(tmp_path / "data" / "people_daily" /
                          "bilstm_crf_predict.pkl")


def test_addon_parser():
    import argparse
    from chinesener_amd.train.addon_parser import Addon, AddonParser
    ap = argparse.ArgumentParser()
    AddonParser([Addon("lambda_adv", 0.05), Addon("asymmetry", None,
                                                  action="store_true")]).append(ap)
    args = ap.parse_args(["--lambda_adv", "0.1", "--asymmetry"])
    assert args.lambda_adv == 0.1 and args.asymmetry is True


def test_convert_bf16_mixed_keeps_ln_fp32():
    import torch
    from chinesener_amd.models import build_model
    from chinesener_amd.models.bert import BertConfig
    from chinesener_amd.train.precision import convert_bf16_mixed
    cfg = BertConfig(vocab_size=100, hidden_size=32, num_hidden_layers=1,
                     num_attention_heads=2, intermediate_size=64)
    m = build_model("bert_bilstm_crf", {
        "vocab_size": 100, "label_size": 5, "bert_config": cfg,
        "rnn_params": {"hidden_units_list": [32], "cell_activation": "relu",
                       "keep_prob_list": [1.0]}, "tag2idx": {},
        "dropout_rate": 0.0})
    convert_bf16_mixed(m)
    dt = {n: p.dtype for n, p in m.named_parameters()}
    assert dt["bert.layers.0.ln1_w"] == torch.float32
    assert dt["crf.transitions"] == torch.float32
    assert dt["bert.layers.0.qkv.weight"] == torch.bfloat16
    assert dt["bilstm.w_ih_f"] == torch.bfloat16


def test_adamw_bf16_master_cpu_math():
    """Eager master-weight path: bf16 params track the fp32 master."""
    import torch
    from chinesener_amd.train.optimizers import AdamWeightDecay
    p = torch.nn.Parameter(torch.randn(32).to(torch.bfloat16))
    opt = AdamWeightDecay([{"params": [p]}], lr=1e-2, weight_decay=0.0)
    for _ in range(5):
        p.grad = torch.randn(32).to(torch.bfloat16)
        opt.step()
    master = opt.state[p]["master"]
    assert master.dtype == torch.float32
    torch.testing.assert_close(p.detach().float(),
                               master.to(torch.bfloat16).float())


def test_summary_logger(tmp_path):
    """verbose=True writes zero-fraction/hist JSONL at summary_steps
    cadence (reference add_layer_summary, tools/utils.py:25-27)."""
    import json
    import torch
    from chinesener_amd.train.trainer import Trainer
    from chinesener_amd.models import build_model
    params = make_tiny_params("bilstm_crf")
    params.update(verbose=True, summary_steps=2, dtype="fp32")
    model = build_model("bilstm_crf", params)
    tr = Trainer(model, "bilstm_crf", params, str(tmp_path), device="cpu")
    for i in range(4):
        tr.train_step(make_tiny_batch("bilstm_crf", batch_size=2, seed=i))
    path = tmp_path / "summaries.jsonl"
    assert path.exists()
    recs = [json.loads(l) for l in path.read_text().splitlines()]
    assert [r["step"] for r in recs] == [2, 4]
    layers = recs[0]["layers"]
    assert layers, "no layer stats captured"
    stat = next(iter(layers.values()))
    assert set(stat) >= {"zero_fraction", "mean", "std", "absmax", "hist"}
    assert len(stat["hist"]) == 16
    assert 0.0 <= stat["zero_fraction"] <= 1.0


def test_main_cli_addon_flags():
    """Per-model AddonParser flags appear in main.py's CLI and land in
    params (reference pattern: model modules declare extra flags)."""
    import subprocess
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "main.py"),
         "--model_name", "bert_dice", "--data", "msra", "--help"],
        capture_output=True, text=True, env=env, timeout=120)
    assert r.returncode == 0
    assert "--alpha" in r.stdout and "--gamma" in r.stdout

    from chinesener_amd.models import apply_addon_values, model_params
    params = model_params("bert_bilstm_crf_adv")
    apply_addon_values("bert_bilstm_crf_adv", params,
                       {"task_weight_1": 0.7, "lambda_adv": 0.2,
                        "shrink_gradient_reverse": None})
    assert params["task_weight"][0] == 0.7
    assert abs(params["task_weight"][1] - 0.3) < 1e-12
    assert params["lambda"] == 0.2
    assert params["shrink_gradient_reverse"] == 0.01   # None kept default


def test_summary_artifacts_dump(tmp_path):
    """flush_artifacts writes the CRF transition matrix npz (the
    reference's TB transition image/histogram parity,
    tools/layer.py:129-130)."""
    import numpy as np
    from chinesener_amd.config import resolve_params
    from chinesener_amd.models import build_model, model_params
    from chinesener_amd.train.summaries import SummaryLogger

    params = resolve_params(model_params("bilstm_crf"), {
        "vocab_size": 50, "label_size": 7, "embedding_dim": 8,
        "model_name": "bilstm_crf",
        "rnn_params": {"hidden_units_list": [8], "cell_activation": "tanh",
                       "keep_prob_list": [1.0]}})
    model = build_model("bilstm_crf", params)
    sl = SummaryLogger(model, str(tmp_path), every=10)
    path = sl.flush_artifacts(model, 10)
    blob = np.load(path)
    trans_keys = [k for k in blob.files if "transitions" in k]
    assert trans_keys, blob.files
    assert blob[trans_keys[0]].shape == (7, 7)
    sl.close()


def test_cpu_training_deterministic(tmp_path):
    """Same seed -> bitwise-identical CPU training losses (SURVEY §5.2:
    the determinism story; GPU runs are statistical — see
    docs/NOTES.md)."""
    import torch
    from chinesener_amd.config import resolve_params
    from chinesener_amd.models import build_model, model_params
    from chinesener_amd.train.trainer import Trainer
    from conftest import make_tiny_batch, make_tiny_params

    def run(tag):
        torch.manual_seed(7)
        params = make_tiny_params("bilstm_crf")
        model = build_model("bilstm_crf", params)
        tr = Trainer(model, "bilstm_crf", params, str(tmp_path / tag))
        return [tr.train_step(make_tiny_batch("bilstm_crf", batch_size=2,
                                              seed=i))
                for i in range(4)]

    assert run("a") == run("b")


def test_profiling_and_tunable_helpers():
    """roctx ranges are safe no-ops without the marker library; the
    TunableOp helpers are inert unless explicitly opted in."""
    from chinesener_amd.utils.profiling import roctx_range, range_push, \
        range_pop
    with roctx_range("unit-test"):
        range_push("inner")
        range_pop()
    import os
    from chinesener_amd.ops import tunable
    assert os.environ.get("CHINESENER_TUNABLE") != "1"
    assert tunable.load_tuned_gemm_table() is False   # opt-in since r2
    tunable.freeze()                                  # never raises


def test_bench_json_driver_contract():
    """bench.py's single JSON line must satisfy the driver contract
    (fields, types, aggregate semantics) — guards future edits."""
    import json
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch_size", "2", "--seq_len", "32"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    r = json.loads(line)
    for k, t in [("metric", str), ("value", (int, float)), ("unit", str),
                 ("n_gpus", int), ("steps", int), ("warmup", int),
                 ("ms_per_step", (int, float)), ("higher_is_better", bool),
                 ("scaling", str), ("dtype", str), ("data", str),
                 ("config", dict)]:
        assert k in r and isinstance(r[k], t), (k, r.get(k))
    assert r["vs_baseline"] is None          # no published baseline
    assert r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    assert r["n_gpus"] == 1 and r["steps"] == 2 and r["warmup"] == 1
    assert r["config"]["global_batch"] == 2
    assert r["config"]["parallelism"] == "dp1"
    # value is the whole-job aggregate: steps*batch/elapsed
    assert abs(r["value"] - 2 * 2 / (r["ms_per_step"] * 2 / 1000)) < 1.0
