"""GPU end-to-end: flagship model train steps on the HIP path, eval
decode, loss decreases on a fixed batch."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from chinesener_amd import ops


def _cuda():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert ops.ext_available()


def _small_bert_params(model_name, label_size=10):
    from chinesener_amd.models.bert import BertConfig
    cfg = BertConfig(vocab_size=2000, hidden_size=768, num_hidden_layers=2,
                     num_attention_heads=12, intermediate_size=3072)
    return {"vocab_size": 2000, "label_size": label_size, "bert_config": cfg,
            "rnn_params": {"hidden_units_list": [128],
                           "cell_activation": "relu", "keep_prob_list": [0.8]},
            "tag2idx": {}, "dropout_rate": 0.1, "lr": 5e-5,
            "num_train_steps": 100, "step_per_epoch": 10,
            "task_list": ["a", "b"], "a": {"label_size": 10},
            "b": {"label_size": 7}}


def test_bert_bilstm_crf_train_steps():
    _cuda()
    torch.manual_seed(0)
    from chinesener_amd.data.loader import make_synthetic_batch
    from chinesener_amd.models import build_model
    from chinesener_amd.train.optimizers import (AdamWeightDecay,
                                                 build_param_groups)
    from chinesener_amd.train.optimizers import LrSchedule, clip_gradients
    model = build_model("bert_bilstm_crf", _small_bert_params("bert_bilstm_crf"))
    model.to("cuda")
    opt = AdamWeightDecay(build_param_groups(model, 1e-4, 0.01), lr=1e-4)
    sched = LrSchedule("bert", 1e-4, num_train_steps=200, warmup_ratio=0.2)
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    losses = []
    for step in range(1, 25):
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(batch)
        out.loss.backward()
        clip_gradients(model, "bert")      # global-norm 1.0, as the trainer
        sched.apply(opt, step)
        opt.step()
        losses.append(float(out.loss.detach()))
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert losses[-1] < losses[0], losses


def test_decode_path_gpu():
    _cuda()
    torch.manual_seed(1)
    from chinesener_amd.data.loader import make_synthetic_batch
    from chinesener_amd.models import build_model
    model = build_model("bert_bilstm_crf",
                        _small_bert_params("bert_bilstm_crf")).to("cuda").eval()
    batch = make_synthetic_batch(4, 96, 10, vocab_size=2000, device="cuda")
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16):
        out = model(batch, compute_pred=True)
    assert out.pred_ids.shape == batch["token_ids"].shape
    assert (out.pred_ids * (1 - batch["mask"])).sum() == 0


def test_mtl_train_step_gpu():
    _cuda()
    torch.manual_seed(2)
    from chinesener_amd.data.loader import make_synthetic_batch
    from chinesener_amd.models import build_model
    model = build_model("bert_bilstm_crf_mtl",
                        _small_bert_params("bert_bilstm_crf_mtl")).to("cuda")
    batch = make_synthetic_batch(8, 64, 7, vocab_size=2000, device="cuda")
    task = (torch.arange(8, device="cuda") % 2)
    batch["task_ids"] = task[:, None].expand(8, 64).clone()
    batch["label_ids"] = batch["label_ids"].clamp(max=6)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = model(batch)
    out.loss.backward()
    assert torch.isfinite(out.loss)


def test_softlexicon_model_gpu():
    _cuda()
    torch.manual_seed(3)
    from chinesener_amd.data.loader import make_synthetic_batch
    from chinesener_amd.models import build_model
    params = _small_bert_params("bert_bilstm_crf_softlexicon")
    params.update(word_vocab_size=5000, word_dim=50,
                  rnn_params={"hidden_units_list": [128],
                              "cell_activation": "tanh",
                              "keep_prob_list": [0.8]})
    model = build_model("bert_bilstm_crf_softlexicon", params).to("cuda")
    batch = make_synthetic_batch(4, 64, 10, vocab_size=2000,
                                 word_enhance="softlexicon", device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = model(batch)
    out.loss.backward()
    assert torch.isfinite(out.loss)


def test_tener_model_gpu():
    _cuda()
    torch.manual_seed(4)
    from chinesener_amd.data.loader import make_synthetic_batch
    from chinesener_amd.models import build_model
    params = {"vocab_size": 2000, "label_size": 10, "embedding_dim": 50,
              "bichar_vocab_size": 1000, "bichar_dim": 50,
              "transformer_params": {"d_model": 160, "num_head": 8,
                                     "ffn_hidden": 320,
                                     "encode_attention_layers": 2},
              "dropout_rate": 0.1, "tag2idx": {}}
    model = build_model("transformer_tener_crf_bichar", params).to("cuda")
    batch = make_synthetic_batch(4, 150, 10, vocab_size=2000, is_bert=False,
                                 device="cuda")
    batch["bichar_ids"] = torch.randint(0, 1000, (4, 150), device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = model(batch)
    out.loss.backward()
    assert torch.isfinite(out.loss)


def test_pure_bf16_trainer_loss_decreases(tmp_path):
    """Pure-bf16 mode (bf16 weights + fp32 masters, no autocast): the
    flagship model trains and the loss decreases on a fixed batch."""
    _cuda()
    torch.manual_seed(3)
    from chinesener_amd.data.loader import make_synthetic_batch
    from chinesener_amd.models import build_model
    from chinesener_amd.train.trainer import Trainer
    params = _small_bert_params("bert_bilstm_crf")
    params.update({"lr": 1e-4, "dtype": "bf16", "num_train_steps": 200,
                   "warmup_ratio": 0.2})
    model = build_model("bert_bilstm_crf", params)
    trainer = Trainer(model, "bert_bilstm_crf", params, str(tmp_path / "ck"))
    assert trainer.pure_bf16
    # LN params stay fp32, linear weights bf16
    dt = {n: p.dtype for n, p in model.named_parameters()}
    assert dt["bert.layers.0.ln1_w"] == torch.float32
    assert dt["bert.layers.0.qkv.weight"] == torch.bfloat16
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    losses = [trainer.train_step(batch) for _ in range(25)]
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert losses[-1] < losses[0], losses


def test_convergence_entity_f1(tmp_path):
    """End-to-end learnability: the synthetic corpus draws entity chars
    from per-type disjoint ranges, so a real training run must push
    entity-level F1 far above chance (loss/decode/eval all correct)."""
    _cuda()
    torch.manual_seed(7)
    from chinesener_amd.config import resolve_params
    from chinesener_amd.data.loader import NerDataset
    from chinesener_amd.models import build_model, model_params
    from chinesener_amd.models.bert import BertConfig
    from chinesener_amd.train.trainer import Trainer
    from chinesener_amd.eval import process_prediction
    from chinesener_amd.eval.entity_eval import entity_report

    name = "bert_bilstm_crf"
    pipe = NerDataset(str(tmp_path / "data"), "msra", 32, 1, name)
    cfg = BertConfig(vocab_size=21128, hidden_size=256, num_hidden_layers=2,
                     num_attention_heads=4, intermediate_size=512)
    params = resolve_params(model_params(name), pipe.params,
                            {"bert_config": cfg, "model_name": name,
                             "num_train_steps": 1200, "warmup_ratio": 0.05,
                             "lr": 1.5e-4})
    model = build_model(name, params)
    trainer = Trainer(model, name, params, str(tmp_path / "ck"))

    def epochs():
        while True:
            yield from pipe.iter_batches("train")

    gen = epochs()
    for _ in range(800):
        trainer.train_step(next(gen))

    rows = trainer.predict(pipe.iter_batches("valid", shuffle=False))
    idx2tag = pipe.params["idx2tag"]
    proc = [process_prediction(r, idx2tag) for r in rows]
    rep = entity_report([p["label_tags"] for p in proc],
                        [p["pred_tags"] for p in proc])
    f1 = rep["micro avg"]["f1"]
    print("entity micro F1 after 800 steps:", f1)
    assert f1 > 0.22, rep["micro avg"]


def test_mrc_train_step_gpu(tmp_path):
    """MRC path on GPU: dynamic padded batch lengths through the fused
    attention kernels + masked-CE."""
    _cuda()
    torch.manual_seed(9)
    from chinesener_amd.config import resolve_params
    from chinesener_amd.models import build_model, model_params
    from chinesener_amd.models.bert import BertConfig
    from chinesener_amd.mrc.dataset import MrcDataset
    from chinesener_amd.train.trainer import Trainer
    cfg = BertConfig(vocab_size=21128, hidden_size=768, num_hidden_layers=2,
                     num_attention_heads=12, intermediate_size=3072)
    params = resolve_params(model_params("mrc_bio"), {
        "bert_config": cfg, "vocab_size": 21128, "label_size": 3,
        "num_train_steps": 100, "step_per_epoch": 10, "model_name": "mrc_bio"})
    model = build_model("mrc_bio", params)
    trainer = Trainer(model, "mrc_bio", params, str(tmp_path / "ck"))
    ds = MrcDataset(str(tmp_path / "msra"), "msra", batch_size=8,
                    max_seq_len=170)
    losses = []
    for i, batch in enumerate(ds.iter_batches("valid", shuffle=False)):
        losses.append(trainer.train_step(batch))
        if i >= 4:
            break
    assert all(torch.isfinite(torch.tensor(losses))), losses


@pytest.mark.parametrize("model_name", [
    "bilstm_crf", "bilstm_crf_softword", "bilstm_crf_ex_softword",
    "bilstm_crf_softlexicon", "bilstm_crf_bichar", "bert_ce", "bert_dice",
    "bert_crf", "bert_bilstm_crf", "bert_cnn_crf", "bert_bilstm_crf_bigram",
    "bert_bilstm_crf_softlexicon", "bert_bilstm_crf_mtl",
    "bert_bilstm_crf_adv", "transformer_crf_bichar",
    "transformer_tener_crf_bichar", "mrc_bio"])
def test_every_model_trains_on_gpu(model_name):
    """All 16 reference models (+ MRC) run forward+backward+optimizer on
    the GPU HIP path with finite losses."""
    _cuda()
    torch.manual_seed(5)
    import sys as _sys
    import os as _os
    _sys.path.insert(0, _os.path.dirname(_os.path.abspath(__file__)))
    from conftest import make_tiny_batch, make_tiny_params
    from chinesener_amd.models import build_model, optimizer_family
    from chinesener_amd.train.optimizers import build_optimizer, clip_gradients
    params = make_tiny_params(model_name)
    params.update({"num_train_steps": 100, "step_per_epoch": 10, "lr": 1e-4})
    model = build_model(model_name, params).to("cuda")
    fam = optimizer_family(model_name)
    opt, sched = build_optimizer(model, fam, params)
    mtl = "mtl" in model_name or "adv" in model_name
    batch = make_tiny_batch(model_name, mtl=mtl)
    dev = {k: v.to("cuda") for k, v in batch.items()}
    losses = []
    for step in range(1, 4):
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(dev, compute_pred=(step == 3))
        out.loss.backward()
        clip_gradients(model, fam)
        sched.apply(opt, step)
        opt.step()
        losses.append(float(out.loss.detach()))
    assert all(torch.isfinite(torch.tensor(losses))), (model_name, losses)
    assert out.pred_ids is not None


@pytest.mark.gpu
def test_device_prefetcher_gpu():
    """Pinned side-stream H2D staging delivers every batch, on-device,
    bit-identical, in order. Ring-buffer contract (allocation-free
    replay loops): a handed-out batch is valid until the NEXT two
    batches are drawn — consume as you iterate, like the train loop."""
    _cuda()
    from chinesener_amd.data.loader import DevicePrefetcher
    src = [{"token_ids": torch.randint(0, 100, (4, 32)),
            "w": torch.randn(4, 8)} for _ in range(16)]
    n = 0
    ptrs = set()
    staged = 0
    for host, dev in zip(src, DevicePrefetcher(iter(src), "cuda:0")):
        n += 1
        for k in host:
            # staging engages only after a sustained stable-shape run
            # (ragged streams must not pay pinned allocations); early
            # batches pass through as host tensors
            if dev[k].is_cuda:
                staged += 1
                ptrs.add(dev[k].data_ptr())
            assert torch.equal(dev[k].cpu(), host[k])
    assert n == 16
    assert staged >= 10   # steady state is device-staged
    # persistent 2-deep ring: 2 buffers per key, not one per batch
    assert len(ptrs) == 4, len(ptrs)
    # and through trainer.train (the wrapping call site)
    from conftest import make_tiny_batch, make_tiny_params
    from chinesener_amd.train.trainer import Trainer
    from chinesener_amd.models import build_model
    params = make_tiny_params("bilstm_crf")
    model = build_model("bilstm_crf", params)
    tr = Trainer(model, "bilstm_crf", params, "/tmp/prefetch_ck",
                 device="cuda:0")
    batches = [make_tiny_batch("bilstm_crf", batch_size=2, seed=i)
               for i in range(6)]
    res = tr.train(iter(batches), log_steps=100, save_steps=0)
    assert res["step"] == 6
