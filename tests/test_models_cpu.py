"""Every registered model: forward + backward on a tiny CPU config."""
import pytest
import torch

from chinesener_amd.models import MODELS, build_model

from conftest import make_tiny_batch, make_tiny_params

SINGLE_TASK = [n for n in MODELS if n not in
               ("bert_bilstm_crf_mtl", "bert_bilstm_crf_adv", "mrc_bio")]


@pytest.mark.parametrize("name", SINGLE_TASK)
def test_model_forward_backward(name):
    torch.manual_seed(0)
    params = make_tiny_params(name)
    model = build_model(name, params)
    batch = make_tiny_batch(name)
    out = model(batch)
    assert out.loss is not None and torch.isfinite(out.loss), name
    out.loss.backward()
    grads = [p.grad for p in model.parameters() if p.requires_grad]
    assert any(g is not None and g.abs().sum() > 0 for g in grads), name


@pytest.mark.parametrize("name", SINGLE_TASK)
def test_model_predict(name):
    torch.manual_seed(0)
    params = make_tiny_params(name)
    model = build_model(name, params).eval()
    batch = make_tiny_batch(name)
    with torch.no_grad():
        out = model(batch, compute_pred=True)
    assert out.pred_ids.shape == batch["token_ids"].shape
    # padded positions decode to 0
    assert (out.pred_ids * (1 - batch["mask"])).sum() == 0


@pytest.mark.parametrize("name", ["bert_bilstm_crf_mtl", "bert_bilstm_crf_adv"])
def test_mtl_models(name):
    torch.manual_seed(0)
    params = make_tiny_params(name)
    model = build_model(name, params)
    batch = make_tiny_batch(name, mtl=True)
    out = model(batch)
    assert torch.isfinite(out.loss)
    out.loss.backward()
    # shared BERT must receive grads from both towers
    bert_grads = [p.grad for n, p in model.named_parameters()
                  if n.startswith("bert.") and p.grad is not None]
    assert bert_grads and any(g.abs().sum() > 0 for g in bert_grads)
    with torch.no_grad():
        out = model(batch, compute_pred=True)
    assert out.pred_ids.shape == batch["token_ids"].shape


def test_adv_flip_gradient_direction():
    """Discriminator loss must push shared encoder the OPPOSITE way."""
    from chinesener_amd.models.base import flip_gradient
    x = torch.randn(3, 4, requires_grad=True)
    y = flip_gradient(x, 0.5)
    y.sum().backward()
    torch.testing.assert_close(x.grad, torch.full_like(x, -0.5))


def test_mrc_model():
    torch.manual_seed(0)
    params = make_tiny_params("mrc_bio")
    model = build_model("mrc_bio", params)
    B, L = 2, 16
    batch = {
        "token_ids": torch.randint(4, 200, (B, L)),
        "mask": torch.ones(B, L, dtype=torch.long),
        "segment_ids": (torch.arange(L)[None, :] >= 6).long().expand(B, L),
        "text_mask": (torch.arange(L)[None, :] >= 6).long().expand(B, L),
        "label_ids": torch.randint(0, 3, (B, L)),
    }
    out = model(batch)
    assert torch.isfinite(out.loss)
    out.loss.backward()


def test_bigram_use_bert_variant():
    params = make_tiny_params("bert_bilstm_crf_bigram")
    params["use_bert"] = True
    model = build_model("bert_bilstm_crf_bigram", params)
    batch = make_tiny_batch("bert_bilstm_crf_bigram")
    out = model(batch)
    assert torch.isfinite(out.loss)


def test_tf_bert_name_mapping_loader():
    """TF-namespace weights load into BertModel with fused qkv and
    produce the mapped values (reference init_from_checkpoint parity,
    tools/train_utils.py:91-102)."""
    import numpy as np
    import torch
    from chinesener_amd.models.bert import BertConfig, BertModel
    from chinesener_amd.models.bert_loader import load_tf_bert
    H, I, V = 32, 64, 100
    cfg = BertConfig(vocab_size=V, hidden_size=H, num_hidden_layers=2,
                     num_attention_heads=2, intermediate_size=I,
                     max_position_embeddings=40)
    model = BertModel(cfg)
    rng = np.random.default_rng(0)
    tf = {"bert/embeddings/word_embeddings": rng.normal(size=(V, H)),
          "bert/embeddings/position_embeddings": rng.normal(size=(40, H)),
          "bert/embeddings/token_type_embeddings": rng.normal(size=(2, H)),
          "bert/embeddings/LayerNorm/gamma": rng.normal(size=(H,)),
          "bert/embeddings/LayerNorm/beta": rng.normal(size=(H,))}
    for i in range(2):
        p = f"bert/encoder/layer_{i}"
        for name, shape in [("attention/self/query", (H, H)),
                            ("attention/self/key", (H, H)),
                            ("attention/self/value", (H, H)),
                            ("attention/output/dense", (H, H)),
                            ("intermediate/dense", (H, I)),
                            ("output/dense", (I, H))]:
            tf[f"{p}/{name}/kernel"] = rng.normal(size=shape)
            tf[f"{p}/{name}/bias"] = rng.normal(size=(shape[1],))
        for ln in ("attention/output/LayerNorm", "output/LayerNorm"):
            tf[f"{p}/{ln}/gamma"] = rng.normal(size=(H,))
            tf[f"{p}/{ln}/beta"] = rng.normal(size=(H,))
    missing, unexpected = load_tf_bert(model, tf)
    assert not unexpected
    # qkv fused correctly: query kernel^T is the first H rows
    q_t = torch.as_tensor(tf["bert/encoder/layer_0/attention/self/query/kernel"]).T
    torch.testing.assert_close(model.layers[0].qkv.weight[:H].double(),
                               q_t.double())
    v_bias = torch.as_tensor(tf["bert/encoder/layer_0/attention/self/value/bias"])
    torch.testing.assert_close(model.layers[0].qkv.bias[2 * H:].double(),
                               v_bias.double())
    emb = torch.as_tensor(tf["bert/embeddings/word_embeddings"])
    torch.testing.assert_close(model.embeddings.word.weight.double(), emb.double())


def test_tf_checkpoint_bundle_roundtrip_and_bert_load(tmp_path):
    """TF v2 tensor-bundle file format: write a bert-base-google-shaped
    checkpoint (the FULL TF variable inventory incl. pooler/cls heads,
    as in /root/reference pretrain_model/ch_google) with the in-tree
    writer, read it back with the from-scratch parser, and load it into
    BertModel via the name mapping with zero unexpected keys."""
    import numpy as np
    import torch
    from chinesener_amd.models.bert import BertConfig, BertModel
    from chinesener_amd.models.bert_loader import load_tf_bert
    from chinesener_amd.models.tf_checkpoint import (read_tf_checkpoint,
                                                     write_tf_checkpoint)

    H, I, V, LYR = 24, 48, 120, 2
    rng = np.random.default_rng(9)
    tensors = {
        "bert/embeddings/word_embeddings": rng.normal(size=(V, H)),
        "bert/embeddings/position_embeddings": rng.normal(size=(40, H)),
        "bert/embeddings/token_type_embeddings": rng.normal(size=(2, H)),
        "bert/embeddings/LayerNorm/gamma": rng.normal(size=(H,)),
        "bert/embeddings/LayerNorm/beta": rng.normal(size=(H,)),
        # heads a real bert_model.ckpt carries (ignored by the mapping)
        "bert/pooler/dense/kernel": rng.normal(size=(H, H)),
        "bert/pooler/dense/bias": rng.normal(size=(H,)),
        "cls/predictions/output_bias": rng.normal(size=(V,)),
        "cls/predictions/transform/dense/kernel": rng.normal(size=(H, H)),
        "cls/predictions/transform/dense/bias": rng.normal(size=(H,)),
        "cls/predictions/transform/LayerNorm/gamma": rng.normal(size=(H,)),
        "cls/predictions/transform/LayerNorm/beta": rng.normal(size=(H,)),
        "cls/seq_relationship/output_weights": rng.normal(size=(2, H)),
        "cls/seq_relationship/output_bias": rng.normal(size=(2,)),
        "global_step": np.array(123, dtype=np.int64),
    }
    for i in range(LYR):
        p = f"bert/encoder/layer_{i}"
        for nm, shape in [("attention/self/query", (H, H)),
                          ("attention/self/key", (H, H)),
                          ("attention/self/value", (H, H)),
                          ("attention/output/dense", (H, H)),
                          ("intermediate/dense", (H, I)),
                          ("output/dense", (I, H))]:
            tensors[f"{p}/{nm}/kernel"] = rng.normal(size=shape)
            tensors[f"{p}/{nm}/bias"] = rng.normal(size=(shape[1],))
        for ln in ("attention/output/LayerNorm", "output/LayerNorm"):
            tensors[f"{p}/{ln}/gamma"] = rng.normal(size=(H,))
            tensors[f"{p}/{ln}/beta"] = rng.normal(size=(H,))
    tensors = {k: (v.astype(np.float32) if v.dtype == np.float64 else v)
               for k, v in tensors.items()}

    prefix = str(tmp_path / "bert_model.ckpt")
    write_tf_checkpoint(prefix, tensors)
    back = read_tf_checkpoint(prefix)
    assert set(back) == set(tensors), (
        set(tensors) - set(back), set(back) - set(tensors))
    for k, v in tensors.items():
        got = back[k]
        assert got.shape == tuple(np.shape(v)), k
        np.testing.assert_array_equal(np.asarray(got), np.asarray(v), err_msg=k)

    cfg = BertConfig(vocab_size=V, hidden_size=H, num_hidden_layers=LYR,
                     num_attention_heads=2, intermediate_size=I,
                     max_position_embeddings=40)
    model = BertModel(cfg)
    missing, unexpected = load_tf_bert(model, back)
    assert not unexpected, unexpected
    assert not missing, missing      # every model tensor received a value
    q = torch.as_tensor(
        tensors["bert/encoder/layer_1/attention/self/query/kernel"]).T
    torch.testing.assert_close(model.layers[1].qkv.weight[:H].double(),
                               q.double())
