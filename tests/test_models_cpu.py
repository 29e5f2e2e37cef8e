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
