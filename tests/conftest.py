import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_bert_config():
    from chinesener_amd.models.bert import BertConfig
    return BertConfig(vocab_size=200, hidden_size=32, num_hidden_layers=2,
                      num_attention_heads=2, intermediate_size=64,
                      max_position_embeddings=64)


def make_tiny_params(model_name: str, label_size: int = 10, seq_len: int = 16):
    """Params dict small enough for CPU forward/backward of any model."""
    from chinesener_amd.models.bert import BertConfig
    import numpy as np
    tag2idx = {"[PAD]": 0, "O": 1, "B-LOC": 2, "I-LOC": 3, "B-PER": 4,
               "I-PER": 5, "B-ORG": 6, "I-ORG": 7, "[CLS]": 8, "[SEP]": 9}
    p = {
        "vocab_size": 200, "label_size": label_size, "max_seq_len": seq_len,
        "embedding_dim": 16, "word_dim": 8, "word_vocab_size": 50,
        "bichar_vocab_size": 100, "bichar_dim": 8,
        "bert_config": BertConfig(vocab_size=200, hidden_size=32,
                                  num_hidden_layers=2, num_attention_heads=2,
                                  intermediate_size=64,
                                  max_position_embeddings=64),
        "rnn_params": {"hidden_units_list": [12], "cell_activation": "tanh",
                       "keep_prob_list": [1.0]},
        "transformer_params": {"d_model": 16, "num_head": 2, "ffn_hidden": 32,
                               "encode_attention_layers": 1},
        "dropout_rate": 0.0, "embedding_dropout": 0.0,
        "tag2idx": tag2idx,
        "idx2tag": {v: k for k, v in tag2idx.items()},
        "task_list": ["msra", "msr"],
        "msra": {"label_size": label_size},
        "msr": {"label_size": 7},
    }
    return p


def make_tiny_batch(model_name: str, batch_size: int = 3, seq_len: int = 16,
                    label_size: int = 10, seed: int = 0, mtl: bool = False):
    from chinesener_amd.data.loader import make_synthetic_batch
    from chinesener_amd.data.preprocess import extract_prefix_surfix
    enhance, tok = extract_prefix_surfix(model_name)
    b = make_synthetic_batch(batch_size, seq_len, label_size, vocab_size=200,
                             word_enhance=enhance, seed=seed,
                             is_bert=(tok == "bert"))
    if model_name.startswith("transformer"):
        import torch as _t
        g = _t.Generator().manual_seed(seed)
        b["bichar_ids"] = _t.randint(0, 100, (batch_size, seq_len), generator=g)
    if enhance == "softlexicon":
        b["softlexicon_ids"] = b["softlexicon_ids"] % 50
    if enhance == "bichar":
        b["bichar_ids"] = b["bichar_ids"] % 100
    if mtl:
        import torch as _t
        task = _t.arange(batch_size) % 2
        b["task_ids"] = task[:, None].expand(batch_size, seq_len).clone()
        b["label_ids"] = b["label_ids"].clamp(max=6)  # valid for both towers
    return b
