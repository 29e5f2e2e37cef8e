"""Pretrained-BERT weight loading by name mapping.

The reference initializes its BERT scope from a TF checkpoint via
get_assignment_map_from_checkpoint + init_from_checkpoint
(tools/train_utils.py:91-102), which pins TF variable names like
``bert/encoder/layer_0/attention/self/query/kernel``. This module maps
that namespace (or the HuggingFace torch equivalent) onto our module
tree — including fusing the separate query/key/value projections into
the packed ``qkv`` linear.

Accepts a dict of numpy arrays / torch tensors keyed by TF names
(e.g. loaded from an npz export of a TF checkpoint — no TF needed).
"""
from __future__ import annotations

from typing import Dict, Iterable, Tuple

import numpy as np
import torch


def _t(arr) -> torch.Tensor:
    return torch.as_tensor(np.asarray(arr))


def map_tf_bert_weights(tf_weights: Dict[str, "np.ndarray"],
                        num_layers: int) -> Dict[str, torch.Tensor]:
    """TF name space -> our BertModel state_dict (qkv fused)."""
    out: Dict[str, torch.Tensor] = {}
    g = tf_weights

    def has(k):
        return k in g

    emb = "bert/embeddings"
    if has(f"{emb}/word_embeddings"):
        out["embeddings.word.weight"] = _t(g[f"{emb}/word_embeddings"])
    if has(f"{emb}/position_embeddings"):
        out["embeddings.position.weight"] = _t(g[f"{emb}/position_embeddings"])
    if has(f"{emb}/token_type_embeddings"):
        out["embeddings.token_type.weight"] = _t(g[f"{emb}/token_type_embeddings"])
    if has(f"{emb}/LayerNorm/gamma"):
        out["embeddings.ln_weight"] = _t(g[f"{emb}/LayerNorm/gamma"])
        out["embeddings.ln_bias"] = _t(g[f"{emb}/LayerNorm/beta"])

    for i in range(num_layers):
        src = f"bert/encoder/layer_{i}"
        dst = f"layers.{i}"
        # TF kernels are [in, out]; torch Linear weights are [out, in]
        qw = _t(g[f"{src}/attention/self/query/kernel"]).T
        kw = _t(g[f"{src}/attention/self/key/kernel"]).T
        vw = _t(g[f"{src}/attention/self/value/kernel"]).T
        out[f"{dst}.qkv.weight"] = torch.cat([qw, kw, vw], dim=0)
        out[f"{dst}.qkv.bias"] = torch.cat([
            _t(g[f"{src}/attention/self/query/bias"]),
            _t(g[f"{src}/attention/self/key/bias"]),
            _t(g[f"{src}/attention/self/value/bias"])])
        out[f"{dst}.attn_out.weight"] = \
            _t(g[f"{src}/attention/output/dense/kernel"]).T
        out[f"{dst}.attn_out.bias"] = _t(g[f"{src}/attention/output/dense/bias"])
        out[f"{dst}.ln1_w"] = _t(g[f"{src}/attention/output/LayerNorm/gamma"])
        out[f"{dst}.ln1_b"] = _t(g[f"{src}/attention/output/LayerNorm/beta"])
        out[f"{dst}.ffn_in.weight"] = _t(g[f"{src}/intermediate/dense/kernel"]).T
        out[f"{dst}.ffn_in_bias"] = _t(g[f"{src}/intermediate/dense/bias"])
        out[f"{dst}.ffn_out.weight"] = _t(g[f"{src}/output/dense/kernel"]).T
        out[f"{dst}.ffn_out.bias"] = _t(g[f"{src}/output/dense/bias"])
        out[f"{dst}.ln2_w"] = _t(g[f"{src}/output/LayerNorm/gamma"])
        out[f"{dst}.ln2_b"] = _t(g[f"{src}/output/LayerNorm/beta"])
    return out


def load_tf_bert(bert_model: torch.nn.Module,
                 tf_weights: Dict[str, "np.ndarray"],
                 strict: bool = False) -> Tuple[Iterable[str], Iterable[str]]:
    """Load mapped TF weights into a BertModel; returns (missing,
    unexpected) like load_state_dict. Non-strict by default so a
    checkpoint without e.g. token_type still loads (reference warm-start
    semantics, tools/utils.py:49-54)."""
    mapped = map_tf_bert_weights(tf_weights, len(bert_model.layers))
    result = bert_model.load_state_dict(mapped, strict=strict)
    return result.missing_keys, result.unexpected_keys


def load_npz_bert(bert_model: torch.nn.Module, path: str) -> None:
    """Convenience: npz file whose keys are TF variable names (the
    offline export format scripts/export_tf_bert.py documents)."""
    blob = np.load(path)
    load_tf_bert(bert_model, {k: blob[k] for k in blob.files})
