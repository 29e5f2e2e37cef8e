"""Model registry keyed by the reference's model names.

Replaces the reference's importlib-by-name + TRAIN_PARAMS import
side-effects (tools/train_utils.py:149, model/bilstm_crf.py:55-62) with
an explicit registry: each entry is (ModelClass,
per-model param overrides). The overrides mirror each reference module's
TRAIN_PARAMS.update block; optimizer family is chosen by name substring
as the reference does (tools/train_utils.py:156-164).
"""
from __future__ import annotations

from typing import Dict, Tuple, Type

from .base import ModelOutput, NerModel, flip_gradient  # noqa: F401
from .bert import BertConfig, BertModel  # noqa: F401
from .mrc import MrcBio
from .mtl import BertBilstmCrfAdv, BertBilstmCrfMtl
from .single import (BertBilstmCrf, BertBilstmCrfBigram,
                     BertBilstmCrfSoftlexicon, BertCe, BertCnnCrf, BertCrf,
                     BertDice, BilstmCrf, BilstmCrfBichar, BilstmCrfExSoftword,
                     BilstmCrfSoftlexicon, BilstmCrfSoftword)
from .transformer_models import TransformerCrfBichar, TransformerTenerCrfBichar

_RNN_TANH = {"rnn_params": {"hidden_units_list": [128],
                            "cell_activation": "tanh", "keep_prob_list": [0.8]}}
_RNN_RELU = {"rnn_params": {"hidden_units_list": [128],
                            "cell_activation": "relu", "keep_prob_list": [0.8]}}
_RNN200 = {"rnn_params": {"hidden_units_list": [200],
                          "cell_activation": "tanh", "keep_prob_list": [0.8]}}
_TRANS = {"transformer_params": {"d_model": 160, "num_head": 8,
                                 "ffn_hidden": 320, "encode_attention_layers": 2},
          "batch_size": 16, "lr": 1e-3}
_BERT_LR = {"lr": 5e-5, "warmup_ratio": 0.1, "weight_decay": 0.01,
            # batch 64 (BASELINE config): at 32 the x500 crf/logit LR
            # groups are too noisy for from-scratch convergence (token
            # F1 plateaus ~0.36 vs entity F1 0.88+ at 64) and an MI355X
            # fits 64 trivially in 288 GB HBM3E
            "batch_size": 64}

MODELS: Dict[str, Tuple[Type[NerModel], Dict]] = {
    # name -> (class, per-model TRAIN_PARAMS overrides); cites: SURVEY.md §2.3
    "bilstm_crf": (BilstmCrf, {**_RNN_TANH, "lr": 1e-3}),
    "bilstm_crf_softword": (BilstmCrfSoftword, {**_RNN_TANH, "lr": 1e-3}),
    "bilstm_crf_ex_softword": (BilstmCrfExSoftword, {**_RNN_TANH, "lr": 1e-3}),
    "bilstm_crf_softlexicon": (BilstmCrfSoftlexicon, {**_RNN200, "lr": 1e-3}),
    "bilstm_crf_bichar": (BilstmCrfBichar, {**_RNN_TANH, "lr": 1e-3}),
    "bert_ce": (BertCe, dict(_BERT_LR)),
    "bert_dice": (BertDice, {**_BERT_LR, "alpha": 0.1, "gamma": 1.0}),
    "bert_crf": (BertCrf, dict(_BERT_LR)),
    "bert_bilstm_crf": (BertBilstmCrf, {
        **_BERT_LR, **_RNN_RELU,
        # per-layer-group differential LR (model/bert_bilstm_crf.py:44-47)
        "diff_lr_times": {"crf": 500, "logit": 500, "lstm": 100}}),
    "bert_cnn_crf": (BertCnnCrf, {
        **_BERT_LR, "cnn_params": {"filters": 128, "kernel_sizes": (2, 3, 4),
                                   "keep_prob": 0.8}}),
    "bert_bilstm_crf_bigram": (BertBilstmCrfBigram, {
        **_BERT_LR, **_RNN_TANH, "use_bert": False}),
    "bert_bilstm_crf_softlexicon": (BertBilstmCrfSoftlexicon, {
        **_BERT_LR, **_RNN200,
        "diff_lr_times": {"crf": 500, "logit": 500, "lstm": 100}}),
    "bert_bilstm_crf_mtl": (BertBilstmCrfMtl, {
        **_BERT_LR, **_RNN_RELU, "task_weight": (0.5, 0.5), "asymmetry": False,
        "diff_lr_times": {"crf": 500, "logit": 500, "lstm": 100}}),
    "bert_bilstm_crf_adv": (BertBilstmCrfAdv, {
        **_BERT_LR, **_RNN_RELU, "task_weight": (0.5, 0.5), "asymmetry": False,
        "lambda": 0.05, "shrink_gradient_reverse": 0.01,
        "diff_lr_times": {"crf": 500, "logit": 500, "lstm": 100}}),
    "transformer_crf_bichar": (TransformerCrfBichar, dict(_TRANS)),
    "transformer_tener_crf_bichar": (TransformerTenerCrfBichar, dict(_TRANS)),
    "mrc_bio": (MrcBio, {"lr": 5e-6, "max_seq_len": 170, "batch_size": 32}),
}


# Per-model extra CLI flags (reference AddonParser pattern,
# tools/train_utils.py:14-44 consumed in each model module's main block):
# the train driver appends these to argparse and merges parsed values
# into the resolved params.
from ..train.addon_parser import Addon  # noqa: E402

MODEL_ADDONS: Dict[str, list] = {
    "bert_dice": [Addon("alpha", 0.1, float, help="dice focal down-weight"),
                  Addon("gamma", 1.0, float, help="dice smoothing")],
    "bert_bilstm_crf_mtl": [
        Addon("task_weight_1", 0.5, float, help="loss weight of task 1"),
        Addon("asymmetry", None, action="store_true",
              help="feed task1 hidden into task2 tower")],
    "bert_bilstm_crf_adv": [
        Addon("task_weight_1", 0.5, float, help="loss weight of task 1"),
        Addon("lambda_adv", 0.05, float, help="adversarial loss weight"),
        Addon("shrink_gradient_reverse", 0.01, float,
              help="flip-gradient scale")],
}


def model_addons(name: str) -> list:
    return list(MODEL_ADDONS.get(name, []))


def apply_addon_values(name: str, params: Dict, values: Dict) -> Dict:
    """Merge parsed addon values into params (None = keep default)."""
    vals = {k: v for k, v in values.items() if v is not None}
    if "task_weight_1" in vals:
        w1 = float(vals.pop("task_weight_1"))
        params["task_weight"] = (w1, 1.0 - w1)
    if "lambda_adv" in vals:
        params["lambda"] = float(vals.pop("lambda_adv"))
    params.update(vals)
    return params


def model_params(name: str) -> Dict:
    if name not in MODELS:
        raise KeyError(f"unknown model '{name}' (known: {sorted(MODELS)})")
    return dict(MODELS[name][1])


def build_model(name: str, params: Dict) -> NerModel:
    cls, _ = MODELS[name]
    return cls(params)


def optimizer_family(name: str) -> str:
    """bert* -> AdamW+warmup+poly; transformer* -> Noam; else Adam+exp decay
    (reference substring dispatch, tools/train_utils.py:156-164)."""
    if name.startswith("bert") or name.startswith("mrc"):
        return "bert"
    if name.startswith("transformer"):
        return "transformer"
    return "custom"
