"""Global defaults + run configuration.

Mirrors the reference's config.py TRAIN_PARAMS/RUN_CONFIG
(/root/reference/config.py:5-29) but replaces import-side-effect dict
mutation with an explicit per-model registry: models
register override dicts via ``chinesener_amd.models.register``; callers
get a merged copy through ``resolve_params``.
"""
from __future__ import annotations

import copy
from typing import Any, Dict

# Defaults shared by every model (reference config.py:5-15).
TRAIN_PARAMS: Dict[str, Any] = {
    "dropout_rate": 0.2,
    "batch_size": 32,
    "epoch_size": 10,
    "max_seq_len": 150,
    "lr": 1e-3,
    "early_stop_ratio": 1.0,      # patience = ratio * steps_per_epoch
    "dtype": "bf16",              # compute dtype on GPU (fp32 on CPU)
}

RUN_CONFIG: Dict[str, Any] = {
    "summary_steps": 10,
    "log_steps": 100,
    "save_steps": 500,
    "keep_checkpoint_max": 3,
    "seed": 1234,                 # reference fixed dropout seed (tools/layer.py:57)
}

CHECKPOINT_DIR = "./checkpoint"    # ./checkpoint/ner_{data}_{model}/ (main.py:17)
EXPORT_DIR = "./serving_model"     # ./serving_model/{model}/{version}/ (tools/infer_utils.py:10)
DATA_DIR = "./data"


def resolve_params(model_params: Dict[str, Any] | None = None,
                   data_params: Dict[str, Any] | None = None,
                   cli_overrides: Dict[str, Any] | None = None) -> Dict[str, Any]:
    """Merge defaults <- per-model overrides <- dataset-derived params <- CLI.

    Matches the reference's resolution order: config.py defaults, model
    module TRAIN_PARAMS.update (model/bilstm_crf.py:55-62), then
    input_pipe.params (main.py:24-25).
    """
    params = copy.deepcopy(TRAIN_PARAMS)
    for layer in (model_params, data_params, cli_overrides):
        if layer:
            params.update(layer)
    return params
