"""Serving export with the reference's layout:
./serving_model/{model}/{version}/ (tools/infer_utils.py:10, main.py:57-60).

The exported artifact is a torch state_dict + the resolved params needed
to rebuild the graph server-side (model name, label maps, word-enhance);
the server wraps it in a hipGraph-captured forward (serve/engine.py).
"""
from __future__ import annotations

import json
import os
import pickle
import time
from typing import Dict, Optional, Tuple

import torch

from ..config import EXPORT_DIR
from ..models import build_model


def export_model(model: torch.nn.Module, name: str, params: Dict,
                 export_root: str = EXPORT_DIR,
                 version: Optional[int] = None) -> str:
    version = version or int(time.time())
    out_dir = os.path.join(export_root, name, str(version))
    os.makedirs(out_dir, exist_ok=True)
    torch.save(model.state_dict(), os.path.join(out_dir, "model.pt"))
    # keep everything picklable (incl. BertConfig and embedding matrices —
    # the server must rebuild the exact graph, serve/engine.py)
    export_params = {}
    for k, v in params.items():
        if isinstance(v, torch.Tensor):
            v = v.detach().cpu()
        try:
            pickle.dumps(v)
        except Exception:
            continue
        export_params[k] = v
    with open(os.path.join(out_dir, "params.pkl"), "wb") as f:
        pickle.dump({"model_name": params.get("model_name", name),
                     "params": export_params}, f)
    with open(os.path.join(out_dir, "meta.json"), "w") as f:
        json.dump({"model": name, "version": version,
                   "exported_at": time.strftime("%Y-%m-%d %H:%M:%S")}, f)
    return out_dir


def latest_version_dir(name: str, export_root: str = EXPORT_DIR) -> str:
    base = os.path.join(export_root, name)
    versions = sorted(int(v) for v in os.listdir(base) if v.isdigit())
    if not versions:
        raise FileNotFoundError(f"no exported versions under {base}")
    return os.path.join(base, str(versions[-1]))


def load_exported(name: str, export_root: str = EXPORT_DIR,
                  device: str = "cpu",
                  version: Optional[int] = None) -> Tuple[torch.nn.Module, Dict]:
    if version is not None:
        vdir = os.path.join(export_root, name, str(version))
    else:
        vdir = latest_version_dir(name, export_root)
    with open(os.path.join(vdir, "params.pkl"), "rb") as f:
        meta = pickle.load(f)
    model_name = meta["model_name"]
    params = meta["params"]
    model = build_model(model_name, params)
    state = torch.load(os.path.join(vdir, "model.pt"), map_location=device,
                       weights_only=False)
    model.load_state_dict(state)
    model.to(device).eval()
    return model, params
