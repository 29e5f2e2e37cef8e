"""Checkpoint/resume with the reference's layout and semantics
(SURVEY.md §5.4): ./checkpoint/ner_{data}_{model}/ dir (main.py:17,69),
keep_checkpoint_max=3 (config.py:21-22), warm-start iff a checkpoint
exists (tools/utils.py:49-61), --clear_model rm-rf (tools/utils.py:16-22).
"""
from __future__ import annotations

import os
import re
import shutil
from typing import Dict, Optional

import torch


def ckpt_dir(data_names: str, model_name: str, root: str = "./checkpoint") -> str:
    return os.path.join(root, f"ner_{data_names}_{model_name}")


def clear_model(path: str) -> None:
    if os.path.exists(path):
        shutil.rmtree(path)


class CheckpointManager:
    def __init__(self, directory: str, keep_max: int = 3):
        self.dir = directory
        self.keep_max = keep_max
        os.makedirs(directory, exist_ok=True)

    def _paths(self):
        out = []
        for f in os.listdir(self.dir):
            m = re.match(r"ckpt-(\d+)\.pt$", f)
            if m:
                out.append((int(m.group(1)), os.path.join(self.dir, f)))
        return sorted(out)

    def latest(self) -> Optional[str]:
        paths = self._paths()
        return paths[-1][1] if paths else None

    def save(self, step: int, model: torch.nn.Module,
             optimizer: Optional[torch.optim.Optimizer] = None,
             extra: Optional[Dict] = None) -> str:
        state = {"step": step, "model": model.state_dict()}
        if optimizer is not None:
            state["optimizer"] = optimizer.state_dict()
        if extra:
            state["extra"] = extra
        path = os.path.join(self.dir, f"ckpt-{step}.pt")
        torch.save(state, path)
        for _, old in self._paths()[:-self.keep_max]:
            os.remove(old)
        return path

    def restore(self, model: torch.nn.Module,
                optimizer: Optional[torch.optim.Optimizer] = None,
                map_location="cpu") -> int:
        """Warm-start if present (reference glob check). Returns step (0 if none)."""
        path = self.latest()
        if path is None:
            return 0
        state = torch.load(path, map_location=map_location, weights_only=False)
        model.load_state_dict(state["model"])
        if optimizer is not None and "optimizer" in state:
            optimizer.load_state_dict(state["optimizer"])
        return int(state.get("step", 0))
