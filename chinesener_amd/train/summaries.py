"""Layer activation summaries (reference tools/utils.py:25-27
``add_layer_summary``: zero-fraction scalar + activation histogram per
layer, written every ``summary_steps``:10 — config.py:19).

TensorBoard is not part of this stack; summaries are JSONL records in
``{model_dir}/summaries.jsonl`` (step, per-module zero_fraction / mean /
std / absmax + a fixed 16-bin histogram), cheap to grep or plot.

Hooks only compute when armed, so steady-state steps pay one boolean
check per module. hipGraph-replayed steps bypass Python hooks entirely;
summaries therefore sample eager steps only (the capture-warmup and any
shape-miss steps), which matches their debugging purpose.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional

import torch

HIST_BINS = 16


def tensor_stats(t: torch.Tensor) -> Dict:
    """Zero-fraction + moments + fixed-bin histogram (host-side floats)."""
    f = t.detach().float()
    absmax = float(f.abs().max()) if f.numel() else 0.0
    hist = torch.histc(f, bins=HIST_BINS, min=-absmax or -1.0,
                       max=absmax or 1.0)
    return {
        "zero_fraction": float((f == 0).float().mean()),
        "mean": float(f.mean()),
        "std": float(f.std()) if f.numel() > 1 else 0.0,
        "absmax": absmax,
        "hist": [int(c) for c in hist.tolist()],
    }


class SummaryLogger:
    """Forward-hook activation summaries on leaf modules.

    Usage::

        sl = SummaryLogger(model, model_dir, every=10)
        sl.maybe_arm(step)     # before forward
        ...forward...
        sl.flush(step)         # after forward (writes if armed)
    """

    def __init__(self, model: torch.nn.Module, model_dir: str,
                 every: int = 10, max_modules: int = 64):
        self.every = max(1, every)
        self.path = os.path.join(model_dir, "summaries.jsonl")
        os.makedirs(model_dir, exist_ok=True)
        self.armed = False
        self._records: Dict[str, torch.Tensor] = {}
        self._handles: List = []
        n = 0
        for name, mod in model.named_modules():
            if len(list(mod.children())):
                continue            # leaves only
            if n >= max_modules:
                break
            n += 1
            self._handles.append(mod.register_forward_hook(
                self._make_hook(name or mod.__class__.__name__)))

    def _make_hook(self, name: str):
        def hook(_mod, _inp, out):
            if not self.armed:
                return
            t = out[0] if isinstance(out, (tuple, list)) else out
            if isinstance(t, torch.Tensor) and t.is_floating_point():
                self._records[name] = t     # stats deferred to flush
        return hook

    def maybe_arm(self, step: int) -> bool:
        self.armed = step % self.every == 0
        self._records = {}
        return self.armed

    def flush(self, step: int) -> Optional[str]:
        if not self.armed:
            return None
        rec = {"step": step,
               "layers": {k: tensor_stats(v) for k, v in self._records.items()}}
        with open(self.path, "a") as f:
            f.write(json.dumps(rec) + "\n")
        self.armed = False
        self._records = {}
        return self.path

    def flush_artifacts(self, model: torch.nn.Module, step: int) -> str:
        """Dump the visual-summary artifacts the reference renders as
        TensorBoard images (CRF transition matrix image+histogram,
        tools/layer.py:129-130; attention image summaries,
        tools/transformer/modules.py:128): the raw matrices land in
        ``{model_dir}/artifacts/step_{N}.npz`` for offline plotting —
        no TensorBoard in this stack, the npz IS the artifact."""
        art_dir = os.path.join(os.path.dirname(self.path), "artifacts")
        os.makedirs(art_dir, exist_ok=True)
        import numpy as np
        blobs = {}
        for name, p in model.named_parameters():
            low = name.lower()
            if "transitions" in low:          # CRF transition matrices
                blobs[name] = p.detach().float().cpu().numpy()
            elif low.endswith(("u", "v")) and p.dim() == 2 and p.shape[0] <= 16:
                blobs[name] = p.detach().float().cpu().numpy()  # TENER u/v
        path = os.path.join(art_dir, f"step_{step}.npz")
        np.savez(path, **blobs) if blobs else None
        return path

    def close(self):
        for h in self._handles:
            h.remove()
        self._handles = []
