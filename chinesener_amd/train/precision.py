"""Pure-bf16 training mode for MI355X.

Instead of autocast (which launches ~190 weight-cast kernels per step on
BERT-base), the model's weights live in bf16 and the fused optimizer
keeps fp32 master copies (csrc/adam.hip). Numerically-sensitive params
stay fp32:

* LayerNorm affine params (``ln*`` names) — the LN kernels compute fp32
  statistics and accept fp32 affine directly;
* CRF ``transitions`` — the CRF forward-backward kernel is fp32;
* pretrained/frozen embedding constants registered as fp32 buffers.
"""
from __future__ import annotations

import torch


def _keep_fp32(name: str) -> bool:
    last = name.split(".")[-1]
    return last.startswith("ln") or last == "transitions"


def convert_bf16_mixed(model: torch.nn.Module) -> torch.nn.Module:
    """Cast parameters to bf16 except LN affines / CRF transitions."""
    keep = {name: p.data.clone() for name, p in model.named_parameters()
            if _keep_fp32(name)}
    model.to(torch.bfloat16)
    for name, p in model.named_parameters():
        if name in keep:
            p.data = keep[name]
    return model


def wants_pure_bf16(params: dict, device: str) -> bool:
    return device.startswith("cuda") and params.get("dtype", "bf16") == "bf16"
