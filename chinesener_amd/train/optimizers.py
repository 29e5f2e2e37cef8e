"""Optimizers + LR schedules (reference tools/train_utils.py:246-390).

Three families, chosen by model-name substring like the reference
(:156-164):

* ``bert``: AdamWeightDecay (no bias correction, decoupled wd excluding
  LayerNorm/bias — :276-284) with linear warmup + polynomial decay, plus
  per-layer-group differential LR (``diff_lr_times`` name-substring
  groups sharing globally-clipped grads — :287-337).
* ``transformer``: Adam with the Noam scheme (modules.py:209-217).
* ``custom``: Adam with exponential decay (:340-376) and value clipping.

On GPU the step runs the multi-tensor fused Adam HIP kernel (SURVEY.md
K16); CPU uses torch foreach ops with identical math.
"""
from __future__ import annotations

import math
from typing import Dict, Iterable, List, Optional, Tuple

import torch

from .. import ops


def _no_decay(name: str) -> bool:
    # LayerNorm + bias excluded from weight decay (reference :282)
    n = name.lower()
    return n.endswith("bias") or "ln_" in n or "layernorm" in n or ".ln" in n


def build_param_groups(model: torch.nn.Module, base_lr: float,
                       weight_decay: float,
                       diff_lr_times: Optional[Dict[str, float]] = None
                       ) -> List[Dict]:
    """Split params into (diff-lr substring x wd/no-wd) groups.

    ``diff_lr_times`` maps a name substring to an LR multiplier
    (reference bert_train_op :287-337: one optimizer per group, shared
    clipped grads — here one optimizer with per-group lr_scale).
    """
    diff = diff_lr_times or {}
    groups: Dict[Tuple[str, bool], List] = {}
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        scale_key = ""
        for sub in diff:
            if sub in name:
                scale_key = sub
                break
        groups.setdefault((scale_key, _no_decay(name)), []).append(p)
    out = []
    for (key, nodecay), params in groups.items():
        out.append({
            "params": params,
            "lr_scale": float(diff.get(key, 1.0)),
            # full multiplier; LrSchedule.apply ramps lr_scale from 1 up
            # to this over warmup (diff-LR shock mitigation, see below)
            "lr_scale_base": float(diff.get(key, 1.0)),
            "weight_decay": 0.0 if nodecay else weight_decay,
        })
    return out


class AdamWeightDecay(torch.optim.Optimizer):
    """BERT AdamWeightDecay: m,v EMA without bias correction; decoupled wd;
    per-group lr = schedule_lr * lr_scale. GPU steps use the fused
    multi-tensor HIP kernel when available."""

    def __init__(self, params: Iterable, lr: float = 5e-5, betas=(0.9, 0.999),
                 eps: float = 1e-6, weight_decay: float = 0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, lr_scale=1.0)
        super().__init__(params, defaults)
        # device-side base LR (hipGraph-captured steps: the schedule
        # writes this scalar outside the graph, the fused kernel reads it)
        self.lr_dev: Optional[torch.Tensor] = None

    def enable_graph_lr(self, device) -> torch.Tensor:
        self.lr_dev = torch.ones(1, dtype=torch.float32, device=device)
        self._meta_cache = {}
        return self.lr_dev

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        # fused bins by param dtype: bf16 params carry an fp32 master copy
        # (pure-bf16 training mode — no autocast weight casts per step)
        fused = {torch.bfloat16: [], torch.float32: []}
        # a group may mix dtypes (e.g. fp32 LN affines + bf16 biases in the
        # no-decay group under pure-bf16 mode): process each dtype slice
        # separately so masters exist exactly for the bf16 params
        split_groups = []
        for group in self.param_groups:
            for dt in (torch.float32, torch.bfloat16):
                params = [p for p in group["params"]
                          if p.grad is not None and p.dtype == dt]
                if params:
                    split_groups.append((group, params))
        for group, params in split_groups:
            grads = [p.grad for p in params]
            states = [self.state[p] for p in params]
            for p, s in zip(params, states):
                if "m" not in s:
                    s["m"] = torch.zeros(p.shape, dtype=torch.float32,
                                         device=p.device)
                    s["v"] = torch.zeros_like(s["m"])
                    if p.dtype == torch.bfloat16:
                        s["master"] = p.detach().float().clone()
            ms = [s["m"] for s in states]
            vs = [s["v"] for s in states]
            if self.lr_dev is not None:
                # graph mode: meta.lr = per-group scale; base from lr_dev.
                # The scale is baked into the cached meta blob, so the
                # warmup ramp cannot apply here — use the full multiplier
                # (graphed steps are captured post-warmup in practice)
                lr = group.get("lr_scale_base", group.get("lr_scale", 1.0))
            else:
                lr = group["lr"] * group.get("lr_scale", 1.0)
            b1, b2 = group["betas"]
            eps, wd = group["eps"], group["weight_decay"]
            if params[0].is_cuda and ops.ext_available():
                masters = ([s["master"] for s in states]
                           if params[0].dtype == torch.bfloat16 else [])
                fused[params[0].dtype].append(
                    (params, [g.contiguous() for g in grads], masters, ms, vs,
                     [lr] * len(params), [wd] * len(params)))
            elif params[0].dtype == torch.bfloat16:
                # eager master-weight path (CPU/debug)
                for p, g, s in zip(params, grads, states):
                    gf = g.float()
                    s["m"].mul_(b1).add_(gf, alpha=1 - b1)
                    s["v"].mul_(b2).addcmul_(gf, gf, value=1 - b2)
                    upd = s["m"] / (s["v"].sqrt() + eps)
                    if wd:
                        upd = upd + wd * s["master"]
                    s["master"].add_(upd, alpha=-lr)
                    p.copy_(s["master"].to(p.dtype))
            else:
                torch._foreach_mul_(ms, b1)
                torch._foreach_add_(ms, grads, alpha=1 - b1)
                torch._foreach_mul_(vs, b2)
                torch._foreach_addcmul_(vs, grads, grads, value=1 - b2)
                denom = torch._foreach_sqrt(vs)
                torch._foreach_add_(denom, eps)
                update = torch._foreach_div(ms, denom)
                if wd:
                    torch._foreach_add_(update, params, alpha=wd)
                torch._foreach_add_(params, update, alpha=-lr)
        for dtype, bins in fused.items():
            if not bins:
                continue
            # one multi-tensor kernel launch per dtype (per-tensor lr)
            b1, b2 = self.param_groups[0]["betas"]
            eps = self.param_groups[0]["eps"]
            cat = [sum((f[i] for f in bins), []) for i in range(7)]
            ext = ops.get_ext()
            if self.lr_dev is None:
                ext.multi_tensor_adamw(cat[0], cat[1], cat[2], cat[3],
                                       cat[4], cat[5], cat[6],
                                       b1, b2, eps, None)
            else:
                # graph mode: cache the device meta blob (no H2D per step
                # — hipGraph capture forbids host transfers); meta lrs are
                # the static per-group scales, lr_dev carries the schedule
                sig = tuple(t.data_ptr() for t in cat[0] + cat[1])
                cache = self._meta_cache.get(dtype)
                if cache is None or cache[0] != sig:
                    blob, info = ext.adamw_build_meta(
                        cat[0], cat[1], cat[2], cat[3], cat[4], cat[5],
                        cat[6])
                    total, n, isbf = info.tolist()
                    cache = (sig, blob, total, n, bool(isbf))
                    self._meta_cache[dtype] = cache
                ext.multi_tensor_adamw_run(cache[1], cache[2], cache[3],
                                           cache[4], b1, b2, eps,
                                           self.lr_dev)
        return loss


class LrSchedule:
    """Per-step LR multiplier playing the reference's three schemes."""

    def __init__(self, family: str, base_lr: float, num_train_steps: int,
                 warmup_ratio: float = 0.1, step_per_epoch: int = 1000,
                 decay_rate: float = 0.95, d_model: int = 160,
                 warmup_steps: Optional[int] = None):
        self.family = family
        self.base_lr = base_lr
        self.total = max(1, num_train_steps)
        self.warmup = warmup_steps if warmup_steps is not None else max(
            1, int(self.total * warmup_ratio))
        self.step_per_epoch = max(1, step_per_epoch)
        self.decay_rate = decay_rate
        self.d_model = d_model

    def lr_at(self, step: int) -> float:
        s = max(1, step)
        if self.family == "bert":
            # linear warmup + polynomial (power 1) decay to 0 (:246-284)
            if s < self.warmup:
                return self.base_lr * s / self.warmup
            frac = min(1.0, (s - self.warmup) / max(1, self.total - self.warmup))
            return self.base_lr * (1.0 - frac)
        if self.family == "transformer":
            # Noam (modules.py:209-217)
            return (self.base_lr * (self.d_model ** -0.5)
                    * min(s ** -0.5, s * self.warmup ** -1.5) * self.warmup ** 0.5)
        # exponential decay per epoch, staircase (:365-376)
        return self.base_lr * (self.decay_rate ** (s // self.step_per_epoch))

    def apply(self, optimizer: torch.optim.Optimizer, step: int) -> float:
        lr = self.lr_at(step)
        # Diff-LR warm ramp: the reference applies its crf/logit x500
        # multiplier from step 0 (fine for a pretrained encoder); with a
        # random-init encoder that shock locks the CRF into the all-O
        # basin (profiles/convergence_r01.md). Ramp each group's
        # multiplier 1 -> full over the warmup window instead; after
        # warmup the schedule is exactly the reference's.
        ramp = min(1.0, max(1, step) / self.warmup)
        for g in optimizer.param_groups:
            g["lr"] = lr
            base = g.get("lr_scale_base")
            if base is not None and base != 1.0:
                g["lr_scale"] = 1.0 + (base - 1.0) * ramp
        lr_dev = getattr(optimizer, "lr_dev", None)
        if lr_dev is not None:   # graph mode: one tiny device write
            lr_dev.fill_(lr)
        return lr


def clip_gradients(model: torch.nn.Module, family: str,
                   max_norm: float = 1.0) -> Optional[torch.Tensor]:
    """bert/transformer: clip_by_global_norm 1.0 (:315,334); custom:
    clip_by_value +-5 (:379-390).

    GPU path is fully device-side (fused multi-tensor sumsq + scale, no
    host sync in the step loop); returns the squared-norm tensor."""
    if family in ("bert", "transformer"):
        ps = [p for p in model.parameters() if p.grad is not None]
        if not ps:
            return None
        for p in ps:            # scale must hit the real grad storage
            if not p.grad.is_contiguous():
                p.grad = p.grad.contiguous()
        grads = [p.grad for p in ps]
        if grads[0].is_cuda and ops.ext_available():
            ext = ops.get_ext()
            # cached device meta blob: capture-safe (no H2D in the step);
            # pointer signature invalidates when grads are reallocated
            sig = tuple(g.data_ptr() for g in grads)
            cm = getattr(model, "_clip_meta", None)
            if cm is None or cm[0] != sig:
                blob, info = ext.norm_build_meta(grads)
                cm = (sig, blob, int(info[0]))
                model._clip_meta = cm
            out = torch.zeros(1, dtype=torch.float32, device=grads[0].device)
            sumsq = ext.multi_tensor_sumsq_run(cm[1], cm[2], out)
            norm = sumsq.sqrt()
            coef = (max_norm / (norm + 1e-6)).clamp(max=1.0)
            ext.multi_tensor_scale_run(cm[1], cm[2], coef)
            return sumsq
        return torch.nn.utils.clip_grad_norm_(model.parameters(), max_norm)
    torch.nn.utils.clip_grad_value_(model.parameters(), 5.0)
    return None


def build_optimizer(model: torch.nn.Module, family: str, params: Dict
                    ) -> Tuple[torch.optim.Optimizer, LrSchedule]:
    base_lr = params.get("lr", 1e-3)
    schedule = LrSchedule(
        family, base_lr,
        num_train_steps=params.get("num_train_steps", 10000),
        warmup_ratio=params.get("warmup_ratio", 0.1),
        step_per_epoch=params.get("step_per_epoch", 1000),
        decay_rate=params.get("lr_decay_rate", 0.95),
        d_model=params.get("transformer_params", {}).get("d_model", 160))
    if family == "bert":
        groups = build_param_groups(model, base_lr,
                                    params.get("weight_decay", 0.01),
                                    params.get("diff_lr_times"))
        opt = AdamWeightDecay(groups, lr=base_lr,
                              weight_decay=params.get("weight_decay", 0.01))
    else:
        opt = torch.optim.Adam(model.parameters(), lr=base_lr)
    return opt, schedule
