"""hipGraph-captured training step.

Captures one whole optimizer step — zero_grad, forward, backward, (DP
all-reduce waits), global-norm clip, fused AdamW — into a hipGraph
(torch.cuda.CUDAGraph == hipGraph on ROCm), then replays it per step:
the ~700 kernel-launch gaps per step (~1.4 ms on bert_bilstm_crf)
disappear. Requirements handled here:

* LR schedule stays live through the optimizer's device-side lr scalar
  (AdamWeightDecay.enable_graph_lr) written outside the graph;
* multi-tensor kernels use cached device meta blobs (optimizers.py) —
  no host→device copies inside the capture;
* dropout uses torch's graph-safe philox state (fresh masks per replay);
* inputs are copied into static buffers before each replay.

Falls back to eager transparently if capture fails (e.g. an RCCL build
without graph support) or when a batch's shapes differ from the
captured ones.
"""
from __future__ import annotations

import logging
import os
from typing import Callable, Dict, Optional

import torch

log = logging.getLogger("chinesener_amd")


class GraphedTrainStep:
    def __init__(self, model, optimizer, schedule, clip_fn: Callable,
                 dp=None, cast: Optional[Callable] = None):
        self.model = model
        self.opt = optimizer
        self.schedule = schedule
        self.clip_fn = clip_fn
        self.dp = dp
        self.cast = cast or (lambda b: b)
        self.graph = None
        self.static: Dict[str, torch.Tensor] = {}
        self.static_loss = None
        self.sig = None
        self.failed = False
        self._params = [p for p in model.parameters() if p.requires_grad]

    def _body(self):
        """One whole training step, capture-safe.

        Gradients come from ``torch.autograd.grad`` — NOT ``backward()``:
        a captured ``backward()`` goes through AccumulateGrad nodes and
        the resulting graph corrupts within a few replays once ANY
        foreign allocator/stream activity happens between replays (NaN
        loss or HSA aperture faults; reproduced and bisected on MI355X,
        scripts/debug/debug_capture_bisect.py: backward-capture arms corrupt,
        autograd.grad arms stay clean — the same reason upstream
        make_graphed_callables captures via autograd.grad). Grads are
        copied into persistent ``p.grad`` buffers so the cached
        clip/optimizer meta blobs keep stable pointers."""
        out = self.model(self.cast(self.static))
        grads = torch.autograd.grad(out.loss, self._params,
                                    allow_unused=True)
        with torch.no_grad():
            for p, g in zip(self._params, grads):
                if p.grad is None:
                    p.grad = torch.zeros_like(p)  # persistent buffer
                if g is None:
                    p.grad.zero_()
                else:
                    p.grad.copy_(g)
        if self.dp is not None:
            # autograd.grad fires no post-accumulate hooks: run the
            # bucketed all-reduce explicitly (recorded into the graph)
            self.dp.reduce_in_graph()
        self.clip_fn(self.model)
        self.opt.step()
        return out.loss

    def _signature(self, batch):
        return tuple(sorted((k, tuple(v.shape), str(v.dtype))
                            for k, v in batch.items()))

    def try_capture(self, batch: Dict[str, torch.Tensor],
                    step: int = 1) -> bool:
        if self.failed:
            return False
        try:
            if not hasattr(self.opt, "enable_graph_lr"):
                raise RuntimeError("optimizer has no device-lr support")
            device = next(self.model.parameters()).device
            self.opt.enable_graph_lr(device)
            self.schedule.apply(self.opt, max(1, step))
            self.static = {k: v.clone() for k, v in batch.items()}
            # The two warmup passes below perform REAL optimizer updates
            # from this one batch; snapshot params so the training
            # trajectory is unchanged by capture (the first replay then
            # applies the genuine step for this batch).
            do_restore = os.environ.get("CHINESENER_CAPTURE_NO_RESTORE") != "1"
            params = [p for p in self.model.parameters() if p.requires_grad]
            snap = [p.detach().clone() for p in params] if do_restore else []
            state_snap = {}
            if do_restore:
                for p in params:
                    st = self.opt.state.get(p)
                    if st:  # mid-training capture: preserve EMA state
                        state_snap[p] = {k: v.clone() for k, v in st.items()
                                         if isinstance(v, torch.Tensor)}
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(2):
                    self._body()
            torch.cuda.current_stream().wait_stream(side)
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self.static_loss = self._body()
            # restore params + reset m/v/master so warmup updates vanish
            # (state tensor ADDRESSES are baked into the graph's meta
            # blobs — reset in place, never reallocate)
            if do_restore:
                with torch.no_grad():
                    for p, s in zip(params, snap):
                        p.copy_(s)
                        st = self.opt.state.get(p)
                        if not st:
                            continue
                        prev = state_snap.get(p)
                        if prev:
                            for k, v in prev.items():
                                st[k].copy_(v)
                        else:   # state born during warmup: pristine init
                            st["m"].zero_()
                            st["v"].zero_()
                            if "master" in st:
                                st["master"].copy_(p.detach().float())
            self.sig = self._signature(batch)
            log.info("training step captured in hipGraph")
            return True
        except Exception as e:
            log.warning("hipGraph step capture failed (%s); eager", e)
            self.opt.lr_dev = None
            self.graph = None
            self.failed = True
            # A failed capture can leave the stream wedged in capture
            # state (any later CUDA op then fails). Probe and fail loudly
            # rather than letting every subsequent step error obscurely.
            try:
                if torch.cuda.is_available():
                    torch.cuda.synchronize()
            except Exception as e2:
                raise RuntimeError(
                    "stream left in invalid capture state after failed "
                    f"hipGraph capture: {e2}; set CHINESENER_NO_STEPGRAPH=1 "
                    "(or CHINESENER_STEPGRAPH=0 for Trainer)") from e
            return False

    def release(self):
        """Drop the captured graph and restore eager-optimizer state.

        Replays are only safe while the allocator sees no FOREIGN
        activity: differently-shaped allocations between replays (an
        eval pass, even plain torch.randn churn) corrupt subsequent
        replays on this stack within a few steps (NaN loss or
        HSA_STATUS_ERROR_MEMORY_APERTURE_VIOLATION — reproduced and
        bisected on MI355X, scripts/debug/debug_trainer_graph.py arms
        alloc/emptyalloc/rngonly). The Trainer therefore releases the
        graph at every eval/predict/checkpoint boundary and re-captures
        on the next repeated-shape step (~1 s per boundary)."""
        self.graph = None
        self.static = {}
        self.static_loss = None
        self.sig = None
        self.opt.lr_dev = None
        if hasattr(self.opt, "_meta_cache"):
            self.opt._meta_cache = {}
        if hasattr(self.model, "_clip_meta"):
            self.model._clip_meta = None

    def matches(self, batch) -> bool:
        return self.graph is not None and self.sig == self._signature(batch)

    def replay(self, batch: Dict[str, torch.Tensor], step: int) -> torch.Tensor:
        for k, v in batch.items():
            self.static[k].copy_(v, non_blocking=True)
        self.schedule.apply(self.opt, step)   # device lr write
        self.graph.replay()
        return self.static_loss
