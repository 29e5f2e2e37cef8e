"""Training loop (the reference's Estimator train_and_evaluate analog,
main.py:47-49): step loop with bf16 autocast on GPU, gradient clipping,
LR schedule, periodic logging/checkpointing, early stopping on eval loss
(stop_if_no_decrease_hook, main.py:43-46), and predict-dump parity
(main.py:52-55)."""
from __future__ import annotations

import logging
import os
import pickle
import time
from typing import Dict, Iterable, Optional

import torch

from ..models import ModelOutput, NerModel, optimizer_family
from .checkpoints import CheckpointManager
from .metrics import TagMetrics
from .graph_step import GraphedTrainStep
from .optimizers import build_optimizer, clip_gradients
from .precision import convert_bf16_mixed, wants_pure_bf16
from ..utils.profiling import roctx_range

log = logging.getLogger("chinesener_amd")


class Trainer:
    def __init__(self, model: NerModel, model_name: str, params: Dict,
                 ckpt_dir: str, device: Optional[str] = None,
                 dp_engine=None, rank: int = 0):
        self.model = model
        self.model_name = model_name
        self.params = params
        self.rank = rank
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        if self.device.startswith("cuda"):
            from ..ops.tunable import load_tuned_gemm_table
            load_tuned_gemm_table()
        self.model.to(self.device)
        # pure-bf16 weights + fp32 masters in the fused optimizer (no
        # autocast weight-cast kernels; see train/precision.py)
        self.pure_bf16 = wants_pure_bf16(params, self.device)
        if self.pure_bf16:
            convert_bf16_mixed(self.model)
        self.family = optimizer_family(model_name)
        self.optimizer, self.schedule = build_optimizer(model, self.family, params)
        self.ckpt = CheckpointManager(ckpt_dir,
                                      params.get("keep_checkpoint_max", 3))
        self.dp = dp_engine
        self.use_bf16 = False  # autocast replaced by pure-bf16 weights
        self.step = self.ckpt.restore(model, self.optimizer,
                                      map_location=self.device)
        # hipGraph-captured step (default ON on GPU; CHINESENER_STEPGRAPH=0
        # disables). Replays are verified bit-exact vs eager steps. Capture
        # is DEFERRED until the diff-LR warm ramp finishes (schedule.warmup)
        # when x500-style multiplier groups exist: the fused kernel bakes
        # each group's full lr_scale_base into its cached meta blob, so a
        # pre-warmup capture would bypass the ramp that prevents all-O-basin
        # divergence (optimizers.py LrSchedule.apply).
        self._graph: GraphedTrainStep | None = None
        self._last_sig = None
        self._staging: Dict[str, torch.Tensor] = {}
        self._staging_sig = None
        self.use_step_graph = (self.device.startswith("cuda")
                               and os.environ.get("CHINESENER_STEPGRAPH")
                               != "0")
        self._has_diff_lr = any(
            g.get("lr_scale_base", 1.0) != 1.0
            for g in self.optimizer.param_groups)
        # layer activation summaries (reference add_layer_summary,
        # tools/utils.py:25-27): opt-in via params["verbose"] — JSONL in
        # the checkpoint dir instead of TensorBoard events
        self.summary = None
        if params.get("verbose") and rank == 0:
            from .summaries import SummaryLogger
            self.summary = SummaryLogger(
                model, ckpt_dir, every=params.get("summary_steps", 10))

    def _drop_graph(self):
        """Release the captured step graph before any foreign GPU work
        (eval, predict, checkpoint): allocator activity between replays
        corrupts them on this stack (see GraphedTrainStep.release).
        Re-capture happens automatically on the next repeated-shape
        train step."""
        if self._graph is not None:
            self._graph.release()
            self._graph = None

    # ------------------------------------------------------------- train
    def _cast(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        if not self.pure_bf16:
            return batch
        return {k: v.to(torch.bfloat16) if v.is_floating_point() else v
                for k, v in batch.items()}

    def _forward(self, batch: Dict[str, torch.Tensor]) -> ModelOutput:
        if self.use_bf16:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                return self.model(batch)
        return self.model(batch)

    def _stage(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        """Copy a batch into PERSISTENT device staging buffers (same
        shapes -> same storage every step). The replay loop must be
        allocation-free: fresh per-step `.to(device)` tensors are
        foreign allocator activity between graph replays, which this
        stack tolerates only statistically (soak runs crashed with HSA
        aperture faults after ~1-2k replays; allocation-free loops are
        stable — scripts/debug/debug_capture_bisect.py premade arms)."""
        sig = tuple(sorted((k, tuple(v.shape), str(v.dtype))
                           for k, v in batch.items()))
        if self._staging_sig != sig:
            self._staging = {k: torch.empty_like(v, device=self.device)
                             for k, v in batch.items()}
            self._staging_sig = sig
        for k, v in batch.items():
            self._staging[k].copy_(v, non_blocking=True)
        return self._staging

    def train_step(self, batch: Dict[str, torch.Tensor]) -> float:
        batch = self._stage(batch)
        graph_ready = (not self._has_diff_lr
                       or self.step >= self.schedule.warmup)
        if self.dp is not None:
            # capture is only safe over RCCL (gloo collectives wedge the
            # stream's capture state — see bench.py)
            import torch.distributed as _dist
            if _dist.is_initialized() and _dist.get_backend() != "nccl":
                graph_ready = False
        if self.use_step_graph and graph_ready:
            # ragged input streams (MRC padded batching) never settle on
            # one shape: every capture would be dropped a step later.
            # Give up on graphs as soon as shape diversity shows up.
            ssig = self._staging_sig
            seen = getattr(self, "_sigs_seen", None)
            if seen is None:
                seen = self._sigs_seen = set()
            seen.add(ssig)
            if len(seen) >= 4 and self._graph is None:
                log.info("ragged batch shapes (%d distinct); disabling "
                         "step graph", len(seen))
                self.use_step_graph = False
        if self.use_step_graph and graph_ready:
            if self._graph is not None and self._graph.matches(batch):
                self.step += 1
                loss = self._graph.replay(batch, self.step)
                return float(loss.detach())
            if self._graph is not None:
                # shape changed (e.g. MRC's ragged padded batches): the
                # eager step below is foreign allocator activity that
                # corrupts a LIVE graph's later replays — drop it first
                self._drop_graph()
                self._shape_misses = getattr(self, "_shape_misses", 0) + 1
                if self._shape_misses >= 3:
                    # shape-unstable input stream: graphs thrash; go eager
                    log.info("batch shapes unstable; disabling step graph")
                    self.use_step_graph = False
            sig = tuple(sorted((k, tuple(v.shape)) for k, v in batch.items()))
            if (self.use_step_graph and self._graph is None
                    and sig == self._last_sig):
                g = GraphedTrainStep(
                    self.model, self.optimizer, self.schedule,
                    lambda m: clip_gradients(m, self.family),
                    dp=self.dp, cast=self._cast)
                if g.try_capture(batch, step=self.step + 1):
                    self._graph = g
                    # a successful re-capture proves the stream is only
                    # periodically ragged (epoch-end partial batches):
                    # reset the instability counter so per-epoch drops
                    # never permanently disable graphs
                    self._shape_misses = 0
                    self.step += 1
                    return float(g.replay(batch, self.step).detach())
                self.use_step_graph = False
            self._last_sig = sig
        batch = self._cast(batch)
        if self.dp is not None:
            self.dp.zero_grad()
        else:
            self.optimizer.zero_grad(set_to_none=True)
        if self.summary is not None:
            self.summary.maybe_arm(self.step + 1)
        with roctx_range("forward"):
            out = self._forward(batch)
        if self.summary is not None:
            self.summary.flush(self.step + 1)
        with roctx_range("backward"):
            out.loss.backward()
        if self.dp is not None:
            with roctx_range("allreduce_wait"):
                self.dp.finalize_backward()  # wait bucketed all-reduces
        with roctx_range("clip+optimizer"):
            clip_gradients(self.model, self.family)
            self.step += 1
            self.schedule.apply(self.optimizer, self.step)
            self.optimizer.step()
        return float(out.loss.detach())

    def train(self, batches: Iterable[Dict[str, torch.Tensor]],
              eval_fn=None, log_steps: int = 100, save_steps: int = 500,
              early_stop_patience: Optional[int] = None,
              max_steps: Optional[int] = None) -> Dict:
        best_eval, since_best = float("inf"), 0
        t0, last_log_step = time.time(), self.step
        losses = []
        if (str(self.device).startswith("cuda")
                and os.environ.get("CHINESENER_NO_PREFETCH") != "1"):
            # one-batch-ahead pinned H2D staging on a side stream
            from ..data.loader import DevicePrefetcher
            batches = DevicePrefetcher(batches, self.device)
        for batch in batches:
            loss = self.train_step(batch)
            losses.append(loss)
            if self.step % log_steps == 0 and self.rank == 0:
                dt = time.time() - t0
                steps = self.step - last_log_step
                sps = steps * batch["token_ids"].shape[0] / max(dt, 1e-9)
                log.info("step %d loss %.4f lr %.2e %.1f samples/s",
                         self.step, sum(losses) / len(losses),
                         self.schedule.lr_at(self.step), sps)
                losses, t0, last_log_step = [], time.time(), self.step
            if save_steps and self.step % save_steps == 0:
                # ALL ranks drop together: re-capture's warmup passes
                # issue DP collectives, which must stay rank-aligned
                self._drop_graph()
            if save_steps and self.step % save_steps == 0 and self.rank == 0:
                self.ckpt.save(self.step, self.model, self.optimizer)
                if self.summary is not None:
                    self.summary.flush_artifacts(self.model, self.step)
                if eval_fn is not None:
                    ev = eval_fn()
                    if ev < best_eval:
                        best_eval, since_best = ev, 0
                    else:
                        since_best += 1
                        if (early_stop_patience is not None
                                and since_best >= early_stop_patience):
                            log.info("early stop at step %d", self.step)
                            break
            if max_steps is not None and self.step >= max_steps:
                break
        if self.rank == 0:
            self.ckpt.save(self.step, self.model, self.optimizer)
        return {"step": self.step, "best_eval": best_eval}

    # -------------------------------------------------------------- eval
    @torch.no_grad()
    def evaluate(self, batches: Iterable[Dict[str, torch.Tensor]],
                 idx2tag: Optional[Dict[int, str]] = None,
                 label_size: Optional[int] = None) -> Dict[str, float]:
        self._drop_graph()
        self.model.eval()
        metrics = TagMetrics(label_size or self.params.get("label_size", 10),
                             idx2tag or self.params.get("idx2tag"))
        total_loss, n = 0.0, 0
        for batch in batches:
            batch = self._cast({k: v.to(self.device) for k, v in batch.items()})
            if self.use_bf16:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    out = self.model(batch, compute_pred=True)
            else:
                out = self.model(batch, compute_pred=True)
            if out.loss is not None:
                total_loss += float(out.loss)
                n += 1
            metrics.update(out.pred_ids, batch["label_ids"], batch["mask"])
        self.model.train()
        result = metrics.compute()
        result["loss"] = total_loss / max(n, 1)
        return result

    # ----------------------------------------------------------- predict
    @torch.no_grad()
    def predict(self, batches: Iterable[Dict[str, torch.Tensor]]):
        """Returns list of per-sample dicts (pred_ids, label_ids, mask) —
        the reference pickles the Estimator.predict list (main.py:52-55)."""
        self._drop_graph()
        self.model.eval()
        out_rows = []
        for batch in batches:
            dev = self._cast({k: v.to(self.device) for k, v in batch.items()})
            if self.use_bf16:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    out = self.model(dev, compute_pred=True)
            else:
                out = self.model(dev, compute_pred=True)
            pred = out.pred_ids.cpu()
            for b in range(pred.shape[0]):
                out_rows.append({
                    "pred_ids": pred[b].numpy(),
                    "label_ids": batch["label_ids"][b].numpy(),
                    "mask": batch["mask"][b].numpy(),
                })
        self.model.train()
        return out_rows

    def dump_predictions(self, rows, data_dir: str, file_prefix: str) -> str:
        path = os.path.join(data_dir, f"{file_prefix}_predict.pkl")
        os.makedirs(data_dir, exist_ok=True)
        with open(path, "wb") as f:
            pickle.dump(rows, f)
        return path
