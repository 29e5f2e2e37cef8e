#!/usr/bin/env python3
"""MRC-NER driver — reference mrc/main.py parity:

  python mrc_main.py --data msra --do_train --do_eval [--do_export]

Trains BERT→dense(3) BIO tagging over [CLS]query[SEP]text samples
(×3 tag queries per sentence), evaluates with a span-level strict
report written to train.log, dumps extracted entities, and exports the
serving model (reference mrc/main.py:37-122)."""
from __future__ import annotations

import argparse
import logging
import os
import pickle
import sys

# GEMM autotuning must be configured before torch init (see bench.py)
if os.environ.get("CHINESENER_NO_TUNABLE") != "1":
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "100")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                          "/tmp/chinesener_tunableop_.csv")

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from chinesener_amd.config import CHECKPOINT_DIR, DATA_DIR, resolve_params
from chinesener_amd.eval.entity_eval import entity_report, report_to_text
from chinesener_amd.models import build_model, model_params
from chinesener_amd.models.mrc import MRC_LABELS
from chinesener_amd.mrc.dataset import MrcDataset
from chinesener_amd.train.checkpoints import ckpt_dir, clear_model
from chinesener_amd.train.trainer import Trainer

log = logging.getLogger("chinesener_amd")
IDX2MRC = {v: k for k, v in MRC_LABELS.items()}


def mrc_rows_to_tags(rows, tag_types):
    """Per-sample BIO-over-text-region tag sequences, typed by the sample's
    query tag ('B' -> 'B-PER' etc.) so entity_report scores spans."""
    y_true, y_pred = [], []
    for row, tag_type in zip(rows, tag_types):
        mask = row["text_mask"].astype(bool)
        to_tags = lambda ids: [  # noqa: E731
            "O" if IDX2MRC[int(i)] == "O" else f"{IDX2MRC[int(i)]}-{tag_type}"
            for i in ids[mask]]
        y_pred.append(to_tags(row["pred_ids"]))
        y_true.append(to_tags(row["label_ids"]))
    return y_true, y_pred


@torch.no_grad()
def mrc_predict(trainer, batches):
    trainer.model.eval()
    rows, tag_types = [], []
    for batch in batches:
        dev = {k: v.to(trainer.device) for k, v in batch.items()}
        if trainer.use_bf16:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                out = trainer.model(dev, compute_pred=True)
        else:
            out = trainer.model(dev, compute_pred=True)
        pred = out.pred_ids.cpu().numpy()
        for b in range(pred.shape[0]):
            rows.append({"pred_ids": pred[b],
                         "label_ids": batch["label_ids"][b].numpy(),
                         "text_mask": batch["text_mask"][b].numpy()})
            tag_types.append("")
    trainer.model.train()
    return rows


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--data", default="msra")
    ap.add_argument("--clear_model", action="store_true")
    ap.add_argument("--use_gpu", action="store_true",
                    help="reference-CLI parity; GPU is auto-detected")
    ap.add_argument("--do_train", action="store_true")
    ap.add_argument("--do_eval", action="store_true")
    ap.add_argument("--do_export", action="store_true")
    ap.add_argument("--epochs", type=int, default=1)
    ap.add_argument("--max_steps", type=int, default=None)
    ap.add_argument("--batch_size", type=int, default=None)
    ap.add_argument("--data_dir", default=None)
    ap.add_argument("--ckpt_root", default=CHECKPOINT_DIR)
    args = ap.parse_args(argv)

    data_dir = args.data_dir or os.path.join(DATA_DIR, args.data)
    mparams = model_params("mrc_bio")
    batch_size = args.batch_size or mparams.get("batch_size", 32)
    pipe = MrcDataset(data_dir, args.data, batch_size,
                      mparams.get("max_seq_len", 170))
    params = resolve_params(mparams, pipe.params,
                            {"model_name": "mrc_bio",
                             "epoch_size": args.epochs})
    params["num_train_steps"] = params["step_per_epoch"] * args.epochs
    model_dir = ckpt_dir(args.data, "MRC", args.ckpt_root)
    if args.clear_model:
        clear_model(model_dir)
    os.makedirs(model_dir, exist_ok=True)
    logging.basicConfig(
        level=logging.INFO, force=True,
        handlers=[logging.StreamHandler(),
                  logging.FileHandler(os.path.join(model_dir, "train.log"))],
        format="%(asctime)s %(message)s")

    model = build_model("mrc_bio", params)
    trainer = Trainer(model, "mrc_bio", params, model_dir)

    if args.do_train:
        log.info("MRC train: data=%s n_sample=%d bs=%d L=%d", args.data,
                 params["n_sample"], batch_size, params["max_seq_len"])
        trainer.train(pipe.iter_batches("train"),
                      log_steps=params.get("log_steps", 100),
                      save_steps=params.get("save_steps", 500),
                      max_steps=args.max_steps)

    if args.do_eval:
        rows = mrc_predict(trainer, pipe.iter_batches("test", shuffle=False))
        # token-level ROC/PR AUC over the B/I classes (reference
        # mrc/model.py:57-83 adds ROC-AUC and PR-AUC eval metrics)
        try:
            from sklearn.metrics import average_precision_score, roc_auc_score
            import numpy as np
            y, s = [], []
            for row in rows:
                m = row["text_mask"].astype(bool)
                y.extend((row["label_ids"][m] > 0).astype(int).tolist())
                s.extend((row["pred_ids"][m] > 0).astype(int).tolist())
            if len(set(y)) == 2:
                log.info("MRC token ROC-AUC %.4f PR-AUC %.4f",
                         roc_auc_score(y, s), average_precision_score(y, s))
        except Exception as e:  # pragma: no cover
            log.warning("AUC metrics skipped: %s", e)
        # tag types cycle per tag query in build order
        tag_types = []
        for _ in range(len(rows) // max(1, len(pipe.tag2query)) + 1):
            tag_types.extend(pipe.tag2query)
        y_true, y_pred = mrc_rows_to_tags(rows, tag_types[:len(rows)])
        rep = entity_report(y_true, y_pred)
        text = report_to_text(rep, f"=== MRC span-level report @ {args.data} ===")
        log.info("\n%s", text)
        os.makedirs(data_dir, exist_ok=True)
        with open(os.path.join(data_dir, "mrc_bio_predict.pkl"), "wb") as f:
            pickle.dump(rows, f)

    if args.do_export:
        from chinesener_amd.serve.export import export_model
        export_model(trainer.model, "mrc_bio", params)
    return 0


if __name__ == "__main__":
    sys.exit(main())
