#!/usr/bin/env python3
"""Train driver — CLI parity with the reference's main.py:120-145:

  python main.py --model_name bert_bilstm_crf --data msra
  python main.py --model_name bert_bilstm_crf_mtl --data msra,people_daily

Comma in --data selects multi-task training (reference main.py:142-143);
--clear_model wipes the checkpoint dir; --device selects the GPU;
--rename renames the checkpoint/serving dirs; --export_only skips
training and exports the serving model. After training it dumps test
predictions to ./data/{data}/{model}_predict.pkl (main.py:52-55) and
exports ./serving_model/{model}/{version}/ (main.py:57-60).

Multi-GPU data parallel (new vs the reference): launch
with  torchrun --nproc-per-node N main.py ... ; ranks are wired over
RCCL/xGMI through chinesener_amd.dist.
"""
from __future__ import annotations

import argparse
import logging
import os
import sys

# GEMM autotuning must be configured before torch init (see bench.py);
# the first ~1 min of training doubles as the tuning warmup.
if os.environ.get("CHINESENER_NO_TUNABLE") != "1":
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "100")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                          "/tmp/chinesener_tunableop_.csv")

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from chinesener_amd.config import CHECKPOINT_DIR, DATA_DIR, RUN_CONFIG, resolve_params
from chinesener_amd.data.loader import MultiDataset, NerDataset
from chinesener_amd.models import build_model, model_params
from chinesener_amd.train.checkpoints import ckpt_dir, clear_model
from chinesener_amd.train.trainer import Trainer


def setup_logging(model_dir: str | None = None):
    handlers = [logging.StreamHandler()]
    if model_dir:
        os.makedirs(model_dir, exist_ok=True)
        handlers.append(logging.FileHandler(os.path.join(model_dir, "train.log")))
    logging.basicConfig(level=logging.INFO, handlers=handlers,
                        format="%(asctime)s %(levelname)s %(message)s", force=True)


def dist_init(args):
    """One process per GPU over RCCL when launched by torchrun."""
    if "RANK" in os.environ and int(os.environ.get("WORLD_SIZE", "1")) > 1:
        from chinesener_amd.dist import init_process_group
        return init_process_group()
    return 0, 1


def build_everything(args, rank: int, world_size: int):
    datas = args.data.split(",")
    name = args.rename or args.model_name
    mparams = model_params(args.model_name)
    if len(datas) > 1 and not ("_mtl" in args.model_name
                               or "_adv" in args.model_name):
        raise SystemExit(
            f"--data {args.data}: comma-separated datasets select "
            f"multi-task training, which needs a shared-BERT mtl/adv "
            f"model (bert_bilstm_crf_mtl or bert_bilstm_crf_adv) — "
            f"'{args.model_name}' is single-task (reference main.py "
            f"multitask_train has the same contract)")
    if len(datas) == 1:
        pipe = NerDataset(args.data_dir, datas[0], args.batch_size or
                          mparams.get("batch_size", 32), args.epochs,
                          args.model_name, rank, world_size)
    else:
        pipe = MultiDataset(args.data_dir, datas, args.batch_size or
                            mparams.get("batch_size", 32), args.epochs,
                            args.model_name, rank, world_size)
    params = resolve_params(mparams, pipe.params, {
        "epoch_size": args.epochs, "model_name": args.model_name, **RUN_CONFIG})
    if args.batch_size:
        params["batch_size"] = args.batch_size
    if args.max_steps:
        # a user-requested step budget extends the LR schedule: otherwise
        # the poly decay reaches 0 at step_per_epoch*epochs and any steps
        # past that train at lr=0
        params["num_train_steps"] = max(
            params.get("num_train_steps", 0), args.max_steps)
    from chinesener_amd.models import apply_addon_values
    apply_addon_values(args.model_name, params,
                       getattr(args, "_addon_values", {}))
    model = build_model(args.model_name, params)
    model_dir = ckpt_dir(args.data.replace(",", "_"), name, args.ckpt_root)
    return pipe, params, model, model_dir, name


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model_name", required=True)
    ap.add_argument("--data", required=True,
                    help="dataset name; comma-separated pair => multi-task")
    ap.add_argument("--clear_model", action="store_true")
    ap.add_argument("--gpu", action="store_true",
                    help="kept for reference-CLI parity; GPU is auto-detected")
    ap.add_argument("--device", type=int, default=None,
                    help="GPU index (reference --device, main.py:128-131)")
    ap.add_argument("--rename", default=None)
    ap.add_argument("--export_only", action="store_true")
    ap.add_argument("--epochs", type=int, default=10)
    ap.add_argument("--batch_size", type=int, default=None)
    ap.add_argument("--max_steps", type=int, default=None)
    ap.add_argument("--data_dir", default=DATA_DIR)
    ap.add_argument("--ckpt_root", default=CHECKPOINT_DIR)
    # per-model extra flags (reference AddonParser: each model module
    # declares additional hyperparameters merged into the CLI)
    from chinesener_amd.models import model_addons
    from chinesener_amd.train.addon_parser import AddonParser
    pre = argparse.ArgumentParser(add_help=False)
    pre.add_argument("--model_name", default=None)
    known, _ = pre.parse_known_args(argv)
    addon_parser = AddonParser(model_addons(known.model_name)
                               if known.model_name else [])
    addon_parser.append(ap)
    args = ap.parse_args(argv)
    args._addon_values = addon_parser.extract(args)

    if args.device is not None:
        os.environ.setdefault("HIP_VISIBLE_DEVICES", str(args.device))

    rank, world_size = dist_init(args)
    torch.manual_seed(int(RUN_CONFIG.get("seed", 1234)) + rank)
    pipe, params, model, model_dir, name = build_everything(args, rank, world_size)
    if args.clear_model and rank == 0:
        clear_model(model_dir)
    setup_logging(model_dir if rank == 0 else None)
    log = logging.getLogger("chinesener_amd")
    log.info("model=%s data=%s params(batch=%s steps/epoch=%s) device=%s",
             args.model_name, args.data, params.get("batch_size"),
             params.get("step_per_epoch"),
             "cuda" if torch.cuda.is_available() else "cpu")

    dp_engine = None
    if world_size > 1:
        from chinesener_amd.dist import BucketedDataParallel
        model = model.to("cuda" if torch.cuda.is_available() else "cpu")
        dp_engine = BucketedDataParallel(model)

    trainer = Trainer(model, args.model_name, params, model_dir,
                      dp_engine=dp_engine, rank=rank)

    if args.export_only:
        from chinesener_amd.serve.export import export_model
        export_model(model, name, params)
        return 0

    datas = args.data.split(",")
    eval_pipe = pipe if len(datas) == 1 else pipe.pipes[0]

    def eval_fn():
        m = trainer.evaluate(eval_pipe.iter_batches("valid", shuffle=False),
                             params.get("idx2tag"), params.get("label_size")
                             or eval_pipe.params["label_size"])
        log.info("eval: loss %.4f acc %.4f micro_f1 %.4f", m["loss"],
                 m["accuracy"], m.get("micro_f1", float("nan")))
        return m["loss"]

    patience = max(1, int(params.get("early_stop_ratio", 1.0)))
    trainer.train(pipe.iter_batches("train"), eval_fn=eval_fn,
                  log_steps=params.get("log_steps", 100),
                  save_steps=params.get("save_steps", 500),
                  early_stop_patience=patience, max_steps=args.max_steps)

    if rank == 0:
        # dump test predictions per task (reference main.py:52-55, :105-112)
        if len(datas) == 1:
            rows = trainer.predict(pipe.iter_batches("test", shuffle=False))
            # attach raw sentences so evaluation.py --topn can show text
            # (the reference pkl carries tokens, main.py:52-55)
            raws = pipe.raw_sentences("test")
            for row, raw in zip(rows, raws):
                row["raw"] = raw
            trainer.dump_predictions(rows, os.path.join(args.data_dir, datas[0]),
                                     name)
        else:
            for task in datas:
                tp = pipe.build_predict_pipe(task)
                rows = trainer.predict(tp.iter_batches("test", shuffle=False))
                trainer.dump_predictions(
                    rows, os.path.join(args.data_dir, task),
                    f"{name}_{'_'.join(datas)}")
        from chinesener_amd.serve.export import export_model
        export_model(model, name, params)
    return 0


if __name__ == "__main__":
    sys.exit(main())
