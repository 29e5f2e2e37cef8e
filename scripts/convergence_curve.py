#!/usr/bin/env python3
"""Entity-F1 convergence curve for the flagship config: full BERT-base
(12L) bert_bilstm_crf on the learnable synthetic MSRA corpus (entity
chars drawn from per-type ranges), batch 64, L=128, pure bf16, graphed
step. Writes gpurun_out/convergence_curve.json."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from chinesener_amd.config import resolve_params
from chinesener_amd.data.loader import NerDataset
from chinesener_amd.eval import process_prediction
from chinesener_amd.eval.entity_eval import entity_report
from chinesener_amd.models import build_model, model_params
from chinesener_amd.train.trainer import Trainer


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(int(os.environ.get("CONV_SEED", "0")))
    name = "bert_bilstm_crf"
    out_dir = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "gpurun_out")
    os.makedirs(out_dir, exist_ok=True)
    pipe = NerDataset("/tmp/conv_data", "msra", 64, 1, name)
    params = resolve_params(model_params(name), pipe.params, {
        "model_name": name, "num_train_steps": 1600, "warmup_ratio": 0.1,
        "lr": 5e-5})
    model = build_model(name, params)
    trainer = Trainer(model, name, params, "/tmp/conv_ck")

    def batches():
        while True:
            yield from pipe.iter_batches("train")

    gen = batches()
    curve = []
    t0 = time.time()
    for step in range(1, int(os.environ.get("CONV_STEPS", "1200")) + 1):
        loss = trainer.train_step(next(gen))
        if step % int(os.environ.get("CONV_EVAL_EVERY", "200")) == 0:
            if os.environ.get("CONV_NO_EVAL") == "1":
                print(f"step {step} loss {loss:.2f}", flush=True)
                continue
            if os.environ.get("CONV_EVAL_ONE") == "1":
                with torch.no_grad():
                    trainer.model.eval()
                    b = next(pipe.iter_batches("valid", shuffle=False))
                    dev = trainer._cast({k: v.to(trainer.device)
                                         for k, v in b.items()})
                    out = trainer.model(dev, compute_pred=True)
                    trainer.model.train()
                print(f"step {step} loss {loss:.2f} evalloss "
                      f"{float(out.loss):.2f}", flush=True)
                continue
            rows = trainer.predict(pipe.iter_batches("valid", shuffle=False))
            idx2tag = pipe.params["idx2tag"]
            proc = [process_prediction(r, idx2tag) for r in rows]
            rep = entity_report([p["label_tags"] for p in proc],
                                [p["pred_tags"] for p in proc])
            f1 = rep["micro avg"]["f1"]
            curve.append({"step": step, "loss": round(loss, 2),
                          "entity_f1": round(f1, 4)})
            print(f"step {step} loss {loss:.2f} F1 {f1:.4f}", flush=True)
    result = {"config": "bert_bilstm_crf 12L, synthetic msra (learnable), "
                        "bs64 L128 bf16, lr 5e-5",
              "wall_s": round(time.time() - t0, 1), "curve": curve}
    with open(os.path.join(out_dir, "convergence_curve.json"), "w") as f:
        json.dump(result, f, indent=1)
    print(json.dumps(result))


if __name__ == "__main__":
    main()
