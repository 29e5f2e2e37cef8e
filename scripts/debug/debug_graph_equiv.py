import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from chinesener_amd.config import resolve_params
from chinesener_amd.data.loader import NerDataset
from chinesener_amd.models import build_model, model_params
from chinesener_amd.models.bert import BertConfig
from chinesener_amd.train.trainer import Trainer

def run(graphed, steps=12):
    torch.manual_seed(0)
    name = "bert_bilstm_crf"
    os.environ["CHINESENER_NO_STEPGRAPH"] = "0" if graphed else "1"
    pipe = NerDataset("/tmp/d", "msra", 64, 1, name)
    cfg = BertConfig(hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0)
    params = resolve_params(model_params(name), pipe.params,
                            {"model_name": name, "num_train_steps": 1600,
                             "bert_config": cfg, "dropout_rate": 0.0})
    params["rnn_params"] = dict(params["rnn_params"], keep_prob_list=[1.0])
    params["embedding_dropout"] = 0.0
    model = build_model(name, params)
    trainer = Trainer(model, name, params, f"/tmp/ck_{graphed}")
    batches = list(pipe.iter_batches("train", shuffle=False))[:6]
    losses = []
    for step in range(steps):
        losses.append(round(trainer.train_step(batches[step % 6]), 3))
    return losses

e = run(False)
g = run(True)
print("eager  :", e)
print("graphed:", g)
import math
print("maxdiff:", max(abs(a-b) for a,b in zip(e,g)))
