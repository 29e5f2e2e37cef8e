import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from chinesener_amd.config import resolve_params
from chinesener_amd.data.loader import NerDataset
from chinesener_amd.models import build_model, model_params
from chinesener_amd.models.bert import BertConfig
from chinesener_amd.train.trainer import Trainer

torch.manual_seed(0)
name = "bert_bilstm_crf"
pipe = NerDataset("/tmp/d", "msra", 64, 1, name)
cfg = BertConfig(hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0)
params = resolve_params(model_params(name), pipe.params,
                        {"model_name": name, "num_train_steps": 1600,
                         "bert_config": cfg, "dropout_rate": 0.0,
                         "embedding_dropout": 0.0})
params["rnn_params"] = dict(params["rnn_params"], keep_prob_list=[1.0])
model = build_model(name, params)
tr = Trainer(model, name, params, "/tmp/ckno")
batch = next(pipe.iter_batches("train", shuffle=False))
prev_nonfinite = set()
for step in range(1, 301):
    loss = tr.train_step(batch)
    if step % 10 == 0 or loss != loss:
        wn = sum(float(p.detach().float().pow(2).sum())
                 for p in model.parameters()) ** 0.5
        mx = max(float(p.detach().float().abs().max())
                 for p in model.parameters())
        lrv = float(tr.optimizer.lr_dev) if tr.optimizer.lr_dev is not None else -1
        print(f"step {step} loss {loss:.3f} wnorm {wn:.2f} wmax {mx:.2f} "
              f"lr_dev {lrv:.2e} graphed {tr._graph is not None}", flush=True)
    if loss != loss:
        bad = [n for n, p in model.named_parameters()
               if not torch.isfinite(p.detach().float()).all()]
        print("non-finite params:", bad[:6])
        # check optimizer masters
        badm = []
        for n, p in model.named_parameters():
            s = tr.optimizer.state.get(p, {})
            for k in ("m", "v", "master"):
                if k in s and not torch.isfinite(s[k]).all():
                    badm.append(f"{n}.{k}")
        print("non-finite opt state:", badm[:6])
        break
