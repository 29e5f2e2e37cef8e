import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from chinesener_amd.config import resolve_params
from chinesener_amd.data.loader import NerDataset
from chinesener_amd.models import build_model, model_params
from chinesener_amd.models.bert import BertConfig
from chinesener_amd.train.trainer import Trainer

K = int(os.environ.get("K", "30"))

def build():
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
    tr = Trainer(model, name, params, f"/tmp/ck_{os.environ.get('G','x')}")
    batch = next(pipe.iter_batches("train", shuffle=False))
    return tr, batch

os.environ["CHINESENER_NO_STEPGRAPH"] = "1"
os.environ["G"] = "e"
tr_e, batch = build()
el = []
for i in range(3 + K):
    el.append(tr_e.train_step(batch))
en = {n: p.detach().float().norm().item() for n, p in tr_e.model.named_parameters()}

os.environ["CHINESENER_NO_STEPGRAPH"] = "0"
os.environ["G"] = "g"
tr_g, batch2 = build()
gl = []
for i in range(K):   # capture does 3 extra updates internally at step 2... 
    gl.append(tr_g.train_step(batch2))
gn = {n: p.detach().float().norm().item() for n, p in tr_g.model.named_parameters()}

print("eager last5 :", [round(x,3) for x in el[-5:]])
print("graph last5 :", [round(x,3) for x in gl[-5:]])
diffs = sorted(((abs(en[n]-gn[n])/(en[n]+1e-9), n) for n in en), reverse=True)
print("top param norm rel-diffs:")
for d, n in diffs[:8]:
    print(f"  {d:.4f} {n} eager={en[n]:.4f} graph={gn[n]:.4f}")
