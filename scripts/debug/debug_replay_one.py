import os, sys, copy
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
                        {"model_name": name, "num_train_steps": 10**9, "lr": 1e-6, "warmup_ratio": 1.0,
                         "bert_config": cfg, "dropout_rate": 0.0,
                         "embedding_dropout": 0.0})
params["rnn_params"] = dict(params["rnn_params"], keep_prob_list=[1.0])
model = build_model(name, params)
tr = Trainer(model, name, params, "/tmp/ck1")
batches = list(pipe.iter_batches("train", shuffle=False))[:3]
# trigger capture (step1 eager, step2 capture+replay)
tr.train_step(batches[0]); tr.train_step(batches[0])
assert tr._graph is not None
g = tr._graph

def snapshot():
    snap = {"p": {n: p.detach().clone() for n, p in model.named_parameters()},
            "s": {}}
    for p, st in tr.optimizer.state.items():
        snap["s"][id(p)] = {k: v.clone() for k, v in st.items()
                            if torch.is_tensor(v)}
    return snap

def restore(snap):
    with torch.no_grad():
        for n, p in model.named_parameters():
            p.copy_(snap["p"][n])
        for p, st in tr.optimizer.state.items():
            for k, v in st.items():
                if torch.is_tensor(v):
                    v.copy_(snap["s"][id(p)][k])

X = {k: v.to(tr.device) for k, v in batches[1].items()}
S0 = snapshot()
N = int(os.environ.get("N", "30"))

gl = []
for i in range(N):
    loss_g = g.replay(X, tr.step + 1 + i)
    gl.append(float(loss_g))
torch.cuda.synchronize()
Wg = {n: p.detach().float().clone() for n, p in model.named_parameters()}

restore(S0)
from chinesener_amd.train.optimizers import clip_gradients
opt = tr.optimizer
el = []
for i in range(N):
    tr.schedule.apply(opt, tr.step + 1 + i)
    opt.zero_grad(set_to_none=False)
    out = model(tr._cast(X))
    out.loss.backward()
    clip_gradients(model, tr.family)
    opt.step()
    el.append(float(out.loss))
torch.cuda.synchronize()
We = {n: p.detach().float().clone() for n, p in model.named_parameters()}

print("replay losses:", [round(x, 3) for x in gl[::max(1, N // 10)]])
print("eager  losses:", [round(x, 3) for x in el[::max(1, N // 10)]])
diffs = sorted(((float((Wg[n] - We[n]).abs().max()), n) for n in Wg),
               reverse=True)
print("top post-chain param maxdiffs:")
for d, n in diffs[:6]:
    print(f"  {d:.3e}  {n}")
