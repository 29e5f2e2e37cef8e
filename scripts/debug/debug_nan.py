"""Trace the first NaN-producing op in the bert_bilstm_crf forward by
wrapping every custom op + nn.functional entry with a NaN check."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import chinesener_amd.ops as ops
import chinesener_amd.ops.functional as fn


def wrap(name, f):
    def g(*args, **kw):
        out = f(*args, **kw)
        t = out if torch.is_tensor(out) else out[0]
        n = int(t.float().isnan().sum())
        if n:
            print(f"FIRST NAN in op {name}: {n} nans, out shape {tuple(t.shape)}")
            for i, a in enumerate(args):
                if torch.is_tensor(a):
                    print(f"  arg{i}: shape={tuple(a.shape)} dtype={a.dtype} "
                          f"nan={int(a.float().isnan().sum())} "
                          f"absmax={float(a.float().abs().max()):.4f}")
            raise SystemExit(1)
        return out
    return g


for opname in ["attention", "add_layernorm", "layernorm", "bias_gelu",
               "bilstm", "crf_nll", "softlexicon_fuse", "masked_cross_entropy"]:
    setattr(ops, opname, wrap(opname, getattr(fn, opname)))

# also patch the modules that imported ops by name
import chinesener_amd.models.bert as bert_mod
import chinesener_amd.models.layers as layers_mod
bert_mod.ops = ops
layers_mod.ops = ops

from chinesener_amd.models import build_model
from chinesener_amd.models.bert import BertConfig
from chinesener_amd.data.loader import make_synthetic_batch

torch.manual_seed(0)
cfg = BertConfig(vocab_size=2000, hidden_size=768, num_hidden_layers=2,
                 num_attention_heads=12, intermediate_size=3072)
params = {"vocab_size": 2000, "label_size": 10, "bert_config": cfg,
          "rnn_params": {"hidden_units_list": [128],
                         "cell_activation": "relu", "keep_prob_list": [0.8]},
          "tag2idx": {}, "dropout_rate": 0.1}
model = build_model("bert_bilstm_crf", params).to("cuda")
batch = make_synthetic_batch(4, 64, 10, vocab_size=2000, device="cuda")
with torch.autocast("cuda", dtype=torch.bfloat16):
    out = model(batch)
print("forward loss:", float(out.loss))
out.loss.backward()
bad = [n for n, p in model.named_parameters()
       if p.grad is not None and p.grad.float().isnan().any()]
print("nan-grad params:", bad[:10])
print("done, no forward NaN" if torch.isfinite(out.loss) else "loss NaN")
