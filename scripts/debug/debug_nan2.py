import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from chinesener_amd import ops
from chinesener_amd.ops import functional as fn
from chinesener_amd.config import resolve_params
from chinesener_amd.data.loader import NerDataset
from chinesener_amd.models import build_model, model_params
from chinesener_amd.train.precision import convert_bf16_mixed

torch.manual_seed(0)
name = "bert_bilstm_crf"
pipe = NerDataset("/tmp/d", "msra", 64, 1, name)
params = resolve_params(model_params(name), pipe.params, {"model_name": name})
model = build_model(name, params).to("cuda")
convert_bf16_mixed(model)
batch = next(pipe.iter_batches("train", shuffle=False))
dev = {k: (v.to("cuda").to(torch.bfloat16) if v.is_floating_point()
           else v.to("cuda")) for k, v in batch.items()}

def one_step(tag):
    for p in model.parameters(): p.grad = None
    out = model(dev)
    out.loss.backward()
    gmax = max(p.grad.float().abs().max().item()
               for p in model.parameters() if p.grad is not None)
    print(tag, "loss", float(out.loss), "gradmax", gmax)

one_step("full(fused dropout)")

# bypass fused dropout
import torch.nn.functional as F
orig = fn.dropout_add_layernorm
def plain(x, res, w, b, eps=1e-12, p=0.1, training=True):
    if training and p > 0:
        x = F.dropout(x, p, training)
    return fn.add_layernorm(x, res, w, b, eps)
ops.dropout_add_layernorm = plain
import chinesener_amd.models.bert as BB
BB.ops.dropout_add_layernorm = plain
one_step("torch-dropout")

model.eval()
with torch.no_grad():
    out = model(dev, compute_pred=True)
    print("eval loss", float(out.loss))
model.train()

# attention isolated at L=150
torch.manual_seed(1)
B, H, L, D = 64, 12, 150, 64
q = torch.randn(B, H, L, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
lens = torch.randint(5, L + 1, (B,), device="cuda")
mask = (torch.arange(L, device="cuda")[None, :] < lens[:, None]).long()
q.requires_grad_(); k.requires_grad_(); v.requires_grad_()
o = fn.attention(q, k, v, mask=mask)
o.sum().backward()
from chinesener_amd.ops import reference as ref
o_ref = ref.attention(q.detach().float(), k.detach().float(), v.detach().float(), mask)
print("attn L150 fwd maxdiff", (o.float() - o_ref).abs().max().item(),
      "dq finite", torch.isfinite(q.grad.float()).all().item(),
      "dq absmax", q.grad.float().abs().max().item())

# dropout_add_ln isolated at H=768
x = torch.randn(9600, 768, device="cuda", dtype=torch.bfloat16, requires_grad=True)
r = torch.randn(9600, 768, device="cuda", dtype=torch.bfloat16)
w = torch.ones(768, device="cuda"); b = torch.zeros(768, device="cuda")
y = orig(x, r, w, b, 1e-12, 0.1, True)
y.float().sum().backward()
print("fused dln finite:", torch.isfinite(y.float()).all().item(),
      torch.isfinite(x.grad.float()).all().item(),
      "y absmax", y.float().abs().max().item())
