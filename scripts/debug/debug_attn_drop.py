import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from chinesener_amd.ops import functional as fn
from chinesener_amd.ops.functional import _rng_counter_for

def splitmix(ctr, idx):
    z = (ctr * 0x9E3779B97F4A7C15 ^ idx) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    z = z ^ (z >> 31)
    return (z >> 40) / 16777216.0

torch.manual_seed(7)
B, L, H, D = 2, 32, 2, 64
keep = 0.7
qkv = torch.randn(B, L, 3, H, D, device="cuda", dtype=torch.bfloat16,
                  requires_grad=True)
mask = torch.ones(B, L, dtype=torch.long, device="cuda")
ctr = _rng_counter_for(qkv.device)
seed_val = int(ctr.item())   # the value fwd will snapshot
out = fn.attention_qkv(qkv, mask=mask, p_drop=1 - keep, training=True)
g = torch.randn_like(out)
out.backward(g)
dqkv_gpu = qkv.grad.float().clone()

# torch replay with the SAME mask
D_mask = torch.zeros(B, H, L, L)
for b in range(B):
    for h in range(H):
        bh = b * H + h
        for i in range(L):
            for j in range(L):
                idx = (bh * L + i) * L + j
                D_mask[b, h, i, j] = 1.0 if splitmix(seed_val, idx) < keep else 0.0
D_mask = D_mask.cuda()

q, k, v = (qkv.detach().float()[:, :, i].transpose(1, 2).requires_grad_(True)
           for i in range(3))
scale = 1.0 / (D ** 0.5)
scores = torch.matmul(q, k.transpose(-1, -2)) * scale
probs = torch.softmax(scores, dim=-1)
probs_d = probs * D_mask / keep
o_ref = torch.matmul(probs_d, v).transpose(1, 2)   # [B,L,H,D]
print("fwd maxdiff:", (out.float() - o_ref).abs().max().item())
o_ref.backward(g.float())
dq_ref = torch.stack([q.grad, k.grad, v.grad], dim=0)
dq_gpu = torch.stack([dqkv_gpu[:, :, i].transpose(1, 2) for i in range(3)])
for name, i in [("dq", 0), ("dk", 1), ("dv", 2)]:
    d = (dq_gpu[i] - dq_ref[i]).abs().max().item()
    m = dq_ref[i].abs().max().item()
    print(f"{name} maxdiff {d:.4f} (ref absmax {m:.2f})")
