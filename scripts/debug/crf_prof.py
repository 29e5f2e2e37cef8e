import sys, time, torch
sys.path.insert(0, "/root/repo")
from chinesener_amd import ops
ext = ops.get_ext()
B, L, T = 64, 128, 10
em = torch.randn(B, L, T, device="cuda")
tags = torch.randint(0, T, (B, L), device="cuda", dtype=torch.int32)
lens = torch.full((B,), L, device="cuda", dtype=torch.int32)
tr = torch.randn(T, T, device="cuda")
for _ in range(5):
    ext.crf_fwd(em, tags, lens, tr); ext.crf_viterbi(em, lens, tr)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    ext.crf_fwd(em, tags, lens, tr)
torch.cuda.synchronize()
print(f"crf_fwd e2e {1e6*(time.perf_counter()-t0)/50:.1f} us/call")
t0 = time.perf_counter()
for _ in range(50):
    ext.crf_viterbi(em, lens, tr)
torch.cuda.synchronize()
print(f"viterbi e2e {1e6*(time.perf_counter()-t0)/50:.1f} us/call")
