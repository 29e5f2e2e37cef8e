import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.cuda.tunable as tunable
from chinesener_amd.ops.tunable import load_tuned_gemm_table, _DEFAULT
print("csv exists:", os.path.exists(_DEFAULT))
ok = load_tuned_gemm_table()
print("load_tuned_gemm_table ->", ok, "enabled:", tunable.is_enabled(),
      "tuning:", tunable.tuning_is_enabled())
print("validators:", tunable.get_validators())
res = tunable.get_results()
print("n results:", len(res)); print(res[:3])
a = torch.randn(8192, 768, device="cuda", dtype=torch.bfloat16)
w = torch.randn(2304, 768, device="cuda", dtype=torch.bfloat16)
b = torch.randn(2304, device="cuda", dtype=torch.bfloat16)
import torch.nn.functional as F
for _ in range(10): y = F.linear(a, w, b)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(100): y = F.linear(a, w, b)
torch.cuda.synchronize()
print("qkv-shape GEMM: %.1f us" % ((time.perf_counter()-t0)/100*1e6))
tunable.enable(False)
for _ in range(10): y = F.linear(a, w, b)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(100): y = F.linear(a, w, b)
torch.cuda.synchronize()
print("untuned:        %.1f us" % ((time.perf_counter()-t0)/100*1e6))
