import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from chinesener_amd import ops
ext = ops.get_ext()
torch.manual_seed(0)

def check(T, N, Kin):
    dy = torch.randn(T, N, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(T, Kin, device="cuda", dtype=torch.bfloat16)
    ref = (dy.float().T @ x.float())
    got = ext.wgrad(dy, x, 0)
    d = (got - ref).abs().max().item()
    rel = d / ref.abs().max().item()
    def bench(fn, n=30):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(n): fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1e6
    t_us = bench(lambda: ext.wgrad(dy, x, 0))
    t_lib = bench(lambda: dy.T @ x)
    fl = 2.0 * T * N * Kin
    print(f"T={T} N={N} K={Kin}: maxrel {rel:.4f}  ours {t_us:.1f}us "
          f"({fl/t_us/1e6:.0f} TF/s)  hipblaslt {t_lib:.1f}us "
          f"({fl/t_lib/1e6:.0f} TF/s)")

check(8192, 2304, 768)
check(8192, 768, 2304)
check(8192, 3072, 768)
check(8192, 768, 3072)
check(8192, 768, 768)
check(9600, 2304, 768)   # bs64 x L150
check(1000, 768, 768)    # ragged T
