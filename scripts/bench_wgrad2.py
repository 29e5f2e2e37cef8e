"""Microbenchmark wgrad v2 vs v3 (wgrad2) vs hipBLASLt on the BERT
wgrad shapes (docs/ROADMAP.md item 1). Writes a summary to
gpurun_out/wgrad2_bench.txt when run under gpurun."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from chinesener_amd import ops  # noqa: E402

ext = ops.get_ext()
torch.manual_seed(0)
lines = []


def bench(fn, n=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


def check(T, N, Kin):
    dy = torch.randn(T, N, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(T, Kin, device="cuda", dtype=torch.bfloat16)
    ref = dy.float().T @ x.float()
    g2 = ext.wgrad(dy, x, 0)
    g3 = ext.wgrad2(dy, x, 0)
    r2 = ((g2 - ref).abs().max() / ref.abs().max()).item()
    r3 = ((g3 - ref).abs().max() / ref.abs().max()).item()
    t2 = bench(lambda: ext.wgrad(dy, x, 0))
    t3 = bench(lambda: ext.wgrad2(dy, x, 0))
    tl = bench(lambda: dy.T @ x)
    fl = 2.0 * T * N * Kin
    line = (f"T={T:5d} N={N:4d} K={Kin:4d}: rel v2={r2:.4f} v3={r3:.4f} | "
            f"v2 {t2:7.1f}us ({fl/t2/1e6:4.0f} TF/s)  "
            f"v3 {t3:7.1f}us ({fl/t3/1e6:4.0f} TF/s)  "
            f"lib {tl:7.1f}us ({fl/tl/1e6:4.0f} TF/s)")
    print(line, flush=True)
    lines.append(line)


# FFN + QKV wgrad shapes for bs64 x L128 (T=8192) and bs64 x L150
for shape in [(8192, 3072, 768), (8192, 768, 3072), (8192, 2304, 768),
              (8192, 768, 768), (9600, 3072, 768), (9600, 768, 3072),
              (1000, 768, 768)]:
    check(*shape)

os.makedirs("gpurun_out", exist_ok=True)
with open("gpurun_out/wgrad2_bench.txt", "w") as f:
    f.write("\n".join(lines) + "\n")
