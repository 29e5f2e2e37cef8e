"""gemm_nt (hand-written NT MFMA GEMM, csrc/gemm_nt.hip) vs hipBLASLt
on the BERT forward linear shapes. Writes gpurun_out/gemm_nt_bench.txt."""
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


def check(M, N, K, bias=True):
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    bv = torch.randn(N, device="cuda") if bias else None
    ref = a.float() @ b.float().T
    if bias:
        ref = ref + bv
    got = ext.gemm_nt(a, b, bv, False)
    rel = ((got.float() - ref).abs().max() / ref.abs().max()).item()
    t_us = bench(lambda: ext.gemm_nt(a, b, bv, False))
    if bias:
        bb = bv.to(torch.bfloat16)
        t_lib = bench(lambda: torch.nn.functional.linear(a, b, bb))
    else:
        t_lib = bench(lambda: a @ b.T)
    fl = 2.0 * M * N * K
    line = (f"M={M:5d} N={N:4d} K={K:4d} bias={int(bias)}: rel={rel:.4f}  "
            f"ours {t_us:7.1f}us ({fl/t_us/1e6:4.0f} TF/s)  "
            f"lib {t_lib:7.1f}us ({fl/t_lib/1e6:4.0f} TF/s)  "
            f"ratio {t_lib/t_us:.2f}x")
    print(line, flush=True)
    lines.append(line)


# bs64 x L128 (T=8192) and bs64 x L150 (T=9600) forward shapes
for shape in [(8192, 2304, 768), (8192, 768, 768), (8192, 3072, 768),
              (8192, 768, 3072), (9600, 2304, 768), (9600, 3072, 768),
              (9600, 768, 3072)]:
    check(*shape, bias=True)
check(8192, 3072, 768, bias=False)
check(4096, 4096, 4096, bias=False)   # guide ladder reference shape


def check_bwd(M, N, K):
    """dgrad/wgrad candidates incl. the transpose cost."""
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    fl_dx = 2.0 * M * N * K
    t1 = bench(lambda: ext.gemm_nt(dy, w.t().contiguous(), None, False))
    t2 = bench(lambda: dy @ w)
    t3 = bench(lambda: ext.gemm_nt(dy.t().contiguous(), x.t().contiguous(),
                                   None, False))
    t4 = bench(lambda: dy.T @ x)
    line = (f"bwd M={M:5d} N={N:4d} K={K:4d}: "
            f"dx ours {t1:6.1f}us ({fl_dx/t1/1e6:4.0f}) lib {t2:6.1f}us "
            f"({fl_dx/t2/1e6:4.0f}) | dw ours {t3:6.1f}us "
            f"({fl_dx/t3/1e6:4.0f}) lib {t4:6.1f}us ({fl_dx/t4/1e6:4.0f})")
    print(line, flush=True)
    lines.append(line)


for shape in [(8192, 3072, 768), (8192, 768, 3072), (8192, 2304, 768),
              (8192, 768, 768), (9600, 3072, 768)]:
    check_bwd(*shape)

os.makedirs("gpurun_out", exist_ok=True)
with open("gpurun_out/gemm_nt_bench.txt", "w") as f:
    f.write("\n".join(lines) + "\n")
