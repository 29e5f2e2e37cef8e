"""BiLSTM fwd/bwd kernel timing on the flagship shape (bs64 L128 h128,
bert_bilstm_crf) + h=200 (softlexicon variant). Correctness is covered
by tests/test_gpu_kernels.py; this times the autograd path."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from chinesener_amd.ops import functional as fn  # noqa: E402

torch.manual_seed(0)


def run(B, L, E, h):
    x = torch.randn(B, L, E, device="cuda", dtype=torch.bfloat16)
    args = []
    for _ in range(2):  # fw, bw
        args += [torch.randn(E, 4 * h, device="cuda", dtype=torch.bfloat16) * 0.05,
                 torch.randn(h, 4 * h, device="cuda", dtype=torch.bfloat16) * 0.05,
                 torch.zeros(4 * h, device="cuda", dtype=torch.bfloat16)]
    lens = torch.full((B,), L, dtype=torch.long, device="cuda")
    x.requires_grad_(True)

    def step():
        out = fn.bilstm(x, args[0], args[1], args[2], args[3], args[4],
                        args[5], lens, activation="relu")
        out.float().pow(2).mean().backward()
        x.grad = None

    for _ in range(5):
        step()
    torch.cuda.synchronize()
    n = 30
    t0 = time.perf_counter()
    for _ in range(n):
        step()
    torch.cuda.synchronize()
    tot = (time.perf_counter() - t0) / n * 1e3
    # fwd-only
    with torch.no_grad():
        def fwd():
            fn.bilstm(x, args[0], args[1], args[2], args[3], args[4],
                      args[5], lens, activation="relu")
        for _ in range(5):
            fwd()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fwd()
        torch.cuda.synchronize()
        f = (time.perf_counter() - t0) / n * 1e3
    print(f"B={B} L={L} E={E} h={h}: fwd {f:.3f} ms  fwd+bwd {tot:.3f} ms "
          f"(bwd ~{tot - f:.3f} ms)", flush=True)


run(64, 128, 768, 128)
run(64, 128, 968, 200)
