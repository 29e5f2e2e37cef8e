import os, sys, time
import torch
import torch.nn.functional as F

def bench(fn, n=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

x = torch.randn(8192, 768, device="cuda", dtype=torch.bfloat16)
w1 = torch.randn(3072, 768, device="cuda", dtype=torch.bfloat16)  # ffn_in
dy = torch.randn(8192, 3072, device="cuda", dtype=torch.bfloat16)
print("default lib:")
print("  ffn fwd  x@w1.T:", round(bench(lambda: x @ w1.T), 1), "us")
print("  ffn dX  dy@w1 :", round(bench(lambda: dy @ w1), 1), "us")
print("  ffn dW  dy.T@x:", round(bench(lambda: dy.T @ x), 1), "us")
try:
    torch.backends.cuda.preferred_blas_library("cublas")
    print("preferred=cublas (rocBLAS):")
    print("  ffn fwd :", round(bench(lambda: x @ w1.T), 1), "us")
    print("  ffn dX  :", round(bench(lambda: dy @ w1), 1), "us")
    print("  ffn dW  :", round(bench(lambda: dy.T @ x), 1), "us")
except Exception as e:
    print("cublas switch failed:", e)
try:
    torch.backends.cuda.preferred_blas_library("cublaslt")
    print("preferred=cublaslt (hipBLASLt):")
    print("  ffn fwd :", round(bench(lambda: x @ w1.T), 1), "us")
    print("  ffn dX  :", round(bench(lambda: dy @ w1), 1), "us")
except Exception as e:
    print("lt switch failed:", e)
