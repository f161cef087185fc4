"""Per-shape GEMM microbenchmark: torch.matmul (hipBLASLt/rocBLAS via
TunableOp) on the flagship training shapes.

  python tools/bench_gemm.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from relora_amd.ops.tunable import enable_tuned_gemms

SHAPES = [
    # (M, K, N, tag)  — y[M,N] = x[M,K] @ w[N,K]^T
    (16384, 2048, 2048, "qkvo fwd (nt)"),
    (16384, 2048, 5461, "gate/up fwd (nt)"),
    (16384, 5461, 2048, "down fwd (nt)"),
    (16384, 2048, 32100, "lm_head fwd (nt)"),
    (16384, 128, 2048, "loraB-ish (nt)"),
]


def main():
    enable_tuned_gemms()
    for M, K, N, tag in SHAPES:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        for _ in range(5):
            y = x @ w.t()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            y = x @ w.t()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        fl = 2.0 * M * K * N
        print(f"{tag:22s} M{M} K{K} N{N}: {dt*1e6:9.1f} us  {fl/dt/1e12:7.0f} TF/s")
        # NN variant (dx = dy @ w): reduction over N
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        for _ in range(5):
            dx = dy @ w
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            dx = dy @ w
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        print(f"{'  bwd dx (nn)':22s} M{M} K{N} N{K}: {dt*1e6:9.1f} us  {fl/dt/1e12:7.0f} TF/s")


if __name__ == "__main__":
    main()
