"""A/B the fused K1+K2 GEMM (fused_gemm.hip) against the composed
hipBLASLt F.linear + lora_add_nt_ path, per shape and end-to-end-relevant.

  python tools/bench_fused_gemm.py [--M 16384] [--shapes qkvo,gateup,...]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from relora_amd.ops import hip
from relora_amd.ops.tunable import enable_tuned_gemms


SHAPES = {
    # name: (N, K) with M from CLI; llama_1b qkvo / llama_250m all / 7b qkvo
    "qkvo_1b": (2048, 2048),
    "qkvo_250m": (768, 768),
    "gateup_250m": (2560, 768),
    "down_250m": (768, 2560),
    "qkvo_7b": (4096, 4096),
}


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--M", type=int, default=16384)
    p.add_argument("--r", type=int, default=128)
    p.add_argument("--iters", type=int, default=30)
    args = p.parse_args()
    enable_tuned_gemms()
    ext = hip.ext()
    torch.manual_seed(0)

    for name, (N, K) in SHAPES.items():
        M, r = args.M, args.r
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.1
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
        t = torch.randn(M, r, device="cuda", dtype=torch.bfloat16) * 0.1
        bw = torch.randn(N, r, device="cuda", dtype=torch.bfloat16) * 0.1
        scale = 0.25
        bs = bw * scale
        empty = x.new_empty(0)

        # numerics
        y_fused = ext.fused_lora_gemm(x, w, t, bw, empty, scale)
        y_ref = (x.float() @ w.float().t() + scale * (t.float() @ bw.float().t()))
        err = (y_fused.float() - y_ref).abs()
        tol = 2e-2 + 2e-2 * y_ref.abs().clamp_min(1.0)
        bad = (err > tol).sum().item()
        flops = 2 * M * (K + r) * N

        def composed():
            y = torch.nn.functional.linear(x, w)
            ext.lora_add_nt_(y, t, bs)
            return y

        t_comp = timeit(composed, args.iters)
        t_fused = timeit(lambda: ext.fused_lora_gemm(x, w, t, bw, empty, scale),
                         args.iters)
        # 3-buffer deep-pipelined variant (numerics + determinism screen)
        y3a = ext.fused_lora_gemm3(x, w, t, bw, empty, scale)
        bad3 = ((y3a.float() - y_ref).abs() > tol).sum().item()
        det3 = sum((ext.fused_lora_gemm3(x, w, t, bw, empty, scale) != y3a).sum().item()
                   for _ in range(4))
        t3 = timeit(lambda: ext.fused_lora_gemm3(x, w, t, bw, empty, scale),
                    args.iters)
        y4a = ext.fused_lora_gemm4(x, w, t, bw, empty, scale)
        bad4 = ((y4a.float() - y_ref).abs() > tol).sum().item()
        det4 = sum((ext.fused_lora_gemm4(x, w, t, bw, empty, scale) != y4a).sum().item()
                   for _ in range(4))
        t4 = timeit(lambda: ext.fused_lora_gemm4(x, w, t, bw, empty, scale),
                    args.iters)
        print(f"{name:12s} M{M} N{N} K{K} r{r}: "
              f"composed {t_comp*1e6:7.1f} us ({flops/t_comp/1e12:6.0f} TF) | "
              f"fused {t_fused*1e6:7.1f} us ({flops/t_fused/1e12:6.0f} TF) | "
              f"3buf {t3*1e6:7.1f} us ({flops/t3/1e12:6.0f} TF) | "
              f"bk32x4 {t4*1e6:7.1f} us ({flops/t4/1e12:6.0f} TF) | "
              f"bad={bad}/{bad3}/{bad4} nondet={det3}/{det4} "
              f"maxerr={err.max().item():.4f}")


if __name__ == "__main__":
    main()
