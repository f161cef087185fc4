"""Attention kernel microbenchmark (MI355X): TF/s for fwd / bwd at the
flagship shape, plus optional SDPA comparison.

  python tools/bench_attention.py [--B 8] [--nh 32] [--S 2048] [--hd 64]
  rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_LDS_BANK_CONFLICT \
      -d out -- python tools/bench_attention.py --iters 3 --what fwd
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from relora_amd.ops import hip


def flops_fwd(B, nh, S, hd):
    # QK^T + PV, causal halves the work
    return 2 * 2 * B * nh * S * S * hd * 0.5


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=8)
    p.add_argument("--nh", type=int, default=32)
    p.add_argument("--S", type=int, default=2048)
    p.add_argument("--hd", type=int, default=64)
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--what", choices=["fwd", "bwd", "both", "sdpa"], default="both")
    args = p.parse_args()

    torch.manual_seed(0)
    B, nh, S, hd = args.B, args.nh, args.S, args.hd
    q = torch.randn(B, nh, S, hd, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    scale = hd ** -0.5
    ext = hip.ext()

    def timeit(fn, iters):
        for _ in range(args.warmup):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    ff = flops_fwd(B, nh, S, hd)
    if args.what in ("fwd", "both"):
        o, lse = ext.attn_fwd(q, k, v, scale)
        t = timeit(lambda: ext.attn_fwd(q, k, v, scale), args.iters)
        print(f"fwd : {t*1e3:8.3f} ms  {ff/t/1e12:7.1f} TF/s")
    if args.what in ("bwd", "both"):
        o, lse = ext.attn_fwd(q, k, v, scale)
        t = timeit(lambda: ext.attn_bwd(q, k, v, o, lse, do, scale), args.iters)
        # bwd: dQ (2 GEMMs) + dKdV (4 GEMMs) + delta
        print(f"bwd : {t*1e3:8.3f} ms  {ff*2.5/t/1e12:7.1f} TF/s")
    if args.what == "sdpa":
        def sdpa():
            return torch.nn.functional.scaled_dot_product_attention(
                q, k, v, is_causal=True, scale=scale)
        t = timeit(sdpa, args.iters)
        print(f"sdpa fwd: {t*1e3:8.3f} ms  {ff/t/1e12:7.1f} TF/s")
        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)

        def sdpa_fb():
            out = torch.nn.functional.scaled_dot_product_attention(
                qg, kg, vg, is_causal=True, scale=scale)
            out.backward(do)
            qg.grad = kg.grad = vg.grad = None
        t = timeit(sdpa_fb, args.iters)
        print(f"sdpa f+b: {t*1e3:8.3f} ms  {ff*3.5/t/1e12:7.1f} TF/s")


if __name__ == "__main__":
    main()
