"""LoRA kernel microbenchmark at flagship shapes (llama_1b, bs8 x seq2048)."""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from relora_amd.ops import hip


def t(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ext = hip.ext()
    M, K, N, r = 16384, 2048, 2048, 128
    y = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    tu = torch.randn(M, r, device="cuda", dtype=torch.bfloat16) * 0.1
    bs = torch.randn(N, r, device="cuda", dtype=torch.bfloat16) * 0.1
    A = torch.randn(r, K, device="cuda", dtype=torch.bfloat16) * 0.1
    us = torch.randn(M, r, device="cuda", dtype=torch.bfloat16) * 0.1
    dx = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    xd, mask = ext.dropout_mask_fwd(x, 0.1, 7)
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)

    d = t(lambda: ext.dropout_mask_fwd(x, 0.1, 7))
    print(f"dropout_mask  : {d*1e6:8.1f} us  ({(2*M*K*2 + M*K//8)/d/1e9:6.0f} GB/s)")
    d = t(lambda: ext.lora_add_nt_(y, tu, bs))
    print(f"lora_add_nt   : {d*1e6:8.1f} us  ({(2*M*N*2)/d/1e9:6.0f} GB/s eff)")
    d = t(lambda: ext.lora_add_nn_(dx, us, A, mask, 1.0 / 0.9))
    print(f"lora_add_nn   : {d*1e6:8.1f} us  ({(2*M*K*2)/d/1e9:6.0f} GB/s eff)")
    nomask = torch.empty(0, device="cuda", dtype=torch.uint8)
    d = t(lambda: ext.skinny_grad(us, x, mask, 1.0 / 0.9, 1.0, False, torch.bfloat16))
    print(f"skinny_grad dA: {d*1e6:8.1f} us  ({(M*K*2)/d/1e9:6.0f} GB/s eff, masked)")
    d = t(lambda: ext.skinny_grad(tu, dy, nomask, 1.0, 0.25, True, torch.bfloat16))
    print(f"skinny_grad dB: {d*1e6:8.1f} us  ({(M*N*2)/d/1e9:6.0f} GB/s eff)")


if __name__ == "__main__":
    main()
