"""Debug harness for the v3 attention kernels: error stats + run-to-run
determinism (race detector) on a given shape.

  python tools/debug_attn.py [--B 2] [--nh 4] [--S 128] [--hd 64] [--runs 10]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from relora_amd.ops import hip


def sdpa_ref_fp32(q, k, v, scale):
    qf, kf, vf = q.float(), k.float(), v.float()
    S = q.shape[-2]
    scores = qf @ kf.transpose(-1, -2) * scale
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
    scores = scores.masked_fill(mask, float("-inf"))
    return torch.softmax(scores, -1) @ vf


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=2)
    p.add_argument("--nh", type=int, default=4)
    p.add_argument("--S", type=int, default=128)
    p.add_argument("--hd", type=int, default=64)
    p.add_argument("--runs", type=int, default=10)
    p.add_argument("--bwd", action="store_true")
    args = p.parse_args()

    torch.manual_seed(0)
    B, nh, S, hd = args.B, args.nh, args.S, args.hd
    scale = hd ** -0.5
    q, k, v = (torch.randn(B, nh, S, hd, device="cuda", dtype=torch.bfloat16)
               for _ in range(3))
    ext = hip.ext()

    ref = sdpa_ref_fp32(q, k, v, scale)
    outs = []
    for i in range(args.runs):
        o, lse = ext.attn_fwd(q, k, v, scale)
        outs.append(o.clone())
    # run-to-run determinism
    for i in range(1, args.runs):
        d = (outs[i] != outs[0]).sum().item()
        if d:
            print(f"NONDETERMINISTIC fwd: run {i} differs in {d} elements")
    err = (outs[0].float() - ref).abs()
    tol = 2e-2 + 2e-2 * ref.abs().clamp_min(1.0)
    bad = err > tol
    print(f"fwd: max_err={err.max().item():.5f} bad={bad.sum().item()} "
          f"({bad.float().mean().item()*100:.4f}%) nan={torch.isnan(outs[0]).sum().item()}")
    if bad.any():
        idx = bad.nonzero()[:8]
        for t in idx:
            b, h, s, d = t.tolist()
            print(f"  [{b},{h},{s},{d}] got={outs[0][b,h,s,d].item():.5f} "
                  f"ref={ref[b,h,s,d].item():.5f}")
        # distribution of bad rows
        rows = bad.any(dim=-1).nonzero()
        print(f"  bad rows ({len(rows)}): first 10: {rows[:10].tolist()}")

    if args.bwd:
        o, lse = ext.attn_fwd(q, k, v, scale)
        do = torch.randn_like(o)
        grads = []
        for i in range(args.runs):
            dq, dk, dv = ext.attn_bwd(q, k, v, o, lse, do, scale)
            grads.append((dq.clone(), dk.clone(), dv.clone()))
        for i in range(1, args.runs):
            for j, name in enumerate(["dq", "dk", "dv"]):
                d = (grads[i][j] != grads[0][j]).sum().item()
                if d:
                    print(f"NONDETERMINISTIC {name}: run {i} differs in {d}")
        qf = q.float().requires_grad_(True)
        kf = k.float().requires_grad_(True)
        vf = v.float().requires_grad_(True)
        scores = qf @ kf.transpose(-1, -2) * scale
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
        r = torch.softmax(scores.masked_fill(mask, float("-inf")), -1) @ vf
        r.backward(do.float())
        for g, refg, name in zip(grads[0], [qf.grad, kf.grad, vf.grad],
                                 ["dq", "dk", "dv"]):
            e = (g.float() - refg).abs()
            t = 3e-2 + 3e-2 * refg.abs().clamp_min(1.0)
            nb = (e > t).sum().item()
            print(f"{name}: max_err={e.max().item():.5f} bad={nb} "
                  f"nan={torch.isnan(g).sum().item()}")


if __name__ == "__main__":
    main()
