#!/usr/bin/env python3
"""Rank analysis of ReLoRA weight updates — the CLI equivalent of the
reference's analysis notebooks (05_check_ranks / 06_svd /
08_ranks_before_and_after.ipynb): singular-value spectra and effective
ranks of either (a) the low-rank update B@A·s carried by a wrapped
checkpoint, or (b) the accumulated full update W_after - W_before between
two checkpoints (this is ReLoRA's central claim: many low-rank updates
compose to a high-rank total update).

Usage:
  python tools/analyze_ranks.py --checkpoint runs/x/model_5000 \
      [--baseline runs/x/model_1000] [--filter attn] [--top 8] \
      [--jsonl out.jsonl] [--plot spectra.png]

With only --checkpoint: if the state dict contains lora_A/lora_B pairs the
per-module update is B@A·s; otherwise plain weights are analyzed.
With --baseline: analyzes the difference of same-named 2-D weights.
"""

import argparse
import json
import math
import os
import sys

import torch


def _load_sd(path):
    f = os.path.join(path, "pytorch_model.bin") if os.path.isdir(path) else path
    return torch.load(f, map_location="cpu", weights_only=True)


def _scaling(ckpt_dir):
    """lora scaling = alpha / r from relora_config.json when present."""
    cfg = os.path.join(ckpt_dir, "relora_config.json")
    if os.path.isdir(ckpt_dir) and os.path.exists(cfg):
        with open(cfg) as fh:
            c = json.load(fh)
        r = c.get("r")
        alpha = c.get("lora_alpha", r)
        if r:
            return float(alpha) / float(r)
    return 1.0


def collect_updates(args):
    """-> list of (name, 2-D fp32 tensor to decompose)."""
    sd = _load_sd(args.checkpoint)
    out = []
    if args.baseline:
        base = _load_sd(args.baseline)
        for k, v in sd.items():
            if v.ndim == 2 and k in base and base[k].shape == v.shape:
                out.append((k, (v.float() - base[k].float())))
    else:
        lora_as = {k: v for k, v in sd.items() if k.endswith("lora_A.weight")}
        if lora_as:
            s = _scaling(args.checkpoint)
            for ka, a in lora_as.items():
                kb = ka.replace("lora_A.", "lora_B.")
                if kb in sd:
                    name = ka[: -len(".lora_A.weight")]
                    out.append((name, sd[kb].float() @ a.float() * s))
        else:
            out = [(k, v.float()) for k, v in sd.items() if v.ndim == 2]
    if args.filter:
        out = [(k, v) for k, v in out if args.filter in k]
    return out


def spectrum_stats(m, top):
    sv = torch.linalg.svdvals(m)
    total = sv.sum()
    if total <= 0:
        return dict(top_sv=[0.0] * top, rank90=0, rank99=0, effective_rank=0.0,
                    frob=0.0)
    c = torch.cumsum(sv, 0) / total
    p = sv / total
    p = p[p > 0]
    # effective rank = exp(entropy of normalized spectrum) [Roy & Vetterli]
    eff = float(torch.exp(-(p * p.log()).sum()))
    return dict(
        top_sv=[float(x) for x in sv[:top]],
        rank90=int((c < 0.90).sum()) + 1,
        rank99=int((c < 0.99).sum()) + 1,
        effective_rank=eff,
        frob=float(torch.linalg.vector_norm(m)),
    )


def main(argv=None):
    ap = argparse.ArgumentParser(description=__doc__,
                                 formatter_class=argparse.RawDescriptionHelpFormatter)
    ap.add_argument("--checkpoint", required=True,
                    help="model_<step> dir or a pytorch_model.bin")
    ap.add_argument("--baseline", default=None,
                    help="earlier checkpoint; analyze W_ckpt - W_baseline")
    ap.add_argument("--filter", default=None, help="substring filter on names")
    ap.add_argument("--top", type=int, default=8, help="top singular values kept")
    ap.add_argument("--jsonl", default=None, help="write one JSON line per module")
    ap.add_argument("--plot", default=None,
                    help="write a singular-spectra PNG (needs matplotlib)")
    args = ap.parse_args(argv)

    updates = collect_updates(args)
    if not updates:
        print("no 2-D weights matched", file=sys.stderr)
        return 1

    rows, spectra = [], []
    for name, m in updates:
        st = spectrum_stats(m, args.top)
        st["name"] = name
        st["shape"] = list(m.shape)
        rows.append(st)
        spectra.append((name, m))

    w = max(len(r["name"]) for r in rows)
    print(f"{'module':<{w}}  {'shape':>12}  {'eff.rank':>8}  {'r@90%':>6}  "
          f"{'r@99%':>6}  {'|Δ|_F':>10}  top σ")
    for r in rows:
        print(f"{r['name']:<{w}}  {str(tuple(r['shape'])):>12}  "
              f"{r['effective_rank']:>8.1f}  {r['rank90']:>6}  {r['rank99']:>6}  "
              f"{r['frob']:>10.3e}  "
              + " ".join(f"{x:.2e}" for x in r["top_sv"][:4]))
    full = min(min(r["shape"]) for r in rows)
    mean_eff = sum(r["effective_rank"] for r in rows) / len(rows)
    print(f"\n{len(rows)} modules; mean effective rank {mean_eff:.1f} "
          f"(full rank would be ≤ {full})")

    if args.jsonl:
        with open(args.jsonl, "w") as fh:
            for r in rows:
                fh.write(json.dumps(r) + "\n")
        print(f"wrote {args.jsonl}")
    if args.plot:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        fig, ax = plt.subplots(figsize=(7, 4.5))
        for name, m in spectra:
            sv = torch.linalg.svdvals(m)
            ax.semilogy(sv.numpy(), lw=0.8, label=name if len(spectra) <= 8 else None)
        ax.set_xlabel("index"), ax.set_ylabel("singular value")
        ax.set_title(os.path.basename(str(args.checkpoint)))
        if len(spectra) <= 8:
            ax.legend(fontsize=6)
        fig.tight_layout()
        fig.savefig(args.plot, dpi=120)
        print(f"wrote {args.plot}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
