#!/usr/bin/env python3
"""Optimizer-checkpoint inspector — CLI equivalent of the reference's
13_zero_optimizer_resets.ipynb: loads a checkpoint's optimizer.pt and
reports, per parameter state, the exp_avg / exp_avg_sq shapes and the
fraction of exactly-zero entries.  After a ReLoRA reset with magnitude
pruning p, LoRA-param states should show ≈p zeros; after zero-reset, ≈1.0;
frozen-param states (full-rank runs) should show ≈0.

  python tools/inspect_optimizer.py runs/x/model_5000/optimizer.pt
"""

import argparse
import os
import sys

import torch


def walk_states(opt_sd):
    """Yield (param_key, state_dict) over both plain torch.optim layouts and
    our ZeRO consolidated layout."""
    state = opt_sd.get("state", opt_sd)
    for k in sorted(state, key=str):
        s = state[k]
        if isinstance(s, dict) and any(torch.is_tensor(v) for v in s.values()):
            yield k, s


def main(argv=None):
    ap = argparse.ArgumentParser(description=__doc__,
                                 formatter_class=argparse.RawDescriptionHelpFormatter)
    ap.add_argument("path", help="optimizer.pt or a model_<step> directory")
    ap.add_argument("--keys", nargs="*", default=["exp_avg", "exp_avg_sq"])
    args = ap.parse_args(argv)

    path = args.path
    if os.path.isdir(path):
        path = os.path.join(path, "optimizer.pt")
    try:
        ckpt = torch.load(path, map_location="cpu", weights_only=True)
    except Exception:
        # the checkpoint carries saved RNG states (numpy tuples), which the
        # weights_only unpickler rejects; these are our own local checkpoints
        ckpt = torch.load(path, map_location="cpu", weights_only=False)
    opt_sd = ckpt.get("optimizer", ckpt)
    if "update_step" in ckpt:
        print(f"update_step {ckpt['update_step']}  global_step "
              f"{ckpt.get('global_step')}  dtype {ckpt.get('dtype')}")

    rows = list(walk_states(opt_sd))
    if not rows:
        print("no tensor states found", file=sys.stderr)
        return 1
    print(f"{'param':>6}  {'shape':>16}  {'numel':>10}  "
          + "  ".join(f"{k}:%zero" for k in args.keys))
    tot = {k: [0, 0] for k in args.keys}
    for pk, s in rows:
        shapes = next((tuple(v.shape) for v in s.values() if torch.is_tensor(v)), ())
        numel = next((v.numel() for v in s.values() if torch.is_tensor(v)), 0)
        cells = []
        for k in args.keys:
            v = s.get(k)
            if torch.is_tensor(v):
                z = int((v == 0).sum())
                tot[k][0] += z
                tot[k][1] += v.numel()
                cells.append(f"{z / max(1, v.numel()):14.3f}")
            else:
                cells.append(f"{'—':>14}")
        print(f"{str(pk):>6}  {str(shapes):>16}  {numel:>10}  " + "  ".join(cells))
    print("\ntotal: " + "  ".join(
        f"{k} {z}/{n} zero ({z / max(1, n):.1%})" for k, (z, n) in tot.items()))
    return 0


if __name__ == "__main__":
    sys.exit(main())
