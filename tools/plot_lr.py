#!/usr/bin/env python3
"""LR-schedule visualizer — CLI equivalent of the reference's
04_plot_lr.ipynb.  Simulates any scheduler the trainer supports with the
trainer's own flags and emits CSV (and optionally a PNG / terminal
sparkline) so a schedule can be sanity-checked before a run.

  python tools/plot_lr.py --scheduler cosine_restarts --lr 4e-4 \
      --num_training_steps 20000 --warmup_steps 1000 --cycle_length 5000 \
      --restart_warmup_steps 100 --min_lr_ratio 0.1 [--adjust_step 0] \
      [--csv lr.csv] [--png lr.png]
"""

import argparse
import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from relora_amd.training_utils import get_scheculer  # noqa: E402


def simulate(args):
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=args.lr)
    sched = get_scheculer(
        optimizer=opt, scheduler_type=args.scheduler,
        num_training_steps=args.num_training_steps,
        warmup_steps=args.warmup_steps, min_lr_ratio=args.min_lr_ratio,
        cycle_length=args.cycle_length,
        restart_warmup_steps=args.restart_warmup_steps,
        adjust_step=args.adjust_step)
    lrs = []
    for _ in range(args.num_training_steps):
        lrs.append(opt.param_groups[0]["lr"])
        opt.step()
        sched.step()
    return lrs


def sparkline(lrs, width=100):
    blocks = " ▁▂▃▄▅▆▇█"
    hi = max(lrs) or 1.0
    step = max(1, len(lrs) // width)
    samp = [max(lrs[i:i + step]) for i in range(0, len(lrs), step)]
    return "".join(blocks[min(8, int(v / hi * 8.999))] for v in samp)


def main(argv=None):
    ap = argparse.ArgumentParser(description=__doc__,
                                 formatter_class=argparse.RawDescriptionHelpFormatter)
    ap.add_argument("--scheduler", default="cosine_restarts",
                    choices=["linear", "cosine", "cosine_restarts"])
    ap.add_argument("--lr", type=float, default=4e-4)
    ap.add_argument("--num_training_steps", type=int, default=20000)
    ap.add_argument("--warmup_steps", type=int, default=1000)
    ap.add_argument("--min_lr_ratio", type=float, default=0.1)
    ap.add_argument("--cycle_length", type=int, default=None)
    ap.add_argument("--restart_warmup_steps", type=int, default=100)
    ap.add_argument("--adjust_step", type=int, default=0)
    ap.add_argument("--csv", default=None, help="write step,lr CSV here")
    ap.add_argument("--png", default=None, help="write a PNG here (matplotlib)")
    args = ap.parse_args(argv)

    lrs = simulate(args)
    print(sparkline(lrs))
    print(f"peak {max(lrs):.3e}  final {lrs[-1]:.3e}  min {min(lrs):.3e}  "
          f"steps {len(lrs)}")
    if args.csv:
        with open(args.csv, "w") as fh:
            fh.write("step,lr\n")
            fh.writelines(f"{i},{v:.10e}\n" for i, v in enumerate(lrs))
        print(f"wrote {args.csv}")
    if args.png:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        fig, ax = plt.subplots(figsize=(8, 3))
        ax.plot(lrs, lw=0.8)
        ax.set_xlabel("update step"), ax.set_ylabel("lr")
        ax.set_title(f"{args.scheduler} lr={args.lr}")
        fig.tight_layout()
        fig.savefig(args.png, dpi=120)
        print(f"wrote {args.png}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
