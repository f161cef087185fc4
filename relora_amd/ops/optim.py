"""Optimizer-side fused ops: multi-tensor AdamW (K11) and grad-norm clip (K12).

`AdamW` matches `torch.optim.AdamW` semantics exactly (decoupled weight
decay, bias correction, state dict layout with `exp_avg`/`exp_avg_sq` per
param — the layout `optimizer_reset` relies on, reference
training_utils.py:305-361). On ROCm GPUs the step runs as one hand-written
multi-tensor HIP kernel per (device, dtype) group; elsewhere it uses
torch._foreach_* (still vectorized, used by CPU tests as the oracle).

`clip_grad_norm_` reproduces `torch.nn.utils.clip_grad_norm_(...,
error_if_nonfinite=True)` (reference torchrun_main.py:805-808) with a fused
multi-tensor L2-norm + scale on GPU.
"""

import math

import torch

from relora_amd.ops import hip


class AdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=1e-2):
        if lr < 0.0:
            raise ValueError(f"Invalid learning rate: {lr}")
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params, grads, exp_avgs, exp_avg_sqs, steps = [], [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.tensor(0.0)
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1
                params.append(p)
                grads.append(p.grad)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
                steps.append(int(state["step"].item()))
            if not params:
                continue

            beta1, beta2 = group["betas"]
            lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]

            if params[0].is_cuda and hip.use_hip(params[0], "adamw"):
                # group by identical step count (true except after partial loads)
                by_step = {}
                for i, s in enumerate(steps):
                    by_step.setdefault(s, []).append(i)
                for s, idxs in by_step.items():
                    hip.ext().fused_adamw(
                        [params[i] for i in idxs],
                        [grads[i] for i in idxs],
                        [exp_avgs[i] for i in idxs],
                        [exp_avg_sqs[i] for i in idxs],
                        lr, beta1, beta2, eps, wd, s,
                    )
            else:
                self._foreach_step(params, grads, exp_avgs, exp_avg_sqs, steps,
                                   beta1, beta2, lr, eps, wd)
        return loss

    @staticmethod
    def _foreach_step(params, grads, exp_avgs, exp_avg_sqs, steps, beta1, beta2, lr, eps, wd):
        if wd != 0:
            torch._foreach_mul_(params, 1 - lr * wd)
        torch._foreach_lerp_(exp_avgs, grads, 1 - beta1)
        torch._foreach_mul_(exp_avg_sqs, beta2)
        torch._foreach_addcmul_(exp_avg_sqs, grads, grads, 1 - beta2)
        # per-tensor bias correction (steps can differ after resets/loads)
        for p, m, v, s in zip(params, exp_avgs, exp_avg_sqs, steps):
            bc1 = 1 - beta1 ** s
            bc2 = 1 - beta2 ** s
            denom = (v.float() / bc2).sqrt_().add_(eps)
            p.data.add_(((-lr / bc1) * m.float() / denom).to(p.dtype))


@torch.no_grad()
def clip_grad_norm_(parameters, max_norm, norm_type=2.0, error_if_nonfinite=True):
    """Global L2-norm clip over `parameters` grads; returns the total norm."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    grads = [p.grad for p in parameters if p.grad is not None]
    if len(grads) == 0:
        return torch.tensor(0.0)
    assert norm_type == 2.0, "only L2 clipping is supported"
    device = grads[0].device

    if grads[0].is_cuda and hip.use_hip(grads[0], "clip"):
        total_norm = hip.ext().multi_tensor_l2norm(grads)
    else:
        total_norm = torch.linalg.vector_norm(
            torch.stack([torch.linalg.vector_norm(g.detach().float()) for g in grads])
        )

    if error_if_nonfinite and (torch.isnan(total_norm) or torch.isinf(total_norm)):
        raise RuntimeError(
            f"The total norm of order {norm_type} for gradients from `parameters` "
            f"is non-finite, so it cannot be clipped."
        )
    clip_coef = max_norm / (total_norm + 1e-6)
    if clip_coef < 1:
        if grads[0].is_cuda and hip.use_hip(grads[0], "clip"):
            hip.ext().multi_tensor_scale_(grads, float(clip_coef))
        else:
            torch._foreach_mul_(grads, clip_coef.to(device))
    return total_norm
