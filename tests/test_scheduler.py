"""Scheduler parity tests: golden-value checks of the cosine_restarts /
cyclical-cosine lambdas against an independent re-derivation of the
reference semantics (reference training_utils.py:173-236)."""

import math

import pytest
import torch

from relora_amd.training_utils import get_scheculer


def make_sched(**kw):
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=1.0)
    return opt, get_scheculer(optimizer=opt, **kw)


def lrs(opt, sched, n):
    out = []
    for _ in range(n):
        out.append(opt.param_groups[0]["lr"])
        opt.step()
        sched.step()
    return out


def expected_cosine_restarts(step, *, total, warmup, rwarm, every, minr, adjust=0):
    if step < warmup:
        return step / max(1, warmup)
    s = step + adjust
    rstep, rnum = s % every, s // every
    if rstep < rwarm and step >= every:
        prog = (rnum * every + rwarm - warmup) / max(1, total - warmup)
        peak = minr + (1 - minr) * 0.5 * (1 + math.cos(math.pi * prog))
        return rstep / max(1, rwarm) * peak
    prog = (s - warmup) / max(1, total - warmup)
    return minr + (1 - minr) * 0.5 * (1 + math.cos(math.pi * prog))


def test_cosine_restarts_golden():
    total, warmup, rwarm, every, minr = 100, 10, 4, 20, 0.1
    opt, sched = make_sched(
        scheduler_type="cosine_restarts", num_training_steps=total,
        warmup_steps=warmup, min_lr_ratio=minr, cycle_length=every,
        restart_warmup_steps=rwarm,
    )
    got = lrs(opt, sched, total)
    for step, lr in enumerate(got):
        exp = expected_cosine_restarts(
            step, total=total, warmup=warmup, rwarm=rwarm, every=every, minr=minr
        )
        assert lr == pytest.approx(exp, rel=1e-12), f"step {step}: {lr} != {exp}"


def test_cosine_restarts_adjust_step():
    total, warmup, rwarm, every, minr, adj = 100, 5, 4, 20, 0.1, 10
    opt, sched = make_sched(
        scheduler_type="cosine_restarts", num_training_steps=total,
        warmup_steps=warmup, min_lr_ratio=minr, cycle_length=every,
        restart_warmup_steps=rwarm, adjust_step=adj,
    )
    got = lrs(opt, sched, 60)
    for step, lr in enumerate(got):
        exp = expected_cosine_restarts(
            step, total=total, warmup=warmup, rwarm=rwarm, every=every, minr=minr, adjust=adj
        )
        assert lr == pytest.approx(exp, rel=1e-12), f"step {step}"


def test_cosine_restarts_warmup_shape():
    total, warmup, rwarm, every = 60, 6, 3, 20
    opt, sched = make_sched(
        scheduler_type="cosine_restarts", num_training_steps=total,
        warmup_steps=warmup, min_lr_ratio=0.1, cycle_length=every,
        restart_warmup_steps=rwarm,
    )
    got = lrs(opt, sched, total)
    # first warmup ramps 0 -> peak
    assert got[0] == 0.0
    assert got[warmup - 1] < got[warmup] or got[warmup] == pytest.approx(1.0, abs=0.1)
    # at every restart boundary, lr drops to 0 then re-warms
    assert got[every] == 0.0
    assert got[every + 1] > 0
    assert got[every + rwarm] > got[every + 1]
    # envelope decays: restart peaks decrease
    assert got[every + rwarm] > got[2 * every + rwarm]


def test_cosine_restarts_divisibility_error():
    with pytest.raises(ValueError):
        make_sched(
            scheduler_type="cosine_restarts", num_training_steps=105,
            warmup_steps=10, min_lr_ratio=0.1, cycle_length=20,
            restart_warmup_steps=4,
        )


def test_first_reset_before_warmup_error():
    # warmup > cycle_length: the lambda asserts on first evaluation
    with pytest.raises(AssertionError):
        opt, sched = make_sched(
            scheduler_type="cosine_restarts", num_training_steps=100,
            warmup_steps=30, min_lr_ratio=0.1, cycle_length=20,
            restart_warmup_steps=5,
        )
        lrs(opt, sched, 5)


def test_cyclical_cosine_resume_guard():
    total, cycle, warmup = 40, 20, 5
    opt, sched = make_sched(
        scheduler_type="cosine", num_training_steps=total,
        warmup_steps=warmup, min_lr_ratio=0.1, cycle_length=cycle,
    )
    got = lrs(opt, sched, total)
    # in the second cycle, the first two warmup steps return the 1e-7 guard
    assert got[cycle] == pytest.approx(1e-7)
    assert got[cycle + 1] == pytest.approx(1e-7)
    assert got[cycle + 2] == pytest.approx(2 / warmup)


def test_linear_schedule():
    opt, sched = make_sched(
        scheduler_type="linear", num_training_steps=20, warmup_steps=5, min_lr_ratio=0.1
    )
    got = lrs(opt, sched, 20)
    assert got[0] == 0.0
    assert got[5] == pytest.approx(1.0)
    assert got[19] == pytest.approx(1 / 15)


def test_adjust_step_only_for_restarts():
    with pytest.raises(ValueError):
        make_sched(
            scheduler_type="cosine", num_training_steps=20, warmup_steps=5,
            min_lr_ratio=0.1, adjust_step=3,
        )
