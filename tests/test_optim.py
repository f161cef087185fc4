"""Fused AdamW / clip_grad_norm oracle tests vs torch.optim equivalents."""

import pytest
import torch

from relora_amd.ops.optim import AdamW, clip_grad_norm_


def _make(n=3, dtype=torch.float32, seed=0):
    torch.manual_seed(seed)
    params = [torch.nn.Parameter(torch.randn(17 * (i + 1), dtype=dtype)) for i in range(n)]
    grads = [torch.randn_like(p) for p in params]
    return params, grads


def test_adamw_matches_torch():
    params, grads = _make()
    ref_params = [torch.nn.Parameter(p.detach().clone()) for p in params]

    opt = AdamW(params, lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.01)
    ref = torch.optim.AdamW(ref_params, lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.01)

    for step in range(5):
        for p, rp, g in zip(params, ref_params, grads):
            p.grad = g * (step + 1)
            rp.grad = g * (step + 1)
        opt.step()
        ref.step()

    for p, rp in zip(params, ref_params):
        assert torch.allclose(p, rp, atol=1e-6), (p - rp).abs().max()


def test_adamw_state_layout():
    params, grads = _make(1)
    opt = AdamW(params, lr=1e-3)
    params[0].grad = grads[0]
    opt.step()
    state = opt.state[params[0]]
    assert "exp_avg" in state and "exp_avg_sq" in state
    assert state["exp_avg"].shape == params[0].shape


def test_adamw_zero_wd():
    params, grads = _make(1)
    ref_params = [torch.nn.Parameter(p.detach().clone()) for p in params]
    opt = AdamW(params, lr=1e-2, weight_decay=0.0)
    ref = torch.optim.AdamW(ref_params, lr=1e-2, weight_decay=0.0)
    for p, rp, g in zip(params, ref_params, grads):
        p.grad = g
        rp.grad = g
    opt.step()
    ref.step()
    assert torch.allclose(params[0], ref_params[0], atol=1e-7)


def test_clip_grad_norm_matches_torch():
    params, grads = _make()
    for p, g in zip(params, grads):
        p.grad = g.clone() * 10
    ref_params = [torch.nn.Parameter(p.detach().clone()) for p in params]
    for rp, g in zip(ref_params, grads):
        rp.grad = g.clone() * 10

    norm = clip_grad_norm_(params, 1.0)
    ref_norm = torch.nn.utils.clip_grad_norm_(ref_params, 1.0)
    assert torch.allclose(norm, ref_norm, atol=1e-5)
    for p, rp in zip(params, ref_params):
        assert torch.allclose(p.grad, rp.grad, atol=1e-6)


def test_clip_grad_norm_nonfinite_raises():
    params, grads = _make(1)
    params[0].grad = grads[0]
    params[0].grad[0] = float("nan")
    with pytest.raises(RuntimeError):
        clip_grad_norm_(params, 1.0, error_if_nonfinite=True)


def test_clip_no_scale_when_below():
    params, grads = _make(1)
    params[0].grad = grads[0] * 1e-6
    before = params[0].grad.clone()
    clip_grad_norm_(params, 1.0)
    assert torch.equal(params[0].grad, before)


def test_delete_old_checkpoints(tmp_path):
    """--keep_checkpoints GC keeps the N latest model_* dirs
    (reference training_utils.py:406-418)."""
    from relora_amd.training_utils import delete_old_checkpoints

    for step in (10, 20, 30, 40):
        d = tmp_path / f"model_{step}"
        d.mkdir()
        (d / "pytorch_model.bin").write_bytes(b"x")
    (tmp_path / "training_config.yaml").write_text("a: 1")
    delete_old_checkpoints(str(tmp_path), keep=2)
    left = sorted(p.name for p in tmp_path.glob("model_*"))
    assert left == ["model_30", "model_40"]
    assert (tmp_path / "training_config.yaml").exists()
    # keep=None is a no-op
    delete_old_checkpoints(str(tmp_path), keep=None)
    assert sorted(p.name for p in tmp_path.glob("model_*")) == ["model_30", "model_40"]


def test_get_last_training_state(tmp_path):
    from relora_amd.training_utils import get_last_training_state

    import json as _json
    for step in (5, 25, 15):
        d = tmp_path / f"model_{step}"
        d.mkdir()
        with open(d / "training_state.json", "w") as f:
            _json.dump({"update_step": step, "wandb_id": f"id{step}"}, f)
    state, ckpt = get_last_training_state(str(tmp_path))
    assert ckpt.endswith("model_25")
    assert state["update_step"] == 25


def test_print_optimizer_state_size(capsys):
    """Counts Adam moment floats; understands the ZeRO `.optim` indirection
    (reference training_utils.py:367-388)."""
    from relora_amd.training_utils import print_optimizer_state_size

    p = torch.nn.Parameter(torch.randn(100, 10))
    opt = torch.optim.Adam([p], lr=1e-3)
    p.grad = torch.randn_like(p)
    opt.step()
    print_optimizer_state_size(opt)
    out = capsys.readouterr().out
    assert "first moment" in out and "0.00M" in out  # 1000 floats = 0.00M


def test_check_lr_and_alert():
    """Warns (and wandb-alerts) when post-reset lr exceeds the bound
    (reference training_utils.py:391-404).  Captures by swapping the
    logger's handler stream (it binds stderr at import time)."""
    import io

    from relora_amd.training_utils import check_lr_and_alert
    from relora_amd.utils.logging import logger

    buf = io.StringIO()
    old = logger._handler.setStream(buf)
    try:
        p = torch.nn.Parameter(torch.zeros(1))
        opt = torch.optim.SGD([p], lr=0.1)
        check_lr_and_alert(opt, max_lr=1.0)   # fine: no warning
        assert "lr after the reset" not in buf.getvalue()
        check_lr_and_alert(opt, max_lr=0.01)  # too large: warns
        assert "lr after the reset" in buf.getvalue()
    finally:
        logger._handler.setStream(old)
