"""Analysis CLIs (tools/) — the framework's replacement for the reference's
analysis notebooks (05_check_ranks / 06_svd / 08_ranks_before_and_after /
04_plot_lr / 13_zero_optimizer_resets.ipynb): exercised end-to-end on a
tiny CPU ReLoRA run's real checkpoint."""

import json
import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "tools"))


@pytest.fixture(scope="module")
def tiny_run(tmp_path_factory):
    """One 6-step ReLoRA run (save_every=3 -> two checkpoints)."""
    tmp = tmp_path_factory.mktemp("toolsrun")
    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_PORT"):
        os.environ.pop(k, None)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    from relora_amd.trainer import main, parse_args
    args = parse_args([
        "--model_config", "configs/llama_9m.json",
        "--synthetic_data", "true", "--use_peft", "true",
        "--relora", "3", "--cycle_length", "3", "--restart_warmup_steps", "1",
        "--scheduler", "cosine_restarts", "--warmup_steps", "2",
        "--num_training_steps", "6", "--batch_size", "2",
        "--total_batch_size", "2", "--max_length", "32", "--lr", "1e-3",
        "--dtype", "float32", "--eval_every", "100", "--save_every", "3",
        "--reset_optimizer_on_relora", "False",
        "--optimizer_magnitude_pruning", "0.9",
        "--workers", "0", "--save_dir", str(tmp / "run"),
    ])
    main(args)
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()
    return tmp / "run"


def test_analyze_ranks_lora_update(tiny_run, tmp_path, capsys):
    import analyze_ranks
    out = tmp_path / "ranks.jsonl"
    rc = analyze_ranks.main(["--checkpoint", str(tiny_run / "model_6"),
                             "--jsonl", str(out), "--top", "4"])
    assert rc == 0
    rows = [json.loads(l) for l in out.read_text().splitlines()]
    assert rows, "no lora modules analyzed"
    cfg = json.load(open(tiny_run / "model_6" / "relora_config.json"))
    for r in rows:
        # a single B@A update has rank <= r
        assert r["rank99"] <= cfg["r"]
        assert r["effective_rank"] <= cfg["r"] + 1e-6
        assert len(r["top_sv"]) == 4
    text = capsys.readouterr().out
    assert "mean effective rank" in text


def test_analyze_ranks_between_checkpoints(tiny_run, tmp_path):
    import analyze_ranks
    out = tmp_path / "diff.jsonl"
    rc = analyze_ranks.main([
        "--checkpoint", str(tiny_run / "model_6"),
        "--baseline", str(tiny_run / "model_3"),
        "--filter", "attn", "--jsonl", str(out)])
    assert rc == 0
    rows = [json.loads(l) for l in out.read_text().splitlines()]
    assert rows and all("attn" in r["name"] for r in rows)
    # the runs trained, so the accumulated update is nonzero somewhere
    assert any(r["frob"] > 0 for r in rows)


def test_plot_lr_csv_matches_scheduler(tmp_path, capsys):
    import plot_lr
    csv = tmp_path / "lr.csv"
    rc = plot_lr.main(["--scheduler", "cosine_restarts", "--lr", "0.01",
                       "--num_training_steps", "60", "--warmup_steps", "5",
                       "--cycle_length", "20", "--restart_warmup_steps", "3",
                       "--min_lr_ratio", "0.1", "--csv", str(csv)])
    assert rc == 0
    lines = csv.read_text().splitlines()
    assert lines[0] == "step,lr"
    lrs = [float(l.split(",")[1]) for l in lines[1:]]
    assert len(lrs) == 60
    # golden cross-check against the scheduler itself
    from relora_amd.training_utils import get_scheculer
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=0.01)
    s = get_scheculer(optimizer=opt, scheduler_type="cosine_restarts",
                      num_training_steps=60, warmup_steps=5, min_lr_ratio=0.1,
                      cycle_length=20, restart_warmup_steps=3, adjust_step=0)
    for i in range(60):
        assert abs(opt.param_groups[0]["lr"] - lrs[i]) < 1e-12, i
        opt.step(); s.step()


def test_inspect_optimizer_reports_pruned_states(tiny_run, capsys):
    import inspect_optimizer
    rc = inspect_optimizer.main([str(tiny_run / "model_6")])
    assert rc == 0
    out = capsys.readouterr().out
    assert "update_step 6" in out
    assert "exp_avg:%zero" in out
    # magnitude pruning 0.9 fired at step 3 and 6 -> a large zero fraction
    # across lora states; parse the total line
    total = [l for l in out.splitlines() if l.startswith("total:")][0]
    frac = float(total.split("(")[1].split("%")[0]) / 100.0
    assert frac > 0.3, total


def test_tools_run_as_scripts(tiny_run, tmp_path):
    """The CLIs work as plain scripts too (the notebook-replacement UX)."""
    env = {**os.environ, "PYTHONPATH": REPO}
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "plot_lr.py"),
         "--num_training_steps", "30", "--warmup_steps", "3",
         "--cycle_length", "10", "--png", str(tmp_path / "lr.png")],
        capture_output=True, text=True, env=env, timeout=120)
    assert r.returncode == 0, r.stderr
    assert (tmp_path / "lr.png").exists()
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "analyze_ranks.py"),
         "--checkpoint", str(tiny_run / "model_6"),
         "--plot", str(tmp_path / "spectra.png")],
        capture_output=True, text=True, env=env, timeout=300)
    assert r.returncode == 0, r.stderr
    assert (tmp_path / "spectra.png").exists()


def test_search_lds_layout_tool():
    """The layout searcher verifies Q_V2 as a zero-conflict solution and
    scores the current rotation's residual conflicts."""
    import subprocess
    env = {**os.environ, "PYTHONPATH": REPO}
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "search_lds_layout.py")],
        capture_output=True, text=True, env=env, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "Q_V2          : 0" in r.stdout
