"""The actual CLI entry (torchrun_main.py) under the real launcher
(`python -m torch.distributed.run`), 2 processes over gloo on CPU —
the exact invocation shape the reference documents (README.md:27-29)."""

import json
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(600)
def test_torchrun_two_proc_cli(tmp_path):
    env = dict(os.environ)
    # a prior in-process test may have exported MASTER_* while its (destroyed)
    # TCPStore still listens on that port; torchrun's children would rendezvous
    # against the stale store and hang — strip all launcher state
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR", "MASTER_PORT"):
        env.pop(k, None)
    env["RELORA_AMD_NO_TQDM"] = "1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--standalone", "--local-addr", "127.0.0.1",
        "--nnodes=1", "--nproc-per-node=2",
        "torchrun_main.py",
        "--model_config", "configs/llama_9m.json",
        "--synthetic_data", "true",
        "--use_peft", "true",
        "--relora", "2", "--cycle_length", "2",
        "--restart_warmup_steps", "1", "--warmup_steps", "1",
        "--scheduler", "cosine_restarts",
        "--num_training_steps", "4",
        "--batch_size", "2", "--total_batch_size", "8",
        "--max_length", "32", "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "100", "--save_every", "100", "--workers", "0",
        "--save_dir", str(tmp_path / "run"),
    ]
    res = subprocess.run(cmd, cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                         env=env, capture_output=True, text=True, timeout=560)
    assert res.returncode == 0, res.stdout[-2000:] + res.stderr[-2000:]
    state = json.load(open(tmp_path / "run" / "model_4" / "training_state.json"))
    assert state["update_step"] == 4
    # tokens counted x world_size: 4 update steps x 8 x 32 tokens
    assert state["tokens_seen"] == 4 * 8 * 32
