"""bench.py driver contract: one JSON line on stdout (rank 0) with the
fields the round driver parses, runnable end-to-end on CPU with a tiny
model (the GPU run only changes device/dtype, not the contract)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def test_bench_json_contract(tmp_path):
    env = {**os.environ, "PYTHONPATH": REPO, "RELORA_AMD_NO_TQDM": "1"}
    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_ADDR", "MASTER_PORT"):
        env.pop(k, None)
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--model", "configs/llama_9m.json", "--steps", "2", "--warmup", "1",
         "--batch_size", "2", "--seq_len", "32", "--dtype", "fp32",
         "--lora_r", "8"],
        capture_output=True, text=True, env=env, cwd=REPO, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, r.stdout[-2000:]
    out = json.loads(json_lines[0])

    assert REQUIRED <= set(out), REQUIRED - set(out)
    assert out["n_gpus"] == 1
    assert out["steps"] == 2 and out["warmup"] == 1
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["unit"] == "tokens/s"
    cfg = out["config"]
    for k in ("model", "global_batch", "seq_len", "parallelism"):
        assert k in cfg, k
    assert cfg["parallelism"] == "dp1"
    assert cfg["global_batch"] == 2 and cfg["seq_len"] == 32
    # whole-job aggregate: value * elapsed == tokens processed
    tokens = cfg["global_batch"] * cfg["seq_len"] * out["steps"]
    elapsed_s = out["ms_per_step"] * out["steps"] / 1000
    assert abs(out["value"] - tokens / elapsed_s) / out["value"] < 1e-6
