"""The actual CLI entry (torchrun_main.py) under the real launcher
(`python -m torch.distributed.run`), 2 processes over gloo on CPU —
the exact invocation shape the reference documents (README.md:27-29)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_torchrun_two_proc_cli(tmp_path):
    # hermetic environment: prior in-process tests leave launcher/rendezvous
    # state (MASTER_*, TCPStore listeners, gloo sockets) that can poison the
    # child rendezvous — start from a minimal env instead of inheriting
    keep = ("PATH", "HOME", "TMPDIR", "LD_LIBRARY_PATH", "ROCM_PATH",
            "HSA_ENABLE_IPC_MODE_LEGACY", "PYTHONPATH", "HIP_VISIBLE_DEVICES")
    env = {k: os.environ[k] for k in keep if k in os.environ}
    env["RELORA_AMD_NO_TQDM"] = "1"
    env["OMP_NUM_THREADS"] = "1"
    if os.environ.get("RELORA_AMD_HANG_DUMP_S"):
        env["RELORA_AMD_HANG_DUMP_S"] = os.environ["RELORA_AMD_HANG_DUMP_S"]
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
    # write through real files, detached from pytest's captured fds (the
    # piped form hangs under pytest's fd-level capture)
    out_path = tmp_path / "torchrun.out"
    with open(out_path, "w") as out:
        res = subprocess.run(
            cmd, cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            env=env, stdin=subprocess.DEVNULL, stdout=out,
            stderr=subprocess.STDOUT, timeout=560, start_new_session=True)
    assert res.returncode == 0, out_path.read_text()[-2000:]
    state = json.load(open(tmp_path / "run" / "model_4" / "training_state.json"))
    assert state["update_step"] == 4
    # tokens counted x world_size: 4 update steps x 8 x 32 tokens
    assert state["tokens_seen"] == 4 * 8 * 32


@pytest.mark.timeout(600)
def test_torchrun_megatron_zero_two_proc(tmp_path):
    """Megatron data path + ZeRO-1 under the real launcher (2-proc gloo):
    DistributedBatchSampler rank sharding, index-map rank-0 build + barrier,
    ZeRO shard broadcasts and consolidation on save."""
    import numpy as np
    import torch as _torch
    import yaml as _yaml

    from relora_amd.data import indexed_dataset as idx_ds

    rng = np.random.RandomState(1)
    prefix = str(tmp_path / "corpus")
    builder = idx_ds.make_builder(idx_ds.data_file_path(prefix), "mmap", vocab_size=32000)
    for _ in range(300):
        doc = rng.randint(1, 31999, size=rng.randint(8, 64)).astype(np.int64)
        builder.add_item(_torch.from_numpy(doc))
        builder.end_document()
    builder.finalize(idx_ds.index_file_path(prefix))
    cfg_yaml = tmp_path / "m.yaml"
    with open(cfg_yaml, "w") as f:
        _yaml.safe_dump({
            "train_data_paths": [prefix], "valid_data_paths": [prefix],
            "test_data_paths": [prefix], "data_impl": "mmap", "seq_length": 32,
            "train_iters": 64, "eval_interval": 32, "eval_iters": 1,
        }, f)

    keep = ("PATH", "HOME", "TMPDIR", "LD_LIBRARY_PATH", "ROCM_PATH",
            "HSA_ENABLE_IPC_MODE_LEGACY", "PYTHONPATH", "HIP_VISIBLE_DEVICES")
    env = {k: os.environ[k] for k in keep if k in os.environ}
    env["RELORA_AMD_NO_TQDM"] = "1"
    env["OMP_NUM_THREADS"] = "1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--standalone", "--local-addr", "127.0.0.1",
        "--nnodes=1", "--nproc-per-node=2",
        "torchrun_main.py",
        "--model_config", "configs/llama_9m.json",
        "--megatron_dataset_config", str(cfg_yaml),
        "--use_peft", "true", "--optimizer", "adam_zero",
        "--relora", "2", "--cycle_length", "2",
        "--restart_warmup_steps", "1", "--warmup_steps", "1",
        "--scheduler", "cosine_restarts",
        "--num_training_steps", "4",
        "--batch_size", "2", "--total_batch_size", "4",
        "--max_length", "32", "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "2", "--save_every", "4", "--workers", "0",
        "--save_dir", str(tmp_path / "run"),
    ]
    out_path = tmp_path / "torchrun_megatron.out"
    with open(out_path, "w") as out:
        res = subprocess.run(
            cmd, cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            env=env, stdin=subprocess.DEVNULL, stdout=out,
            stderr=subprocess.STDOUT, timeout=560)
    assert res.returncode == 0, out_path.read_text()[-2500:]
    state = json.load(open(tmp_path / "run" / "model_4" / "training_state.json"))
    assert state["update_step"] == 4
    assert state["n_lora_restarts"] >= 1
    # ZeRO consolidated optimizer state is in the checkpoint
    opt = __import__("torch").load(tmp_path / "run" / "model_4" / "optimizer.pt",
                                   map_location="cpu", weights_only=False)
    assert "optimizer" in opt and opt["optimizer"]["state"]


@pytest.mark.parametrize("script", ["torchrun_main.py", "run_glue.py",
                                    "pretokenize.py", "bench.py"])
def test_cli_help_exits_zero(script):
    """Every entry-point script imports cleanly and prints --help (guards
    import-time breakage of the CLI surface)."""
    import subprocess
    import sys as _sys

    r = subprocess.run([_sys.executable, os.path.join(REPO, script), "--help"],
                       capture_output=True, text=True, timeout=180,
                       env={**os.environ, "PYTHONPATH": REPO}, cwd=REPO)
    assert r.returncode == 0, r.stderr[-1500:]
    assert "usage" in r.stdout.lower() or "usage" in r.stderr.lower()
