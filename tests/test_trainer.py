"""End-to-end trainer tests on CPU synthetic data: loop runs, resets fire,
checkpoints have the reference layout, autoresume restores counters, args
validation matches the reference contract."""

import json
import os

import pytest
import torch

from relora_amd.trainer import main, parse_args


def run_args(tmp_path, extra=None, steps=6):
    base = [
        "--model_config", "configs/llama_9m.json",
        "--synthetic_data", "true",
        "--use_peft", "true",
        "--relora", "3", "--cycle_length", "3",
        "--restart_warmup_steps", "1",
        "--scheduler", "cosine_restarts",
        "--warmup_steps", "2",
        "--num_training_steps", str(steps),
        "--batch_size", "2", "--total_batch_size", "4",
        "--max_length", "32",
        "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "100", "--save_every", "100",
        "--workers", "0",
        "--save_dir", str(tmp_path / "run"),
    ]
    return parse_args(base + (extra or []))


@pytest.fixture(autouse=True)
def _single_proc_env(monkeypatch):
    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_ADDR", "MASTER_PORT"):
        monkeypatch.delenv(k, raising=False)
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    yield
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


def test_trainer_end_to_end(tmp_path):
    args = run_args(tmp_path)
    main(args)
    ckpt = tmp_path / "run" / "model_6"
    assert ckpt.exists()
    for f in ("pytorch_model.bin", "config.json", "relora_config.json",
              "optimizer.pt", "training_state.json"):
        assert (ckpt / f).exists(), f
    state = json.load(open(ckpt / "training_state.json"))
    assert state["update_step"] == 6
    assert state["n_lora_restarts"] >= 1
    assert state["n_optimizer_resets"] >= 1
    assert (tmp_path / "run" / "training_config.yaml").exists()
    opt_ckpt = torch.load(ckpt / "optimizer.pt", map_location="cpu", weights_only=False)
    assert "optimizer" in opt_ckpt and "scheduler" in opt_ckpt
    assert opt_ckpt["update_step"] == 6


def test_trainer_autoresume(tmp_path):
    main(run_args(tmp_path, steps=6))
    # resume to 9 steps (num_training_steps must stay divisible by cycle_length)
    args = run_args(tmp_path, extra=["--autoresume", "true"], steps=9)
    main(args)
    state = json.load(open(tmp_path / "run" / "model_9" / "training_state.json"))
    assert state["update_step"] == 9
    assert state["tokens_seen"] > 0


def test_resume_is_bit_exact(tmp_path):
    """Interrupted-then-resumed training produces exactly the weights and
    optimizer state of an uninterrupted run: optimizer/scheduler restore,
    dataloader fast-forward, and per-rank RNG stream restore (dropout is
    active through lora_dropout) all have to line up.  The reference does
    not restore RNG states, so it cannot make this guarantee."""
    import torch.distributed as dist

    # uninterrupted: 6 steps, checkpoint at 3 and 6
    main(run_args(tmp_path / "full", extra=["--save_every", "3"], steps=6))
    if dist.is_initialized():
        dist.destroy_process_group()
    # interrupted: stop at 3 ...
    main(run_args(tmp_path / "half", extra=["--save_every", "3"], steps=3))
    if dist.is_initialized():
        dist.destroy_process_group()
    # ... resume to 6
    main(run_args(tmp_path / "half", extra=["--save_every", "3",
                                            "--autoresume", "true"], steps=6))

    a = torch.load(tmp_path / "full" / "run" / "model_6" / "pytorch_model.bin",
                   map_location="cpu", weights_only=True)
    b = torch.load(tmp_path / "half" / "run" / "model_6" / "pytorch_model.bin",
                   map_location="cpu", weights_only=True)
    assert set(a) == set(b)
    for k in a:
        assert torch.equal(a[k], b[k]), k
    oa = torch.load(tmp_path / "full" / "run" / "model_6" / "optimizer.pt",
                    map_location="cpu", weights_only=False)
    ob = torch.load(tmp_path / "half" / "run" / "model_6" / "optimizer.pt",
                    map_location="cpu", weights_only=False)
    sa, sb = oa["optimizer"]["state"], ob["optimizer"]["state"]
    assert set(map(str, sa)) == set(map(str, sb))
    for k in sa:
        for key in ("exp_avg", "exp_avg_sq"):
            assert torch.equal(sa[k][key], sb[k][key]), (k, key)


def test_trainer_zero_optimizer(tmp_path):
    args = run_args(tmp_path, extra=["--optimizer", "adam_zero"])
    main(args)
    assert (tmp_path / "run" / "model_6" / "optimizer.pt").exists()


def test_trainer_full_rank(tmp_path):
    base = [
        "--model_config", "configs/llama_9m.json",
        "--synthetic_data", "true",
        "--num_training_steps", "3",
        "--batch_size", "2", "--total_batch_size", "2",
        "--max_length", "32", "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "100", "--save_every", "100", "--workers", "0",
        "--save_dir", str(tmp_path / "runf"),
    ]
    args = parse_args(base)
    assert args.use_peft is False
    main(args)
    assert (tmp_path / "runf" / "model_3" / "pytorch_model.bin").exists()


# ---------------------------------------------------------------------------
# args contract
# ---------------------------------------------------------------------------


def test_args_batch_algebra():
    args = parse_args([
        "--synthetic_data", "true", "--batch_size", "4", "--gradient_accumulation", "3",
    ])
    assert args.total_batch_size == 12


def test_args_requires_batch_size():
    with pytest.raises(ValueError):
        parse_args(["--synthetic_data", "true"])


def test_args_data_source_exclusive():
    with pytest.raises(ValueError):
        parse_args(["--batch_size", "2"])
    with pytest.raises(ValueError):
        parse_args(["--synthetic_data", "true", "--dataset_path", "/tmp/x",
                    "--batch_size", "2"])


def test_args_fp16_rejected():
    with pytest.raises(NotImplementedError):
        parse_args(["--synthetic_data", "true", "--batch_size", "2", "--dtype", "float16"])


def test_args_fsdp_rejected():
    """FSDP is hard-disabled, matching the reference (save_model_fsdp raises,
    torchrun_main.py:227-253)."""
    with pytest.raises(NotImplementedError):
        parse_args(["--synthetic_data", "true", "--batch_size", "2",
                    "--total_batch_size", "2", "--distributed_type", "fsdp"])


def test_args_relora_implies_peft():
    args = parse_args(["--synthetic_data", "true", "--batch_size", "2", "--relora", "10"])
    assert args.use_peft is True


def test_args_reset_modes_exclusive():
    with pytest.raises(ValueError):
        parse_args([
            "--synthetic_data", "true", "--batch_size", "2",
            "--reset_optimizer_on_relora", "true",
            "--optimizer_magnitude_pruning", "0.9",
        ])


def test_args_skip_batches_parsing():
    args = parse_args([
        "--synthetic_data", "true", "--batch_size", "2", "--skip_batches", "3,5,9",
    ])
    assert args.skip_batches == {3, 5, 9}


def test_args_yaml_override(tmp_path):
    import yaml

    cfg = {
        "synthetic_data": True, "batch_size": 2, "total_batch_size": 4,
        "lr": "5e-4", "num_training_steps": 10,
    }
    path = tmp_path / "cfg.yaml"
    path.write_text(yaml.dump(cfg))
    args = parse_args(["--training_config", str(path)])
    assert args.batch_size == 2
    assert args.lr == 5e-4
    assert isinstance(args.lr, float)


# ---------------------------------------------------------------------------
# the reference README.dev.md regime matrix as scripted smoke tests
# (SURVEY.md §4 item 5): each regime must complete a short run on CPU
# ---------------------------------------------------------------------------


def _pythia_config(tmp_path):
    import json as _json
    cfg = {
        "architectures": ["GPTNeoXForCausalLM"], "model_type": "gpt_neox",
        "hidden_size": 32, "intermediate_size": 128, "num_attention_heads": 4,
        "num_hidden_layers": 2, "vocab_size": 128, "max_position_embeddings": 64,
        "rotary_pct": 0.25, "rotary_emb_base": 10000, "use_parallel_residual": True,
        "layer_norm_eps": 1e-5, "initializer_range": 0.02, "tie_word_embeddings": False,
    }
    p = tmp_path / "pythia_tiny.json"
    p.write_text(_json.dumps(cfg))
    return str(p)


def test_regime_pythia_zero_relora(tmp_path):
    args = run_args(tmp_path, extra=[
        "--model_config", _pythia_config(tmp_path),
        "--optimizer", "adam_zero",
    ], steps=6)
    main(args)
    assert (tmp_path / "run" / "model_6" / "training_state.json").exists()


def test_regime_pythia_full_rank(tmp_path):
    args = parse_args([
        "--model_config", _pythia_config(tmp_path),
        "--synthetic_data", "true",
        "--num_training_steps", "3",
        "--batch_size", "2", "--total_batch_size", "2",
        "--max_length", "32", "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "100", "--save_every", "100", "--workers", "0",
        "--save_dir", str(tmp_path / "run"),
    ])
    main(args)
    assert (tmp_path / "run" / "model_3").exists()


def test_regime_relora_magnitude_pruning_warm_start(tmp_path):
    # stage 1: short full-rank warmup checkpoint
    args = parse_args([
        "--model_config", "configs/llama_9m.json",
        "--synthetic_data", "true",
        "--num_training_steps", "2",
        "--batch_size", "2", "--total_batch_size", "2",
        "--max_length", "32", "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "100", "--save_every", "2", "--workers", "0",
        "--save_dir", str(tmp_path / "warm"),
    ])
    main(args)
    warm = tmp_path / "warm" / "model_2"
    assert warm.exists()
    # stage 2: relora from the warm checkpoint with magnitude pruning
    args = run_args(tmp_path, extra=[
        "--warmed_up_model", str(warm),
        "--reset_optimizer_on_relora", "false",
        "--optimizer_magnitude_pruning", "0.9",
    ], steps=8)
    main(args)
    state = json.load(open(tmp_path / "run" / "model_8" / "training_state.json"))
    assert state["n_optimizer_resets"] >= 1


def test_auto_batch_size(tmp_path):
    """--batch_size auto sizes the micro-batch for the device and trains."""
    args = parse_args([
        "--model_config", "configs/llama_9m.json",
        "--synthetic_data", "true",
        "--batch_size", "auto",
        "--total_batch_size", "8",
        "--num_training_steps", "2",
        "--max_length", "32", "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "100", "--save_every", "100", "--workers", "0",
        "--save_dir", str(tmp_path / "run"),
    ])
    assert args.batch_size == "auto"
    main(args)
    assert (tmp_path / "run" / "model_2").exists()


def test_auto_micro_batch_estimator():
    from relora_amd.models import load_model_config
    from relora_amd.utils.memory import auto_micro_batch, estimate_step_bytes

    cfg = load_model_config("configs/llama_1b.json")
    # llama-1b at seq 2048 should fit micro-batch >= 8 in 288 GB
    bs = auto_micro_batch(cfg, 2048, 1024, 8, lora_r=128,
                          trainable_ratio=0.08, hbm_bytes=288 << 30)
    assert bs >= 8, bs
    # monotone in micro-batch
    a = estimate_step_bytes(cfg, 4, 2048, lora_r=128)
    b = estimate_step_bytes(cfg, 8, 2048, lora_r=128)
    assert b > a
    # a tiny budget degrades gracefully to 1
    assert auto_micro_batch(cfg, 2048, 1024, 8, hbm_bytes=1 << 30) == 1


def test_training_config_yaml_recipes_parse(tmp_path):
    """The shipped training_configs/ recipes must parse through the YAML
    override path (reference args_utils.py:9-21 semantics)."""
    import yaml as _yaml

    for recipe in ("training_configs/1B_v1.0.yaml", "training_configs/250M_v1.0.yaml"):
        with open(recipe) as f:
            overrides = _yaml.safe_load(f)
        args = parse_args(["--training_config", recipe])
        assert args.use_peft is True
        assert args.lora_r == overrides["lora_r"]
        assert args.total_batch_size == overrides["total_batch_size"]
        assert float(args.lr) == float(overrides["lr"])
        assert args.num_training_steps == int(str(overrides["num_training_steps"]).replace("_", ""))


def test_deterministic_training(tmp_path):
    """Two runs with the same seed produce identical losses and weights
    (philox dropout, synthetic data, and init are all seed-keyed)."""
    import torch as _t

    losses = []
    for run in ("a", "b"):
        args = run_args(tmp_path / run, steps=6)
        main(args)
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()
        sd = _t.load(tmp_path / run / "run" / "model_6" / "pytorch_model.bin",
                     map_location="cpu", weights_only=True)
        losses.append(sd)
    a, b = losses
    assert set(a) == set(b)
    for k in a:
        assert _t.equal(a[k], b[k]), k


def test_profiler_flag(tmp_path, monkeypatch):
    """--profile true produces torch.profiler traces (reference
    maybe_make_profiler, torchrun_main.py:322-335)."""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    args = run_args(tmp_path, extra=["--profile", "true"])
    monkeypatch.chdir(tmp_path)  # after parse: config path was resolved in repo
    args.model_config = os.path.join(repo, "configs", "llama_9m.json")
    main(args)
    import glob as _g
    logs = _g.glob(str(tmp_path / "profiler_logs" / "*" / "*"))
    assert logs, "no profiler trace written"


def test_regime_quantized_relora(tmp_path):
    """--quantize 4bit end-to-end on CPU: frozen weights stored NF4, training
    steps run, merge requantizes, checkpoint saves (reference regime:
    relora.py 4-bit bnb path; ours is the HIP NF4 kernel / python oracle)."""
    args = run_args(tmp_path, extra=["--quantize", "4bit"])
    main(args)
    ckpt = tmp_path / "run" / "model_6"
    assert ckpt.exists()
    state = json.load(open(ckpt / "training_state.json"))
    assert state["update_step"] == 6
    assert state["n_lora_restarts"] >= 1


def test_args_max_train_tokens():
    """--max_train_tokens accepts 1M/1B suffixes and derives
    num_training_steps = tokens // total_batch_size (reference
    args_utils.py:49-51, training_utils.max_train_tokens_to_number)."""
    args = parse_args([
        "--synthetic_data", "true", "--batch_size", "2",
        "--total_batch_size", "4", "--max_train_tokens", "2M",
    ])
    assert args.max_train_tokens == 2_000_000
    assert args.num_training_steps == 2_000_000 // 4
    args = parse_args([
        "--synthetic_data", "true", "--batch_size", "2",
        "--total_batch_size", "4", "--max_train_tokens", "1B",
    ])
    assert args.max_train_tokens == 1_000_000_000


def test_regime_trainable_scaling(tmp_path):
    """--train_scaling (tanh-parameterized lora scale, reference
    relora.py trainable_scaling) through the full loop incl. merges."""
    args = run_args(tmp_path, extra=["--train_scaling"], steps=6)
    main(args)
    state = json.load(open(tmp_path / "run" / "model_6" / "training_state.json"))
    assert state["n_lora_restarts"] >= 1
    sd = torch.load(tmp_path / "run" / "model_6" / "pytorch_model.bin",
                    map_location="cpu", weights_only=True)
    assert any("scaling" in k for k in sd), list(sd)[:5]


def test_regime_quantized_int8(tmp_path):
    """--quantize 8bit end-to-end on CPU (int8 blockwise frozen weights)."""
    args = run_args(tmp_path, extra=["--quantize", "8bit"])
    main(args)
    state = json.load(open(tmp_path / "run" / "model_6" / "training_state.json"))
    assert state["update_step"] == 6 and state["n_lora_restarts"] >= 1


def test_regime_combo_quantized_zero_magprune_resume(tmp_path):
    """The heaviest interaction: NF4 frozen weights + ZeRO-1 + magnitude
    pruning + interrupt/autoresume, end-to-end (a combination matrix the
    reference never exercised)."""
    extra = ["--quantize", "4bit", "--optimizer", "adam_zero",
             "--reset_optimizer_on_relora", "False",
             "--optimizer_magnitude_pruning", "0.9",
             "--save_every", "3"]
    import torch.distributed as dist

    main(run_args(tmp_path, extra=extra, steps=3))
    if dist.is_initialized():
        dist.destroy_process_group()
    main(run_args(tmp_path, extra=extra + ["--autoresume", "true"], steps=6))
    state = json.load(open(tmp_path / "run" / "model_6" / "training_state.json"))
    assert state["update_step"] == 6
    assert state["n_optimizer_resets"] >= 1


def test_nan_consensus_batch_skip(tmp_path, monkeypatch):
    """NaN losses skip the optimizer update (C3 consensus path, reference
    torchrun_main.py:813-822) but training continues and completes."""
    from relora_amd.models import llama as llama_mod

    orig_forward = llama_mod.LlamaForCausalLM.forward
    calls = {"n": 0}

    def nan_every_third(self, *a, **k):
        out = orig_forward(self, *a, **k)
        if self.training and out.loss is not None:
            calls["n"] += 1
            if calls["n"] % 3 == 0:
                out.loss = out.loss * float("nan")
        return out

    monkeypatch.setattr(llama_mod.LlamaForCausalLM, "forward", nan_every_third)
    # clipping must be off: with error_if_nonfinite clipping, NaN grads crash
    # loudly BEFORE the consensus (reference semantics, torchrun_main.py:806)
    args = run_args(tmp_path, extra=["--clip_grad_norm", "0"], steps=6)
    main(args)
    # with 1/3 of batches NaN, the 5%-skipped abort fires (reference
    # torchrun_main.py:819-822) and the final save happens at the abort step
    ckpts = sorted((tmp_path / "run").glob("model_*"))
    assert ckpts, "no final checkpoint written"
    state = json.load(open(ckpts[-1] / "training_state.json"))
    assert state["update_step"] <= 6
    sd = torch.load(ckpts[-1] / "pytorch_model.bin",
                    map_location="cpu", weights_only=True)
    for k, v in sd.items():
        assert torch.isfinite(v).all(), k


def test_eval_nan_raises(tmp_path, monkeypatch):
    """A NaN during evaluation raises (reference evaluate_model,
    torchrun_main.py:176-178) rather than silently logging."""
    from relora_amd.models import llama as llama_mod

    orig_forward = llama_mod.LlamaForCausalLM.forward

    def nan_in_eval(self, *a, **k):
        out = orig_forward(self, *a, **k)
        if not self.training and out.loss is not None:
            out.loss = out.loss * float("nan")
        return out

    monkeypatch.setattr(llama_mod.LlamaForCausalLM, "forward", nan_in_eval)
    with pytest.raises(RuntimeError, match="nan"):
        main(run_args(tmp_path, steps=3))


def test_skip_batches_runtime(tmp_path):
    """--skip_batches skips the named update steps (no loss logged for
    them) while still reaching num_training_steps (reference
    torchrun_main.py:772-775)."""
    args = run_args(tmp_path, extra=["--skip_batches", "2"], steps=6)
    main(args)
    state = json.load(open(tmp_path / "run" / "model_6" / "training_state.json"))
    assert state["update_step"] == 6
    logged = [json.loads(l) for l in
              open(tmp_path / "run" / "wandb_offline.jsonl")]
    steps_with_loss = {r["update_step"] for r in logged if "loss" in r}
    # the skip matches the PRE-increment update_step (reference semantics:
    # "update_step numbers", torchrun_main.py:772), so skipping "2"
    # suppresses the update that would log as step 3
    assert 3 not in steps_with_loss
    assert {1, 2, 4, 5, 6} <= steps_with_loss


def test_mid_run_eval_fires(tmp_path):
    """--eval_every triggers mid-run evals, logged to the recorder
    (reference torchrun_main.py eval window)."""
    args = run_args(tmp_path, extra=["--eval_every", "3"], steps=6)
    main(args)
    logged = [json.loads(l) for l in
              open(tmp_path / "run" / "wandb_offline.jsonl")]
    evals = [r for r in logged if "final_eval_loss" in r]
    assert len(evals) >= 2  # at steps 3 and 6
