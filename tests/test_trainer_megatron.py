"""The full trainer driving the Megatron data path end-to-end on CPU:
mmap corpus -> NeoXArgs yaml -> GPT2Dataset -> train loop with ReLoRA."""

import json

import numpy as np
import pytest
import torch
import yaml

from relora_amd.data import indexed_dataset as idx_ds
from relora_amd.trainer import main, parse_args


@pytest.fixture(autouse=True)
def _clean_dist(monkeypatch):
    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE"):
        monkeypatch.delenv(k, raising=False)
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", "29741")
    yield
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


def test_trainer_with_megatron_dataset(tmp_path):
    # corpus: 200 random docs in an mmap .bin/.idx pair
    rng = np.random.RandomState(0)
    prefix = str(tmp_path / "corpus")
    builder = idx_ds.make_builder(idx_ds.data_file_path(prefix), "mmap", vocab_size=32000)
    for _ in range(200):
        doc = rng.randint(1, 31999, size=rng.randint(8, 80)).astype(np.int64)
        builder.add_item(torch.from_numpy(doc))
        builder.end_document()
    builder.finalize(idx_ds.index_file_path(prefix))

    cfg_yaml = tmp_path / "megatron.yaml"
    with open(cfg_yaml, "w") as f:
        yaml.safe_dump({
            "train_data_paths": [prefix],
            "valid_data_paths": [prefix],
            "test_data_paths": [prefix],
            "data_impl": "mmap",
            "seq_length": 32,
            "train_iters": 64,
            "eval_interval": 32,
            "eval_iters": 2,
        }, f)

    args = parse_args([
        "--model_config", "configs/llama_9m.json",
        "--megatron_dataset_config", str(cfg_yaml),
        "--use_peft", "true",
        "--relora", "3", "--cycle_length", "3",
        "--restart_warmup_steps", "1", "--warmup_steps", "2",
        "--scheduler", "cosine_restarts",
        "--num_training_steps", "6",
        "--batch_size", "2", "--total_batch_size", "2",
        "--max_length", "32",
        "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "3", "--save_every", "100",
        "--workers", "0",
        "--save_dir", str(tmp_path / "run"),
    ])
    main(args)
    state = json.load(open(tmp_path / "run" / "model_6" / "training_state.json"))
    assert state["update_step"] == 6
    assert state["n_lora_restarts"] >= 1
    # index-map caches were produced next to the corpus
    caches = list(tmp_path.glob("corpus_train_0_indexmap_*.npy"))
    assert len(caches) == 3
