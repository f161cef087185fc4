"""pretokenize.py offline end-to-end: local dataset + local tokenizer ->
on-disk tokenized layout -> consumed by the trainer via --dataset_path."""

import json
import os

import pytest
import torch  # noqa: F401


@pytest.fixture
def local_corpus(tmp_path):
    import datasets as hfds
    from tokenizers import Tokenizer, models, pre_tokenizers
    from transformers import PreTrainedTokenizerFast

    words = [f"w{i}" for i in range(60)]
    vocab = {"[PAD]": 0, "[UNK]": 1, "</s>": 2}
    for w in words:
        vocab[w] = len(vocab)
    tok = Tokenizer(models.WordLevel(vocab=vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    fast = PreTrainedTokenizerFast(tokenizer_object=tok, pad_token="[PAD]",
                                   unk_token="[UNK]", eos_token="</s>")
    tok_dir = tmp_path / "tok"
    fast.save_pretrained(str(tok_dir))

    import numpy as np
    rng = np.random.RandomState(0)
    texts = [" ".join(rng.choice(words, size=rng.randint(5, 40)))
             for _ in range(200)]
    ds = hfds.DatasetDict({
        "train": hfds.Dataset.from_dict({"text": texts[:180]}),
        "validation": hfds.Dataset.from_dict({"text": texts[180:]}),
    })
    ds_dir = tmp_path / "rawds"
    ds.save_to_disk(str(ds_dir))
    return tmp_path, str(tok_dir), str(ds_dir)


def test_pretokenize_then_train(local_corpus, monkeypatch):
    tmp_path, tok_dir, ds_dir = local_corpus
    import pretokenize

    args = pretokenize.parse_args([
        "--tokenizer", tok_dir,
        "--dataset", ds_dir,
        "--sequence_length", "32",
        "--num_cpu", "1",
        "--save_dir", str(tmp_path / "pretok"),
    ])
    pretokenize.main(args)
    out_dirs = list((tmp_path / "pretok").iterdir())
    assert len(out_dirs) == 1
    out = out_dirs[0]
    assert (out / "args.json").exists()
    meta = json.loads((out / "args.json").read_text())
    assert meta["sequence_length"] == 32

    # the trainer consumes it through --dataset_path
    from relora_amd.trainer import main as train_main, parse_args as train_args

    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE"):
        monkeypatch.delenv(k, raising=False)
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", "29751")
    # model vocab must match the tokenizer vocab (trainer enforces it,
    # reference torchrun_main.py:481-486)
    tiny_cfg = {
        "architectures": ["LlamaForCausalLM"], "model_type": "llama",
        "hidden_size": 32, "intermediate_size": 64, "num_attention_heads": 4,
        "num_hidden_layers": 2, "max_position_embeddings": 64,
        "max_sequence_length": 64, "rms_norm_eps": 1e-6, "hidden_act": "silu",
        "initializer_range": 0.02, "vocab_size": 63,
    }
    cfg_path = tmp_path / "tiny63.json"
    cfg_path.write_text(json.dumps(tiny_cfg))
    targs = train_args([
        "--model_config", str(cfg_path),
        "--dataset_path", str(out),
        "--num_training_steps", "3",
        "--batch_size", "2", "--total_batch_size", "2",
        "--max_length", "32", "--lr", "1e-3", "--dtype", "float32",
        "--eval_every", "100", "--save_every", "100", "--workers", "0",
        "--save_dir", str(tmp_path / "run"),
    ])
    train_main(targs)
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()
    assert (tmp_path / "run" / "model_3" / "pytorch_model.bin").exists()
