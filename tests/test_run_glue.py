"""run_glue.py end-to-end on CPU: tiny model, offline tokenizer, csv task."""

import csv
import json
import sys

import pytest
import torch  # noqa: F401


@pytest.fixture
def tiny_setup(tmp_path):
    # offline word-level tokenizer
    from tokenizers import Tokenizer, models, pre_tokenizers
    from transformers import PreTrainedTokenizerFast

    words = ["good", "bad", "great", "awful", "fine", "terrible", "movie", "film"]
    vocab = {"[PAD]": 0, "[UNK]": 1, "[EOS]": 2}
    for w in words:
        vocab[w] = len(vocab)
    tok = Tokenizer(models.WordLevel(vocab=vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    fast = PreTrainedTokenizerFast(tokenizer_object=tok, pad_token="[PAD]",
                                   unk_token="[UNK]", eos_token="[EOS]")
    tok_dir = tmp_path / "tok"
    fast.save_pretrained(str(tok_dir))

    # tiny architecture json
    cfg = {
        "architectures": ["LlamaForCausalLM"], "hidden_size": 32,
        "intermediate_size": 64, "num_attention_heads": 4, "num_hidden_layers": 2,
        "vocab_size": len(vocab), "max_position_embeddings": 64,
        "max_sequence_length": 64, "rms_norm_eps": 1e-6, "bos_token_id": 2,
        "eos_token_id": 2, "pad_token_id": 0, "model_type": "llama",
        "hidden_act": "silu", "initializer_range": 0.02,
    }
    cfg_path = tmp_path / "tiny.json"
    cfg_path.write_text(json.dumps(cfg))

    # csv train/validation: single-sentence binary classification
    rng_rows = [("good movie", 1), ("great film", 1), ("fine movie", 1),
                ("bad movie", 0), ("awful film", 0), ("terrible movie", 0)] * 4
    for split in ("train", "validation"):
        with open(tmp_path / f"{split}.csv", "w", newline="") as f:
            w = csv.writer(f)
            w.writerow(["sentence", "label"])
            w.writerows(rng_rows)
    return tmp_path, tok_dir, cfg_path


def test_run_glue_end_to_end(tiny_setup, monkeypatch):
    tmp_path, tok_dir, cfg_path = tiny_setup
    out_dir = tmp_path / "out"
    argv = [
        "run_glue.py",
        "--model_config", str(cfg_path),
        "--tokenizer_name", str(tok_dir),
        "--train_file", str(tmp_path / "train.csv"),
        "--validation_file", str(tmp_path / "validation.csv"),
        "--output_dir", str(out_dir),
        "--do_train", "--do_eval",
        "--max_seq_length", "16",
        "--per_device_train_batch_size", "4",
        "--max_steps", "3",
        "--learning_rate", "1e-3",
        "--report_to", "none",
        "--use_cpu", "True",
    ]
    monkeypatch.setattr(sys, "argv", argv)
    import run_glue

    run_glue.main()
    metrics = json.loads((out_dir / "eval_results.json").read_text())
    assert "eval_accuracy" in metrics
    assert 0.0 <= metrics["eval_accuracy"] <= 1.0


def test_glue_metrics_helpers():
    import numpy as np

    from run_glue import glue_metrics

    preds = np.array([1, 0, 1, 1])
    labels = np.array([1, 0, 0, 1])
    m = glue_metrics("mrpc", preds, labels, is_regression=False)
    assert set(m) == {"accuracy", "f1", "combined_score"}
    assert m["accuracy"] == 0.75
    m = glue_metrics("cola", preds, labels, is_regression=False)
    assert "matthews_correlation" in m
    m = glue_metrics("stsb", np.array([0.1, 0.5, 0.9]), np.array([0.2, 0.4, 1.0]),
                     is_regression=True)
    assert m["pearson"] > 0.9
