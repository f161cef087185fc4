"""Megatron-style data path: indexed storage, index maps, blending,
sharding, and the YAML->dataloader glue (all CPU)."""

import json
import os

import numpy as np
import pytest
import torch
import yaml

from relora_amd.data import indexed_dataset as idx_ds
from relora_amd.data.blendable import BlendableDataset, blending_indices_py
from relora_amd.data.gpt2_dataset import (
    GPT2Dataset, _num_epochs, build_sample_idx_py)
from relora_amd.data.megatron import (
    build_train_valid_test_dataloaders, get_train_valid_test_split_,
    get_normalized_weights_and_num_samples, weights_by_num_docs)
from relora_amd.data.neox_args import NeoXArgs
from relora_amd.data.samplers import DistributedBatchSampler


def _write_corpus(tmp_path, name="corpus", n_docs=37, vocab=1000, seed=0):
    """Build an mmap .bin/.idx pair of random docs; returns (prefix, docs)."""
    rng = np.random.RandomState(seed)
    prefix = str(tmp_path / name)
    builder = idx_ds.make_builder(idx_ds.data_file_path(prefix), "mmap", vocab_size=vocab)
    docs = []
    for _ in range(n_docs):
        doc = rng.randint(1, vocab, size=rng.randint(3, 60)).astype(np.int64)
        docs.append(doc)
        builder.add_item(torch.from_numpy(doc))
        builder.end_document()
    builder.finalize(idx_ds.index_file_path(prefix))
    return prefix, docs


def test_mmap_roundtrip(tmp_path):
    prefix, docs = _write_corpus(tmp_path)
    ds = idx_ds.make_dataset(prefix, "mmap")
    assert len(ds) == len(docs)
    assert ds.sizes.tolist() == [len(d) for d in docs]
    for i in (0, 5, len(docs) - 1):
        np.testing.assert_array_equal(np.asarray(ds[i], dtype=np.int64), docs[i])
    # offset/length reads
    np.testing.assert_array_equal(
        np.asarray(ds.get(3, offset=2, length=4), dtype=np.int64), docs[3][2:6])
    # dtype: vocab < 65500 -> uint16 on disk
    assert ds[0].dtype == np.uint16
    assert idx_ds.infer_dataset_impl(prefix) == "mmap"


def test_mmap_builder_merge(tmp_path):
    p1, docs1 = _write_corpus(tmp_path, "a", n_docs=5, seed=1)
    p2, docs2 = _write_corpus(tmp_path, "b", n_docs=7, seed=2)
    merged = str(tmp_path / "merged")
    b = idx_ds.MMapIndexedDatasetBuilder(idx_ds.data_file_path(merged), dtype=np.uint16)
    b.merge_file_(p1)
    b.merge_file_(p2)
    b.finalize(idx_ds.index_file_path(merged))
    ds = idx_ds.make_dataset(merged, "mmap")
    assert len(ds) == 12
    np.testing.assert_array_equal(np.asarray(ds[5], dtype=np.int64), docs2[0])


def test_legacy_roundtrip(tmp_path):
    prefix = str(tmp_path / "legacy")
    b = idx_ds.IndexedDatasetBuilder(idx_ds.data_file_path(prefix), dtype=np.int32)
    docs = [np.array([5, 6, 7], dtype=np.int32), np.array([9, 10], dtype=np.int32)]
    for d in docs:
        b.add_item(torch.from_numpy(d))
        b.end_document()
    b.finalize(idx_ds.index_file_path(prefix))
    assert idx_ds.infer_dataset_impl(prefix) == "cached"
    ds = idx_ds.make_dataset(prefix, "cached")
    np.testing.assert_array_equal(ds[1], docs[1])
    np.testing.assert_array_equal(ds.get(0, offset=1, length=2), docs[0][1:3])


def test_sample_idx_cpp_matches_python_oracle():
    helpers = pytest.importorskip("relora_amd.data._index_helpers")
    rng = np.random.RandomState(3)
    for trial in range(5):
        n_docs = rng.randint(2, 30)
        sizes = rng.randint(1, 50, size=n_docs).astype(np.int32)
        num_epochs = rng.randint(1, 4)
        doc_idx = np.tile(np.arange(n_docs, dtype=np.int32), num_epochs)
        rng.shuffle(doc_idx)
        seq_length = int(rng.randint(2, 17))
        tokens_per_epoch = int(sizes.sum())
        ours = helpers.build_sample_idx_int32(
            sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
        oracle = build_sample_idx_py(
            sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
        np.testing.assert_array_equal(ours, oracle)
        ours64 = helpers.build_sample_idx_int64(
            sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
        np.testing.assert_array_equal(ours64, oracle)


def test_blending_indices_cpp_matches_python_oracle():
    helpers = pytest.importorskip("relora_amd.data._index_helpers")
    w = np.array([0.6, 0.25, 0.15], dtype=np.float64)
    size = 1000
    di = np.zeros(size, dtype=np.uint8)
    dsi = np.zeros(size, dtype=np.int64)
    helpers.build_blending_indices(di, dsi, w, 3, size, False)
    di2, dsi2 = blending_indices_py(w, size)
    np.testing.assert_array_equal(di, di2)
    np.testing.assert_array_equal(dsi, dsi2)
    # achieved ratios track the weights
    counts = np.bincount(di, minlength=3) / size
    np.testing.assert_allclose(counts, w, atol=0.01)


def test_build_mapping_exports_exist():
    helpers = pytest.importorskip("relora_amd.data._index_helpers")
    docs = np.array([0, 3, 6], dtype=np.int64)       # 2 docs, 3 sentences each
    sizes = np.array([10, 20, 30, 10, 20, 30], dtype=np.int32)
    m = helpers.build_mapping(docs, sizes, 2, 100, 64, 0.1, 1234, False)
    assert m.ndim == 2 and m.shape[1] == 3 and m.shape[0] > 0
    # deterministic in the seed
    m2 = helpers.build_mapping(docs, sizes, 2, 100, 64, 0.1, 1234, False)
    np.testing.assert_array_equal(m, m2)
    titles = np.array([2, 2], dtype=np.int32)
    b = helpers.build_blocks_mapping(docs, sizes, titles, 2, 100, 64, 1234, False, False)
    assert b.ndim == 2 and b.shape[1] == 4 and b.shape[0] > 0


def test_gpt2_dataset_windows_and_cache(tmp_path):
    prefix, docs = _write_corpus(tmp_path, n_docs=25, seed=4)
    indexed = idx_ds.make_dataset(prefix, "mmap")
    documents = np.arange(len(docs), dtype=np.int32)
    seq_len = 16
    ds = GPT2Dataset("train", prefix, documents, indexed,
                     num_samples=40, seq_length=seq_len, seed=7)
    flat_reference = {}
    assert len(ds) >= 40
    sample = ds[0]
    assert sample["input_ids"].shape == (seq_len + 1,)
    assert sample["input_ids"].dtype == np.int64

    # every sample window must be a contiguous run of the shuffled doc stream
    stream = np.concatenate([docs[d] for d in np.asarray(ds.doc_idx)])
    for i in range(0, 40, 7):
        idx = ds.shuffle_idx[i]
        start = idx * seq_len
        expect = stream[start:start + seq_len + 1]
        np.testing.assert_array_equal(ds[i]["input_ids"], expect, err_msg=f"sample {i}")
    del flat_reference

    # second construction must reuse the cached .npy maps and agree exactly
    ds2 = GPT2Dataset("train", prefix, documents, indexed,
                      num_samples=40, seq_length=seq_len, seed=7)
    np.testing.assert_array_equal(ds[3]["input_ids"], ds2[3]["input_ids"])
    cache_files = [f for f in os.listdir(tmp_path) if f.endswith(".npy")]
    assert len(cache_files) == 3


def test_num_epochs_minimality():
    # smallest epoch count such that (E*tokens-1)//seq >= samples
    assert _num_epochs(100, 10, 9) == 1
    assert _num_epochs(100, 10, 10) == 2   # (100-1)//10 = 9 < 10
    assert _num_epochs(101, 10, 10) == 1


def test_blendable_dataset(tmp_path):
    pa, _ = _write_corpus(tmp_path, "pa", n_docs=10, seed=5)
    pb, _ = _write_corpus(tmp_path, "pb", n_docs=10, seed=6)
    sets = []
    for prefix in (pa, pb):
        indexed = idx_ds.make_dataset(prefix, "mmap")
        documents = np.arange(10, dtype=np.int32)
        sets.append(GPT2Dataset(os.path.basename(prefix), prefix, documents,
                                indexed, num_samples=20, seq_length=8, seed=1))
    blend = BlendableDataset(sets, [0.7, 0.3])
    assert len(blend) == len(sets[0]) + len(sets[1])
    s = blend[0]
    assert s["input_ids"].shape == (9,)


def test_distributed_batch_sampler_sharding():
    sampler = list(range(32))
    shards = []
    for rank in range(4):
        bs = DistributedBatchSampler(
            torch.utils.data.SequentialSampler(sampler), batch_size=8,
            drop_last=True, rank=rank, world_size=4)
        shards.append([b for b in bs])
    # each global batch of 8 splits into 4 contiguous shards of 2
    for gb in range(4):
        got = sum((shards[r][gb] for r in range(4)), [])
        assert got == sampler[gb * 8:(gb + 1) * 8]

    # start_iter fast-forward skips whole global batches
    bs = DistributedBatchSampler(torch.utils.data.SequentialSampler(sampler),
                                 batch_size=8, drop_last=True,
                                 rank=0, world_size=4)
    bs.start_iter = 2
    assert [b for b in bs] == [[16, 17], [24, 25]]


def test_neox_args_batch_solver():
    t, m, g = NeoXArgs.calculate_batch_parameters(8, train_batch=1024, micro_batch=8)
    assert (t, m, g) == (1024, 8, 16)
    t, m, g = NeoXArgs.calculate_batch_parameters(8, micro_batch=4, grad_acc=2)
    assert (t, m, g) == (64, 4, 2)
    with pytest.raises(ValueError):
        NeoXArgs.calculate_batch_parameters(8)


def test_split_string():
    b = get_train_valid_test_split_("969, 30, 1", 1000)
    assert b == [0, 969, 999, 1000]
    b = get_train_valid_test_split_("8/1/1", 100)
    assert b[-1] == 100 and len(b) == 4


def test_weights_by_num_docs():
    w = weights_by_num_docs([1000, 10], alpha=0.3)
    assert len(w) == 2 and abs(sum(w) - 1) < 1e-9
    # α<1 boosts the small corpus above its natural share
    assert w[1] > 10 / 1010
    assert weights_by_num_docs([42]) == [1.0]
    w, n = get_normalized_weights_and_num_samples([2.0, 2.0], 100)
    assert w == [0.5, 0.5] and n == [51, 51]  # 0.5% headroom, ceil


def test_build_dataloaders_end_to_end(tmp_path):
    """YAML -> NeoXArgs -> three loaders; batch shapes; resume start_iter."""
    prefix, _ = _write_corpus(tmp_path, n_docs=60, vocab=500, seed=8)
    cfg = {
        "train_data_paths": [prefix],
        "valid_data_paths": [prefix],
        "test_data_paths": [prefix],
        "data_impl": "mmap",
        "seq_length": 16,
        "train_iters": 20,
        "eval_interval": 10,
        "eval_iters": 2,
        "num_workers": 0,
        "global_num_gpus": 1,
        "train_micro_batch_size_per_gpu": 4,
        "gradient_accumulation_steps": 1,
        "train_batch_size": 4,
        "pipe_parallel_size": 0,
        "model_parallel_size": 1,
    }
    neox_args = NeoXArgs.from_dict(cfg)
    train_loader, valid_loader, test_loader = build_train_valid_test_dataloaders(neox_args)
    assert neox_args.do_train and neox_args.do_valid and neox_args.do_test
    batch = next(iter(train_loader))
    assert batch["input_ids"].shape == (4, 17)
    assert batch["input_ids"].dtype == torch.int64

    # resume: iteration=3 -> start_iter 3, first batch equals 4th batch of a fresh run
    fresh = [b["input_ids"] for _, b in zip(range(5), iter(train_loader))]
    cfg2 = dict(cfg, iteration=3)
    neox2 = NeoXArgs.from_dict(cfg2)
    t2, _, _ = build_train_valid_test_dataloaders(neox2)
    assert t2.batch_sampler.start_iter == 3
    resumed = next(iter(t2))
    torch.testing.assert_close(resumed["input_ids"], fresh[3])


def test_load_megatron_dataset_glue(tmp_path):
    import argparse

    from relora_amd.data.megatron import load_megatron_dataset

    prefix, _ = _write_corpus(tmp_path, n_docs=40, vocab=500, seed=9)
    yaml_path = tmp_path / "ds.yaml"
    with open(yaml_path, "w") as f:
        yaml.safe_dump({
            "train_data_paths": [prefix],
            "valid_data_paths": [prefix],
            "test_data_paths": [prefix],
            "data_impl": "mmap",
            "seq_length": 16,
            "train_iters": 10,
            "eval_interval": 5,
            "eval_iters": 1,
        }, f)
    args = argparse.Namespace(
        megatron_dataset_config=str(yaml_path), batch_size=2,
        gradient_accumulation=1, total_batch_size=2, workers=0,
        max_length=16, num_training_steps=5)
    train_loader, valid_loader, test_loader, tok = load_megatron_dataset(
        args, world_size=1, start_iteration=0)
    b = next(iter(train_loader))
    assert b["input_ids"].shape == (2, 17)
    assert valid_loader is not None and test_loader is not None

    # num_training_steps > train_iters must be rejected
    args.num_training_steps = 11
    with pytest.raises(ValueError):
        load_megatron_dataset(args, world_size=1, start_iteration=0)


def test_args_json_sidecar_roundtrip(tmp_path):
    # sanity: yaml safe_dump/load of the injected fields keeps types
    p = tmp_path / "c.yaml"
    with open(p, "w") as f:
        yaml.safe_dump({"seq_length": 2048, "train_iters": 100}, f)
    with open(p) as f:
        d = yaml.safe_load(f)
    assert json.loads(json.dumps(d)) == d


def test_tokenize_and_chunk_offline(tmp_path):
    """HF-path preprocessing (dataloader.py tokenize_and_chunk) with an
    offline word-level tokenizer: concat + chunk to block_size, EOS joined,
    attention_mask dropped (reference dataloader.py:57-124)."""
    import datasets as hfds
    from tokenizers import Tokenizer as Tk, models as tkm, pre_tokenizers as tkp
    from transformers import PreTrainedTokenizerFast

    from relora_amd.data.dataloader import tokenize_and_chunk

    vocab = {"[PAD]": 0, "[UNK]": 1, "</s>": 2}
    for w in ("aa", "bb", "cc", "dd", "ee"):
        vocab[w] = len(vocab)
    tk = Tk(tkm.WordLevel(vocab=vocab, unk_token="[UNK]"))
    tk.pre_tokenizer = tkp.Whitespace()
    tok = PreTrainedTokenizerFast(tokenizer_object=tk, pad_token="[PAD]",
                                  unk_token="[UNK]", eos_token="</s>")

    ds = hfds.DatasetDict({"train": hfds.Dataset.from_dict(
        {"text": ["aa bb cc", "dd ee", "aa aa aa aa aa"]})})
    out = tokenize_and_chunk(tok, ds, text_field="text", sequence_length=4,
                             num_cpu=1)["train"]
    assert out.column_names == ["input_ids"]
    rows = [r["input_ids"] for r in out]
    assert all(len(r) == 4 for r in rows)
    # total tokens = sum(doc tokens + 1 eos each), floored to blocks of 4
    total = sum(len(tok(t)["input_ids"]) + 1 for t in ["aa bb cc", "dd ee", "aa aa aa aa aa"])
    assert len(rows) == total // 4
    flat = [t for r in rows for t in r]
    assert tok.eos_token_id in flat


def test_neox_args_type_validation():
    """Consumed fields are type-checked (reference arguments.py:109-1240
    equivalent for the surface this trainer reads)."""
    import pytest
    from relora_amd.data.neox_args import NeoXArgs

    base = {"global_num_gpus": 2, "train_micro_batch_size_per_gpu": 4,
            "data_path": "x", "train_iters": 10}
    NeoXArgs.from_dict(dict(base))
    with pytest.raises(ValueError, match="seq_length"):
        NeoXArgs.from_dict(dict(base, seq_length="2048"))
    with pytest.raises(ValueError, match="train_data_paths"):
        NeoXArgs.from_dict(dict(base, train_data_paths="not-a-list"))
    with pytest.raises(ValueError, match="mmap_warmup"):
        NeoXArgs.from_dict(dict(base, mmap_warmup=3))
    # float accepts int
    a = NeoXArgs.from_dict(dict(base, weighted_sampler_alpha=1))
    assert a.weighted_sampler_alpha == 1.0


def test_neox_args_strict_unknown_keys():
    import pytest
    from relora_amd.data.neox_args import NeoXArgs

    base = {"global_num_gpus": 1, "train_micro_batch_size_per_gpu": 2,
            "data_path": "x", "attention_dropout": 0.1}
    a = NeoXArgs.from_dict(dict(base))  # non-strict: kept + warned
    assert a.extra_args == {"attention_dropout": 0.1}
    with pytest.raises(ValueError, match="unknown config keys"):
        NeoXArgs.from_dict(dict(base), strict=True)
