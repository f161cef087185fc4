"""Megatron-style train/valid/test dataloader assembly.

Equivalent of the reference's
peft_pretraining/megatron_dataset/data_utils.py:308-467: sizes the three
splits from the iteration schedule, builds weighted multi-corpus
(BlendableDataset) or split-string single-corpus GPT2Datasets, wraps them in
``DistributedBatchSampler`` loaders, and fast-forwards ``start_iter`` on
resume.  No runtime ``make`` step: the C++ helpers are part of the package
build (relora_amd/data/csrc/index_helpers.cpp).
"""

import math
from functools import partial
from itertools import zip_longest

import torch.distributed as dist
import torch.utils.data

from relora_amd.data.blendable import BlendableDataset
from relora_amd.data.gpt2_dataset import GPT2Dataset
from relora_amd.data.indexed_dataset import make_dataset as make_indexed_dataset
from relora_amd.data.samplers import DistributedBatchSampler
from relora_amd.utils.logging import logger


def make_data_loader(dataset, neox_args):
    if dataset is None:
        return None
    world_size, rank = 1, 0
    if dist.is_initialized():
        world_size = dist.get_world_size()
        rank = dist.get_rank()

    sampler = torch.utils.data.SequentialSampler(dataset)
    batch_sampler = DistributedBatchSampler(
        sampler=sampler,
        batch_size=neox_args.batch_size * world_size,
        drop_last=True,
        rank=rank,
        world_size=world_size,
    )
    # dedicated generator: DataLoader.__iter__ otherwise draws its _base_seed
    # from the global RNG, which would offset the global stream by when the
    # iterator is created and break bit-exact resume
    return torch.utils.data.DataLoader(
        dataset, batch_sampler=batch_sampler,
        num_workers=neox_args.num_workers, pin_memory=True,
        generator=torch.Generator().manual_seed(int(neox_args.seed) * 100003 + 3))


def build_the_dataset(data_prefix, name, data_impl, num_samples, seq_length,
                      seed, skip_warmup, build_index_mappings=True,
                      label_prefix=None):
    indexed = make_indexed_dataset(data_prefix, data_impl, skip_warmup)
    label_ds = (make_indexed_dataset(label_prefix, data_impl, skip_warmup)
                if label_prefix else None)
    import numpy as np
    documents = np.arange(indexed.sizes.shape[0], dtype=np.int32)
    return GPT2Dataset(name, data_prefix, documents, indexed, num_samples,
                       seq_length, seed, build_index_mappings=build_index_mappings,
                       label_dataset=label_ds)


def get_train_valid_test_split_(splits_string, size):
    """'969, 30, 1' (or '/'-separated) -> 4 cumulative document boundaries."""
    if "," in splits_string:
        parts = [float(s) for s in splits_string.split(",")]
    elif "/" in splits_string:
        parts = [float(s) for s in splits_string.split("/")]
    else:
        parts = [float(splits_string)]
    parts = (parts + [0.0, 0.0, 0.0])[:3]
    total = sum(parts)
    if total <= 0:
        raise ValueError(f"bad split string: {splits_string!r}")
    parts = [p / total for p in parts]
    bounds = [0]
    for p in parts:
        bounds.append(bounds[-1] + int(round(p * float(size))))
    diff = bounds[-1] - size
    for i in range(1, len(bounds)):
        bounds[i] -= diff
    assert bounds[-1] == size
    return bounds


def build_train_valid_test_datasets(data_prefix, use_shared_fs, data_impl,
                                    splits_string, train_valid_test_num_samples,
                                    seq_length, seed, skip_warmup):
    import numpy as np
    indexed = make_indexed_dataset(data_prefix, data_impl, skip_warmup)
    splits = get_train_valid_test_split_(splits_string, indexed.sizes.shape[0])

    def build(index, name):
        if splits[index + 1] <= splits[index]:
            return None
        documents = np.arange(splits[index], splits[index + 1], dtype=np.int32)
        return GPT2Dataset(name, data_prefix, documents, indexed,
                           train_valid_test_num_samples[index], seq_length,
                           seed, use_shared_fs=use_shared_fs)

    return build(0, "train"), build(1, "valid"), build(2, "test")


def get_normalized_weights_and_num_samples(weights, num_samples):
    total = sum(weights)
    if total <= 0:
        raise ValueError("weights must sum to > 0")
    weights = [w / total for w in weights]
    # 0.5% headroom so uneven blending never starves a dataset
    return weights, [int(math.ceil(num_samples * w * 1.005)) for w in weights]


def weights_by_num_docs(n_docs, alpha=0.3):
    """α-sampling (arXiv 1911.02116): p(L) ∝ |L|^α upsamples small corpora."""
    if len(n_docs) == 1:
        return [1.0]
    total = sum(n_docs)
    probs = [n / total for n in n_docs]
    boosted = [p ** alpha for p in probs]
    s = sum(boosted)
    boosted = [b / s for b in boosted]
    weights = [b * (1 - p) for b, p in zip(boosted, probs)]
    s = sum(weights)
    return [w / s for w in weights]


def build_weighted_datasets(neox_args, train_num_samples, valid_num_samples,
                            test_num_samples, build_index_mappings=True):
    train, valid, test = [], [], []
    for i, (tr, lab, va, te) in enumerate(zip_longest(
            neox_args.train_data_paths,
            neox_args.label_data_paths or [],
            neox_args.valid_data_paths or [],
            neox_args.test_data_paths or [])):
        common = dict(data_impl=neox_args.data_impl, seq_length=neox_args.seq_length,
                      seed=neox_args.seed, skip_warmup=not neox_args.mmap_warmup,
                      build_index_mappings=build_index_mappings)
        if tr:
            train.append(build_the_dataset(tr, f"train_{i}", num_samples=train_num_samples[i],
                                           label_prefix=lab, **common))
        if va:
            valid.append(build_the_dataset(va, f"valid_{i}", num_samples=valid_num_samples[i],
                                           **common))
        if te:
            test.append(build_the_dataset(te, f"test_{i}", num_samples=test_num_samples[i],
                                          **common))
    return train, valid, test


def build_train_valid_test_dataloaders(neox_args):
    if neox_args.is_pipe_parallel:
        raise ValueError("pipeline parallelism is not part of the ReLoRA data path")

    train_iters = neox_args.train_iters
    eval_iters = (train_iters // neox_args.eval_interval + 1) * neox_args.eval_iters
    test_iters = neox_args.eval_iters
    num_samples = [train_iters * neox_args.train_batch_size,
                   eval_iters * neox_args.train_batch_size,
                   test_iters * neox_args.train_batch_size]

    if neox_args.train_data_paths:
        train_w, train_n = get_normalized_weights_and_num_samples(
            neox_args.train_data_weights, num_samples[0])
        valid_w, valid_n = get_normalized_weights_and_num_samples(
            neox_args.valid_data_weights or [1.0], num_samples[1])
        test_w, test_n = get_normalized_weights_and_num_samples(
            neox_args.test_data_weights or [1.0], num_samples[2])

        train_ds_list, valid_ds_list, test_ds_list = build_weighted_datasets(
            neox_args, train_n, valid_n, test_n,
            build_index_mappings=not neox_args.weight_by_num_documents)

        if neox_args.weight_by_num_documents:
            fn = partial(weights_by_num_docs, alpha=neox_args.weighted_sampler_alpha)
            counts = lambda dsl: [d.indexed_dataset.sizes.shape[0] for d in dsl]  # noqa: E731
            train_w, train_n = get_normalized_weights_and_num_samples(
                fn(counts(train_ds_list)), num_samples[0])
            valid_w, valid_n = get_normalized_weights_and_num_samples(
                fn(counts(valid_ds_list)), num_samples[1])
            test_w, test_n = get_normalized_weights_and_num_samples(
                fn(counts(test_ds_list)), num_samples[2])
            train_ds_list, valid_ds_list, test_ds_list = build_weighted_datasets(
                neox_args, train_n, valid_n, test_n)

        train_ds = BlendableDataset(train_ds_list, train_w) if train_ds_list else None
        valid_ds = BlendableDataset(valid_ds_list, valid_w) if valid_ds_list else None
        test_ds = BlendableDataset(test_ds_list, test_w) if test_ds_list else None
    else:
        train_ds, valid_ds, test_ds = build_train_valid_test_datasets(
            data_prefix=neox_args.data_path,
            use_shared_fs=neox_args.use_shared_fs,
            data_impl=neox_args.data_impl,
            splits_string=neox_args.split,
            train_valid_test_num_samples=num_samples,
            seq_length=neox_args.seq_length,
            seed=neox_args.seed,
            skip_warmup=not neox_args.mmap_warmup)

    train_loader = make_data_loader(train_ds, neox_args)
    valid_loader = make_data_loader(valid_ds, neox_args)
    test_loader = make_data_loader(test_ds, neox_args)

    neox_args.do_train = int(train_loader is not None and neox_args.train_iters > 0)
    neox_args.do_valid = int(valid_loader is not None and neox_args.eval_iters > 0)
    neox_args.do_test = int(test_loader is not None and neox_args.eval_iters > 0)

    iteration = neox_args.iteration or 0
    if train_loader is not None:
        train_loader.batch_sampler.start_iter = (
            iteration * neox_args.gradient_accumulation_steps) % len(train_loader)
        logger.info(f"train data start_iter = {train_loader.batch_sampler.start_iter}")
    if valid_loader is not None:
        start_val = ((iteration * neox_args.gradient_accumulation_steps)
                     // neox_args.eval_interval) * neox_args.eval_iters
        valid_loader.batch_sampler.start_iter = start_val % len(valid_loader)
        logger.info(f"valid data start_iter = {valid_loader.batch_sampler.start_iter}")

    return train_loader, valid_loader, test_loader


def load_megatron_dataset(args, world_size, start_iteration):
    """Trainer glue: YAML config -> NeoXArgs -> the three dataloaders
    (reference torchrun_main.py:276-319).  Injects the trainer's world/batch
    geometry into the config before solving derived values."""
    import yaml

    from relora_amd.data.neox_args import NeoXArgs

    logger.info(f"Loading Megatron dataset config from {args.megatron_dataset_config}")
    with open(args.megatron_dataset_config) as f:
        cfg = yaml.safe_load(f)

    cfg["global_num_gpus"] = world_size
    cfg["train_micro_batch_size_per_gpu"] = args.batch_size
    cfg["gradient_accumulation_steps"] = args.gradient_accumulation
    cfg["train_batch_size"] = args.total_batch_size
    cfg["num_workers"] = args.workers

    if args.max_length != cfg["seq_length"]:
        logger.warning(f"max_length ({args.max_length}) != config seq_length "
                       f"({cfg['seq_length']}); using seq_length")
        args.max_length = cfg["seq_length"]

    if args.num_training_steps > cfg["train_iters"]:
        raise ValueError(f"num_training_steps ({args.num_training_steps}) exceeds "
                         f"train_iters ({cfg['train_iters']})")

    tokenizer = None
    if cfg.get("vocab_file"):
        try:
            from tokenizers import Tokenizer
            tokenizer = Tokenizer.from_file(cfg["vocab_file"])
            tokenizer.name_or_path = cfg["vocab_file"]
        except Exception as e:  # offline / missing vocab file
            logger.warning(f"could not load tokenizer from {cfg['vocab_file']}: {e}")

    neox_args = NeoXArgs.from_dict(cfg)
    if neox_args.iteration is None:
        neox_args.iteration = start_iteration
    if neox_args.train_batch_size != args.total_batch_size:
        raise ValueError(
            f"megatron train_batch_size ({neox_args.train_batch_size}) must equal "
            f"total_batch_size ({args.total_batch_size})")

    train_loader, valid_loader, test_loader = build_train_valid_test_dataloaders(neox_args)
    logger.info("Megatron dataset built")
    return train_loader, valid_loader, test_loader, tokenizer
