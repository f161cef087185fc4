"""GPT2-style pretraining dataset over an indexed token store.

A sample is a ``seq_length + 1`` token window read from the shuffled
concatenation of documents, addressed through three cached numpy index maps
(doc_idx / sample_idx / shuffle_idx) built once by rank 0 and mmap-loaded by
every rank.  Behavior- and cache-file-compatible with the reference
(peft_pretraining/megatron_dataset/dataset.py): same ``.npy`` file naming,
same ``np.random.RandomState(seed)`` stream, same sample addressing, so an
existing cache directory is reused as-is.
"""

import os
import time

import numpy as np
import torch
import torch.distributed as dist
import torch.utils.data

from relora_amd.utils.logging import logger


class GPT2Dataset(torch.utils.data.Dataset):
    def __init__(self, name, data_prefix, documents, indexed_dataset,
                 num_samples, seq_length, seed, build_index_mappings=True,
                 use_shared_fs=True, label_dataset=None):
        self.name = name
        self.indexed_dataset = indexed_dataset
        self.label_dataset = label_dataset

        if len(documents) == 0:
            raise ValueError(f"{name}: empty document selection")
        if np.min(documents) < 0 or np.max(documents) >= indexed_dataset.sizes.shape[0]:
            raise ValueError(f"{name}: document indices out of range")

        if build_index_mappings:
            self.doc_idx, self.sample_idx, self.shuffle_idx = _build_index_mappings(
                name, data_prefix, documents, indexed_dataset.sizes,
                num_samples, seq_length, seed, use_shared_fs=use_shared_fs)
            self.shuffle_idx_len = self.shuffle_idx.shape[0] - 1
            self.sample_idx_len = self.sample_idx.shape[0] - 1

    def __len__(self):
        return min(self.shuffle_idx_len, self.sample_idx_len)

    def __getitem__(self, idx):
        try:
            return self._get(idx)
        except IndexError:
            new_idx = idx % len(self)
            logger.warning(f"{self.name}: index {idx} out of bounds, wrapping to {new_idx}")
            return self[new_idx]

    def _get(self, idx):
        idx = self.shuffle_idx[idx]
        doc_f, offset_f = self.sample_idx[idx]
        doc_l, offset_l = self.sample_idx[idx + 1]
        stores = [self.indexed_dataset]
        if self.label_dataset is not None:
            stores.append(self.label_dataset)
        outs = []
        for store in stores:
            if doc_f == doc_l:
                outs.append(store.get(self.doc_idx[doc_f], offset=offset_f,
                                      length=offset_l - offset_f + 1))
            else:
                parts = [store.get(self.doc_idx[doc_f], offset=offset_f)]
                parts.extend(store.get(self.doc_idx[i]) for i in range(doc_f + 1, doc_l))
                parts.append(store.get(self.doc_idx[doc_l], length=offset_l + 1))
                outs.append(np.concatenate(parts))
        sample = {"input_ids": np.asarray(outs[0], dtype=np.int64)}
        if len(outs) > 1:
            sample["label"] = np.asarray(outs[1], dtype=np.int64)
        return sample


def _num_tokens(documents, sizes):
    return np.sum(sizes[documents])


def _num_epochs(tokens_per_epoch, seq_length, num_samples):
    """Smallest epoch count whose token total yields >= num_samples windows.
    The -1: each sample's last token is the next sample's first."""
    epochs, total = 0, 0
    while True:
        epochs += 1
        total += tokens_per_epoch
        if (total - 1) // seq_length >= num_samples:
            return epochs


def _build_doc_idx(documents, num_epochs, np_rng):
    doc_idx = np.tile(np.asarray(documents, dtype=np.int32), num_epochs)
    np_rng.shuffle(doc_idx)
    return doc_idx


def _build_shuffle_idx(size, np_rng):
    dtype = np.uint32 if size < np.iinfo(np.uint32).max - 1 else np.int64
    shuffle_idx = np.arange(size, dtype=dtype)
    np_rng.shuffle(shuffle_idx)
    return shuffle_idx


def build_sample_idx_py(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch):
    """Pure-python fallback / oracle for the C++ builder (tests diff the two)."""
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    sample_idx = np.zeros([num_samples + 1, 2], dtype=np.int64)
    doc_cursor, doc_offset = 0, 0
    for s in range(1, num_samples + 1):
        remaining = seq_length + 1
        while remaining != 0:
            doc_len = sizes[doc_idx[doc_cursor]] - doc_offset
            if doc_len >= remaining:
                doc_offset += remaining - 1
                remaining = 0
            else:
                remaining -= doc_len
                doc_cursor += 1
                doc_offset = 0
        sample_idx[s] = (doc_cursor, doc_offset)
    return sample_idx


def _build_index_mappings(name, data_prefix, documents, sizes, num_samples,
                          seq_length, seed, use_shared_fs=True):
    tokens_per_epoch = int(_num_tokens(documents, sizes))
    num_epochs = _num_epochs(tokens_per_epoch, seq_length, num_samples)
    np_rng = np.random.RandomState(seed=seed)

    # Cache naming is part of the on-disk contract (reference dataset.py:152-159).
    stem = f"{data_prefix}_{name}_indexmap_{num_samples}ns_{seq_length}sl_{seed}s"
    doc_idx_file = stem + "_doc_idx.npy"
    sample_idx_file = stem + "_sample_idx.npy"
    shuffle_idx_file = stem + "_shuffle_idx.npy"

    if not use_shared_fs:
        is_builder = int(os.environ.get("LOCAL_RANK", 0)) == 0
    elif dist.is_initialized():
        is_builder = dist.get_rank() == 0
    else:
        is_builder = True

    if is_builder and not all(os.path.isfile(f) for f in
                              (doc_idx_file, sample_idx_file, shuffle_idx_file)):
        t0 = time.time()
        doc_idx = _build_doc_idx(documents, num_epochs, np_rng)
        np.save(doc_idx_file, doc_idx, allow_pickle=True)

        sizes32 = np.asarray(sizes, dtype=np.int32)
        try:
            from relora_amd.data import _index_helpers as helpers
            ns = (num_epochs * tokens_per_epoch - 1) // seq_length
            if 2 * (ns + 1) < np.iinfo(np.int32).max:
                sample_idx = helpers.build_sample_idx_int32(
                    sizes32, doc_idx, seq_length, num_epochs, tokens_per_epoch)
            else:
                sample_idx = helpers.build_sample_idx_int64(
                    sizes32, doc_idx, seq_length, num_epochs, tokens_per_epoch)
        except ImportError:
            logger.warning("_index_helpers extension missing; using python sample_idx builder")
            sample_idx = build_sample_idx_py(
                sizes32, doc_idx, seq_length, num_epochs, tokens_per_epoch)
        np.save(sample_idx_file, sample_idx, allow_pickle=True)

        shuffle_idx = _build_shuffle_idx(sample_idx.shape[0] - 1, np_rng)
        np.save(shuffle_idx_file, shuffle_idx, allow_pickle=True)
        logger.info(f"{name}: built index mappings in {time.time() - t0:.2f}s "
                    f"({sample_idx.shape[0] - 1} samples, {num_epochs} epochs)")

    if dist.is_initialized():
        # Barrier so non-builder ranks don't read half-written .npy files.
        # gloo/CPU-safe (the reference used an all-reduce on a CUDA tensor).
        dist.barrier()

    doc_idx = np.load(doc_idx_file, allow_pickle=True, mmap_mode="r")
    sample_idx = np.load(sample_idx_file, allow_pickle=True, mmap_mode="r")
    shuffle_idx = np.load(shuffle_idx_file, allow_pickle=True, mmap_mode="r")
    return doc_idx, sample_idx, shuffle_idx
