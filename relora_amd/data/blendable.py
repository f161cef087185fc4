"""Weighted mixture over multiple GPT2Datasets.

Per-sample (dataset, sample) assignment follows the greedy largest-deficit
interleaving computed by the C++ ``build_blending_indices`` (reference
peft_pretraining/megatron_dataset/blendable_dataset.py:27-79).
"""

import time

import numpy as np
import torch.utils.data

from relora_amd.utils.logging import logger


def blending_indices_py(weights, size):
    """Pure-python oracle for the C++ builder (used by tests and as fallback)."""
    dataset_index = np.zeros(size, dtype=np.uint8)
    dataset_sample_index = np.zeros(size, dtype=np.int64)
    taken = np.zeros(len(weights), dtype=np.int64)
    for i in range(size):
        n = max(float(i), 1.0)
        err = weights * n - taken
        best = int(np.argmax(err))
        dataset_index[i] = best
        dataset_sample_index[i] = taken[best]
        taken[best] += 1
    return dataset_index, dataset_sample_index


class BlendableDataset(torch.utils.data.Dataset):
    def __init__(self, datasets, weights):
        self.datasets = datasets
        if len(datasets) != len(weights):
            raise ValueError("one weight per dataset required")
        if len(datasets) >= 255:
            raise ValueError("at most 254 datasets (uint8 index)")

        self.size = sum(len(d) for d in datasets)
        weights = np.asarray(weights, dtype=np.float64)
        total = weights.sum()
        if total <= 0:
            raise ValueError("weights must sum to > 0")
        weights = weights / total

        t0 = time.time()
        try:
            from relora_amd.data import _index_helpers as helpers
            self.dataset_index = np.zeros(self.size, dtype=np.uint8)
            self.dataset_sample_index = np.zeros(self.size, dtype=np.int64)
            helpers.build_blending_indices(
                self.dataset_index, self.dataset_sample_index,
                weights, len(datasets), self.size, False)
        except ImportError:
            logger.warning("_index_helpers extension missing; python blending fallback")
            self.dataset_index, self.dataset_sample_index = blending_indices_py(
                weights, self.size)
        if time.time() - t0 > 5.0:
            logger.info(f"blending indices built in {time.time() - t0:.2f}s")

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        try:
            d = self.dataset_index[idx]
            s = self.dataset_sample_index[idx]
            return self.datasets[d][s]
        except IndexError:
            new_idx = idx % len(self)
            logger.warning(f"blendable index {idx} out of bounds, wrapping to {new_idx}")
            return self[new_idx]
