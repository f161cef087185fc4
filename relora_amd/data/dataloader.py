"""HF-path data utilities (parity with reference
`peft_pretraining/dataloader.py`): `tokenize_and_chunk` (:57-124),
`PreprocessedIterableDataset` (:13-54), `SkipBatchSampler`/`SkipDataLoader`
resume fast-forward (:128-170); plus `SyntheticDataset`, an offline
random-token dataset used by bench.py and tests (this image has no network
for real corpora — BASELINE.json: synthetic data, random-init weights)."""

import itertools
from itertools import chain

import torch
from torch.utils.data import BatchSampler, DataLoader, IterableDataset, get_worker_info


class PreprocessedIterableDataset(IterableDataset):
    """Tokenize raw-text examples on the fly and yield fixed-length batches."""

    def __init__(self, data, tokenizer, batch_size, max_length):
        super().__init__()
        self.data = data
        self.tokenizer = tokenizer
        self.batch_size = batch_size
        self.max_length = max_length

    def __iter__(self):
        worker_info = get_worker_info()
        if worker_info is None:
            iter_data = iter(self.data)
        else:
            iter_data = itertools.islice(
                self.data, worker_info.id, None, worker_info.num_workers
            )

        batch = []
        for example in iter_data:
            tokenized = self.tokenizer(
                example["text"],
                max_length=self.max_length,
                truncation=True,
                padding="max_length",
                return_tensors="pt",
            )
            batch.append(tokenized)
            if len(batch) == self.batch_size:
                yield self._format_batch(batch)
                batch = []
        if batch:
            yield self._format_batch(batch)

    @staticmethod
    def _format_batch(batch):
        input_ids = torch.stack([item["input_ids"].squeeze(0) for item in batch])
        attention_mask = torch.stack([item["attention_mask"].squeeze(0) for item in batch])
        return {"input_ids": input_ids, "attention_mask": attention_mask}


def tokenize_and_chunk(tokenizer, dataset, text_field, sequence_length, num_cpu=None):
    """Tokenize (+EOS per document), concatenate, and chunk into
    `sequence_length` blocks; drops attention_mask (we never pad for LM)."""
    import multiprocessing

    if num_cpu is None:
        num_cpu = multiprocessing.cpu_count()
    extra_map_kwargs = {"num_proc": num_cpu}
    if isinstance(dataset, IterableDataset):
        extra_map_kwargs = {}

    _len_pre = len(dataset)
    tokenized = dataset.map(
        lambda example: tokenizer([t + tokenizer.eos_token for t in example[text_field]]),
        batched=True,
        remove_columns=[text_field],
        **extra_map_kwargs,
    )
    assert "input_ids" in tokenized["train"].features
    assert len(tokenized["train"]) > 0
    assert len(tokenized) == _len_pre

    block_size = sequence_length

    def group_texts(examples):
        concatenated = {k: list(chain(*examples[k])) for k in examples.keys()}
        total_length = len(concatenated["input_ids"])
        if total_length >= block_size:
            total_length = (total_length // block_size) * block_size
        return {
            k: [t[i : i + block_size] for i in range(0, total_length, block_size)]
            for k, t in concatenated.items()
            if k != "attention_mask"
        }

    return tokenized.map(
        group_texts, batched=True, remove_columns=["attention_mask"], **extra_map_kwargs
    )


class SkipBatchSampler(BatchSampler):
    """BatchSampler that skips the first `skip_batches` batches (resume)."""

    def __init__(self, batch_sampler, skip_batches=0):
        self.batch_sampler = batch_sampler
        self.skip_batches = skip_batches

    def __iter__(self):
        for index, samples in enumerate(self.batch_sampler):
            if index >= self.skip_batches:
                yield samples

    @property
    def total_length(self):
        return len(self.batch_sampler)

    def __len__(self):
        return len(self.batch_sampler) - self.skip_batches


class SkipDataLoader(DataLoader):
    """DataLoader that skips the first `skip_batches` batches (resume)."""

    def __init__(self, dataset, skip_batches=0, **kwargs):
        super().__init__(dataset, **kwargs)
        self.skip_batches = skip_batches

    def __iter__(self):
        for index, batch in enumerate(super().__iter__()):
            if index >= self.skip_batches:
                yield batch


class SyntheticDataset(torch.utils.data.Dataset):
    """Deterministic random-token dataset of `length` sequences of
    `seq_len` tokens over `vocab_size` (offline benchmarking / tests)."""

    def __init__(self, vocab_size, seq_len, length, seed=1234):
        self.vocab_size = vocab_size
        self.seq_len = seq_len
        self.length = length
        self.seed = seed

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        return {
            "input_ids": torch.randint(0, self.vocab_size, (self.seq_len,), generator=g)
        }
