"""Batch-sampler-level data-parallel sharding.

``DistributedBatchSampler`` builds the *global* batch first and hands each
rank its contiguous (or interleaved) slice — so the global sample order is
identical at any world size, and ``start_iter`` gives exact dataloader
fast-forward on resume (reference
peft_pretraining/megatron_dataset/samplers.py:88-165).
"""

import torch
import torch.utils.data


class RandomSampler(torch.utils.data.Sampler):
    """Epoch-seeded random sampler with optional wrap-around replacement
    (reference samplers.py:24-85)."""

    def __init__(self, data_source, replacement=False, num_samples=None):
        self.data_source = data_source
        self.replacement = replacement
        self._num_samples = num_samples
        self.epoch = -1
        self.wrap_around = 0

        if self._num_samples is not None and not replacement:
            raise ValueError("num_samples requires replacement=True")
        if self.num_samples <= 0:
            raise ValueError("empty data source")

    @property
    def num_samples(self):
        if self._num_samples is None:
            return len(self.data_source)
        return self._num_samples

    def __len__(self):
        return self.num_samples

    def set_epoch(self, epoch):
        self.epoch = epoch

    def __iter__(self):
        n = len(self.data_source)
        g = torch.Generator()
        if self.epoch >= 0:
            g.manual_seed(self.epoch)
        if self.replacement:
            for _ in range(self.num_samples):
                yield int(torch.randint(0, n, (1,), generator=g).item())
        else:
            yield from (int(i) for i in torch.randperm(n, generator=g))


class DistributedBatchSampler(torch.utils.data.sampler.BatchSampler):
    def __init__(self, sampler, batch_size, drop_last, rank=-1, world_size=2,
                 wrap_last=False, interleave=False):
        super().__init__(sampler, batch_size, drop_last)
        if rank < 0:
            raise ValueError("rank must be provided explicitly")
        self.rank = rank
        self.world_size = world_size
        self.sampler.wrap_around = 0
        self.wrap_around = 0
        self.wrap_last = wrap_last
        self.start_iter = 0
        self.interleave = interleave

    def __iter__(self):
        batch = []
        i = 0
        for idx in self._iterate(self.sampler, wrap_around=False):
            batch.append(idx)
            if len(batch) == self.batch_size:
                shard = self._shard(batch)
                if i >= self.start_iter:
                    yield shard
                    self.start_iter = 0
                i += 1
                batch = []
        if batch and not self.drop_last:
            if self.wrap_last:
                self.sampler.wrap_around -= self.batch_size
                self.wrap_around += len(batch)
                self.wrap_around %= self.batch_size
            yield self._shard(batch)
        if self.wrap_last:
            self.sampler.wrap_around += self.batch_size

    def _iterate(self, source, wrap_around=False):
        for i, idx in enumerate(source):
            if i < self.wrap_around % self.batch_size:
                continue
            if wrap_around:
                self.wrap_around += 1
                self.wrap_around %= self.batch_size
            yield idx

    def _shard(self, batch):
        if self.interleave:
            return batch[self.rank:self.batch_size:self.world_size]
        # Floor-division endpoints so ragged global batches still partition
        # exactly across ranks.
        start = self.rank * self.batch_size // self.world_size
        end = (self.rank + 1) * self.batch_size // self.world_size
        return batch[start:end]
