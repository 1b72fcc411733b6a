"""paddle.io parity (reference: python/paddle/io/reader.py:262 DataLoader,
dataloader/ worker machinery).

DataLoader delegates to torch.utils.data (multiprocess workers,
pin-memory, prefetch, persistent workers) with paddle's API shape:
paddle samplers/batch-samplers adapt onto the torch loader.
"""
from __future__ import annotations

import math
from typing import Iterable, Optional

import numpy as np
import torch
import torch.utils.data as tud


class Dataset(tud.Dataset):
    pass


class IterableDataset(tud.IterableDataset):
    pass


class TensorDataset(tud.TensorDataset):
    def __init__(self, tensors):
        super().__init__(*tensors)

    def __getitem__(self, index):
        return tuple(t[index] for t in self.tensors)


class ComposeDataset(Dataset):
    def __init__(self, datasets):
        self.datasets = datasets

    def __len__(self):
        return min(len(d) for d in self.datasets)

    def __getitem__(self, idx):
        out = []
        for d in self.datasets:
            item = d[idx]
            out.extend(item if isinstance(item, (tuple, list)) else [item])
        return tuple(out)


class ChainDataset(tud.ChainDataset):
    pass


class Subset(tud.Subset):
    pass


def random_split(dataset, lengths, generator=None):
    return tud.random_split(dataset, lengths, generator)


class Sampler(tud.Sampler):
    def __init__(self, data_source=None):
        self.data_source = data_source


class SequenceSampler(Sampler):
    def __iter__(self):
        return iter(range(len(self.data_source)))

    def __len__(self):
        return len(self.data_source)


class RandomSampler(Sampler):
    def __init__(self, data_source, replacement=False, num_samples=None, generator=None):
        super().__init__(data_source)
        self.replacement = replacement
        self._num = num_samples
        self.generator = generator

    def __iter__(self):
        n = len(self.data_source)
        num = self._num or n
        if self.replacement:
            yield from torch.randint(0, n, (num,)).tolist()
        else:
            yield from torch.randperm(n).tolist()[:num]

    def __len__(self):
        return self._num or len(self.data_source)


class BatchSampler(tud.Sampler):
    def __init__(self, dataset=None, sampler=None, shuffle=False, batch_size=1,
                 drop_last=False):
        self.batch_size = batch_size
        self.drop_last = drop_last
        if sampler is not None:
            self.sampler = sampler
        elif shuffle:
            self.sampler = RandomSampler(dataset)
        else:
            self.sampler = SequenceSampler(dataset)

    def __iter__(self):
        batch = []
        for idx in self.sampler:
            batch.append(idx)
            if len(batch) == self.batch_size:
                yield batch
                batch = []
        if batch and not self.drop_last:
            yield batch

    def __len__(self):
        n = len(self.sampler)
        if self.drop_last:
            return n // self.batch_size
        return (n + self.batch_size - 1) // self.batch_size


class DistributedBatchSampler(BatchSampler):
    """reference: python/paddle/io/dataloader/batch_sampler.py
    DistributedBatchSampler -- shards the dataset across ranks."""

    def __init__(self, dataset, batch_size, num_replicas=None, rank=None,
                 shuffle=False, drop_last=False):
        self.dataset = dataset
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.drop_last = drop_last
        import torch.distributed as dist
        if num_replicas is None:
            num_replicas = dist.get_world_size() if dist.is_initialized() else 1
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        self.nranks = num_replicas
        self.local_rank = rank
        self.epoch = 0
        self.num_samples = int(math.ceil(len(dataset) / num_replicas))
        self.total_size = self.num_samples * num_replicas

    def __iter__(self):
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        indices += indices[: (self.total_size - len(indices))]
        indices = indices[self.local_rank:self.total_size:self.nranks]
        batch = []
        for idx in indices:
            batch.append(idx)
            if len(batch) == self.batch_size:
                yield batch
                batch = []
        if batch and not self.drop_last:
            yield batch

    def set_epoch(self, epoch):
        self.epoch = epoch

    def __len__(self):
        if self.drop_last:
            return self.num_samples // self.batch_size
        return (self.num_samples + self.batch_size - 1) // self.batch_size


def _default_collate(batch):
    return tud.default_collate(batch)


class DataLoader:
    def __init__(self, dataset, feed_list=None, places=None, return_list=True,
                 batch_sampler=None, batch_size=1, shuffle=False, drop_last=False,
                 collate_fn=None, num_workers=0, use_buffer_reader=True,
                 prefetch_factor=2, use_shared_memory=True, timeout=0,
                 worker_init_fn=None, persistent_workers=False):
        self.dataset = dataset
        self.return_list = return_list
        kwargs = dict(
            num_workers=num_workers,
            collate_fn=collate_fn,
            timeout=timeout,
            worker_init_fn=worker_init_fn,
            pin_memory=torch.cuda.is_available(),
            persistent_workers=persistent_workers and num_workers > 0,
        )
        if num_workers > 0:
            kwargs["prefetch_factor"] = prefetch_factor
        if batch_sampler is not None:
            self._loader = tud.DataLoader(dataset, batch_sampler=batch_sampler, **kwargs)
            self.batch_sampler = batch_sampler
        elif batch_size is None:
            self._loader = tud.DataLoader(dataset, batch_size=None, **kwargs)
            self.batch_sampler = None
        else:
            self.batch_sampler = BatchSampler(dataset, shuffle=shuffle,
                                              batch_size=batch_size, drop_last=drop_last)
            self._loader = tud.DataLoader(dataset, batch_sampler=self.batch_sampler, **kwargs)

    def __iter__(self):
        dev = None
        if torch.cuda.is_available():
            dev = torch.device("cuda", torch.cuda.current_device())
        for batch in self._loader:
            if dev is not None:
                batch = _move(batch, dev)
            yield batch

    def __len__(self):
        return len(self._loader)


def _move(batch, dev):
    if isinstance(batch, torch.Tensor):
        return batch.to(dev, non_blocking=True)
    if isinstance(batch, (list, tuple)):
        return type(batch)(_move(b, dev) for b in batch)
    if isinstance(batch, dict):
        return {k: _move(v, dev) for k, v in batch.items()}
    return batch


def get_worker_info():
    return tud.get_worker_info()


# long-tail samplers/datasets (reference: io/__init__.py)
from torch.utils.data import (  # noqa: F401
    ConcatDataset,
    SubsetRandomSampler,
    WeightedRandomSampler,
)
