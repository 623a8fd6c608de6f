"""DataLoader (reference gluon/data/dataloader.py).

MI355X-native design note: the reference moves worker->main NDArrays
through POSIX shared memory with ForkingPickler reductions
(dataloader.py:50-93).  Here multiprocessing workers exchange torch CPU
tensors, which already use shared-memory file descriptors via torch's
ForkingPickler reductions — the same mechanism, supplied by the substrate.
Pinned-memory staging + async H2D copy on the copy stream happens in the
consumer via ``pin_memory=True``.
"""
import multiprocessing as mp

import torch

from ...ndarray.ndarray import NDArray
from .sampler import SequentialSampler, RandomSampler, BatchSampler
from . import batchify as _batchify_mod


def default_batchify_fn(data):
    """Stack samples into a batch (reference default_batchify_fn)."""
    return _batchify_mod.Stack()(data)


class _WorkerDataset(torch.utils.data.Dataset):
    def __init__(self, dataset):
        self._dataset = dataset

    def __len__(self):
        return len(self._dataset)

    def __getitem__(self, idx):
        item = self._dataset[idx]

        def unwrap(x):
            if isinstance(x, NDArray):
                # native samples cross the process boundary as numpy
                # (torch CPU tensors ride shm reductions either way)
                return torch.as_tensor(x.asnumpy()) if x.is_native else x._t
            if isinstance(x, tuple):
                return tuple(unwrap(i) for i in x)
            return x
        return unwrap(item)


class DataLoader:
    """Iterates a Dataset in mini-batches with multiprocessing prefetch."""

    def __init__(self, dataset, batch_size=None, shuffle=False, sampler=None,
                 last_batch=None, batch_sampler=None, batchify_fn=None,
                 num_workers=0, pin_memory=False, prefetch=None,
                 thread_pool=False, timeout=120):
        self._dataset = dataset
        self._pin_memory = pin_memory
        if batch_sampler is None:
            if batch_size is None:
                raise ValueError('batch_size required')
            if sampler is None:
                sampler = RandomSampler(len(dataset)) if shuffle \
                    else SequentialSampler(len(dataset))
            batch_sampler = BatchSampler(sampler, batch_size,
                                         last_batch or 'keep')
        self._batch_sampler = batch_sampler
        self._num_workers = num_workers
        self._batchify_fn = batchify_fn or default_batchify_fn

    def __len__(self):
        return len(self._batch_sampler)

    def __iter__(self):
        if self._num_workers == 0:
            for batch_idx in self._batch_sampler:
                samples = [self._dataset[i] for i in batch_idx]
                yield self._batchify_fn(samples)
            return
        # multiprocessing path via torch's shared-memory loader machinery
        loader = torch.utils.data.DataLoader(
            _WorkerDataset(self._dataset),
            batch_sampler=list(self._batch_sampler),
            num_workers=self._num_workers,
            pin_memory=self._pin_memory,
            collate_fn=lambda samples: self._batchify_fn(
                [_rewrap(s) for s in samples]))
        yield from loader


def _rewrap(x):
    if isinstance(x, torch.Tensor):
        return NDArray(x)
    if isinstance(x, tuple):
        return tuple(_rewrap(i) for i in x)
    return x
