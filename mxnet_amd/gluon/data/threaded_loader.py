"""Native threaded DataLoader (reference src/io/dataloader.cc
ThreadedDataLoader): batch assembly runs on C++ worker threads
(mxnet_amd._dataloader), not python processes — no fork, no IPC, and
the GIL is released during the row gather."""
import numpy as np
import torch

from ...ndarray.ndarray import NDArray

__all__ = ['ThreadedDataLoader']


class ThreadedDataLoader:
    """Iterate array-backed data in mini-batches assembled natively.

    ``data``: one array or a tuple/list of arrays (numpy / NDArray /
    torch CPU tensors) sharing dim 0 — e.g. ``(X, Y)``.
    """

    def __init__(self, data, batch_size, shuffle=False, num_workers=2,
                 last_batch='keep'):
        from ... import _dataloader as _dl
        self._dl_mod = _dl
        if not isinstance(data, (tuple, list)):
            data = (data,)
        self._arrays = []
        for a in data:
            if isinstance(a, NDArray):
                a = a.handle
            if isinstance(a, torch.Tensor):
                a = a.detach().cpu().contiguous().numpy()
            self._arrays.append(np.ascontiguousarray(a))
        n = self._arrays[0].shape[0]
        assert all(a.shape[0] == n for a in self._arrays)
        self._n = n
        self._bs = batch_size
        self._shuffle = shuffle
        self._workers = num_workers
        self._drop_last = last_batch == 'discard'

    def __len__(self):
        return self._n // self._bs if self._drop_last \
            else (self._n + self._bs - 1) // self._bs

    def __iter__(self):
        order = np.random.permutation(self._n) if self._shuffle \
            else np.arange(self._n)
        batcher = self._dl_mod.ThreadedBatcher(
            list(self._arrays), [int(i) for i in order], self._bs,
            self._workers, self._drop_last)
        try:
            for _ in range(len(self)):
                arrays = batcher.next()
                batch = [NDArray(torch.from_numpy(a)) for a in arrays]
                yield batch[0] if len(batch) == 1 else tuple(batch)
        finally:
            batcher.shutdown()
