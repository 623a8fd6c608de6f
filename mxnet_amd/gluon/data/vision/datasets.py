"""Vision datasets (reference gluon/data/vision/datasets.py).

No network in this environment: the MNIST/CIFAR classes load from a local
``root`` if the standard raw files are present, and raise a clear error
otherwise.  ``SyntheticImageDataset`` provides the shapes for benchmarks
and tests (BASELINE: synthetic data, random-init weights).
"""
import gzip
import os
import struct

import numpy as _np
import torch

from ..dataset import Dataset
from ....ndarray.ndarray import NDArray


class SyntheticImageDataset(Dataset):
    """Random images + labels of a given shape, deterministic per index."""

    def __init__(self, length=1024, shape=(28, 28, 1), num_classes=10,
                 dtype='float32', seed=0):
        self._length = length
        self._shape = tuple(shape)
        self._classes = num_classes
        self._seed = seed

    def __len__(self):
        return self._length

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self._seed + idx)
        img = torch.rand(self._shape, generator=g)
        label = int(torch.randint(0, self._classes, (1,), generator=g))
        return NDArray(img), label


class _IdxDataset(Dataset):
    """MNIST-style idx-format loader."""

    def __init__(self, root, image_file, label_file, train):
        image_path = os.path.join(os.path.expanduser(root), image_file)
        label_path = os.path.join(os.path.expanduser(root), label_file)
        if not (os.path.exists(image_path) and os.path.exists(label_path)):
            raise FileNotFoundError(
                f'{image_path} not found and no network to download; '
                f'place the raw files under {root}')
        with gzip.open(label_path, 'rb') if label_path.endswith('.gz') \
                else open(label_path, 'rb') as f:
            struct.unpack('>II', f.read(8))
            self._labels = _np.frombuffer(f.read(), dtype=_np.uint8)
        with gzip.open(image_path, 'rb') if image_path.endswith('.gz') \
                else open(image_path, 'rb') as f:
            _, num, rows, cols = struct.unpack('>IIII', f.read(16))
            data = _np.frombuffer(f.read(), dtype=_np.uint8)
            self._images = data.reshape(num, rows, cols, 1)

    def __len__(self):
        return len(self._labels)

    def __getitem__(self, idx):
        return (NDArray(torch.from_numpy(self._images[idx].copy())),
                int(self._labels[idx]))


class MNIST(_IdxDataset):
    def __init__(self, root='~/.mxnet/datasets/mnist', train=True, transform=None):
        prefix = 'train' if train else 't10k'
        super().__init__(root, f'{prefix}-images-idx3-ubyte.gz',
                         f'{prefix}-labels-idx1-ubyte.gz', train)


class FashionMNIST(_IdxDataset):
    def __init__(self, root='~/.mxnet/datasets/fashion-mnist', train=True,
                 transform=None):
        prefix = 'train' if train else 't10k'
        super().__init__(root, f'{prefix}-images-idx3-ubyte.gz',
                         f'{prefix}-labels-idx1-ubyte.gz', train)


class _CIFAR(Dataset):
    _files_train = []
    _files_test = []

    def __init__(self, root, train=True, transform=None, fine_label=False):
        root = os.path.expanduser(root)
        files = self._files_train if train else self._files_test
        imgs, labels = [], []
        for fname in files:
            path = os.path.join(root, fname)
            if not os.path.exists(path):
                raise FileNotFoundError(
                    f'{path} not found and no network to download')
            raw = _np.fromfile(path, dtype=_np.uint8)
            rec = raw.reshape(-1, 3073)
            labels.append(rec[:, 0])
            imgs.append(rec[:, 1:].reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1))
        self._images = _np.concatenate(imgs)
        self._labels = _np.concatenate(labels)

    def __len__(self):
        return len(self._labels)

    def __getitem__(self, idx):
        return (NDArray(torch.from_numpy(self._images[idx].copy())),
                int(self._labels[idx]))


class CIFAR10(_CIFAR):
    _files_train = [f'data_batch_{i}.bin' for i in range(1, 6)]
    _files_test = ['test_batch.bin']

    def __init__(self, root='~/.mxnet/datasets/cifar10', train=True,
                 transform=None):
        super().__init__(root, train, transform)


class CIFAR100(_CIFAR):
    _files_train = ['train.bin']
    _files_test = ['test.bin']

    def __init__(self, root='~/.mxnet/datasets/cifar100', train=True,
                 transform=None, fine_label=False):
        super().__init__(root, train, transform, fine_label)
