"""Gluon data API (reference python/mxnet/gluon/data)."""
from .dataset import Dataset, ArrayDataset, SimpleDataset, RecordFileDataset
from .sampler import Sampler, SequentialSampler, RandomSampler, BatchSampler
from .dataloader import DataLoader
from . import vision
from . import batchify

from .threaded_loader import ThreadedDataLoader  # noqa: F401
