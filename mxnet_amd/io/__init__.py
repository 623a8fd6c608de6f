"""Legacy data iterators (reference python/mxnet/io/io.py)."""
from collections import namedtuple

import numpy as _np
import torch

from ..ndarray.ndarray import NDArray, array
from . import recordio  # noqa: F401

DataDesc = namedtuple('DataDesc', ['name', 'shape'])


class DataBatch:
    def __init__(self, data, label=None, pad=0, index=None,
                 provide_data=None, provide_label=None):
        self.data = data
        self.label = label
        self.pad = pad
        self.index = index
        self.provide_data = provide_data
        self.provide_label = provide_label


class DataIter:
    def __init__(self, batch_size=0):
        self.batch_size = batch_size

    def __iter__(self):
        return self

    def reset(self):
        pass

    def next(self):
        raise NotImplementedError

    def __next__(self):
        return self.next()


class NDArrayIter(DataIter):
    """Iterate dense NDArray/numpy data (reference io.py NDArrayIter)."""

    def __init__(self, data, label=None, batch_size=1, shuffle=False,
                 last_batch_handle='pad', data_name='data',
                 label_name='softmax_label'):
        super().__init__(batch_size)
        self.data = self._init_data(data, data_name)
        self.label = self._init_data(label, label_name) if label is not None else []
        self.num_data = self.data[0][1].shape[0]
        self.shuffle = shuffle
        self.last_batch_handle = last_batch_handle
        self.cursor = -batch_size
        self._order = _np.arange(self.num_data)

    @staticmethod
    def _init_data(data, default_name):
        if data is None:
            return []
        if isinstance(data, (NDArray, _np.ndarray, torch.Tensor)):
            data = {default_name: data}
        elif isinstance(data, (list, tuple)):
            data = {f'{default_name}_{i}' if i else default_name: d
                    for i, d in enumerate(data)}
        out = []
        for k, v in data.items():
            if not isinstance(v, NDArray):
                v = array(v)
            out.append((k, v))
        return out

    @property
    def provide_data(self):
        return [DataDesc(k, (self.batch_size,) + v.shape[1:])
                for k, v in self.data]

    @property
    def provide_label(self):
        return [DataDesc(k, (self.batch_size,) + v.shape[1:])
                for k, v in self.label]

    def reset(self):
        self.cursor = -self.batch_size
        if self.shuffle:
            _np.random.shuffle(self._order)

    def iter_next(self):
        self.cursor += self.batch_size
        return self.cursor < self.num_data

    def next(self):
        if not self.iter_next():
            raise StopIteration
        idx = self._order[self.cursor:self.cursor + self.batch_size]
        pad = 0
        if len(idx) < self.batch_size:
            if self.last_batch_handle == 'discard':
                raise StopIteration
            pad = self.batch_size - len(idx)
            idx = _np.concatenate([idx, self._order[:pad]])
        sel = torch.as_tensor(idx, dtype=torch.long)
        data = [NDArray(v._t[sel]) for _, v in self.data]
        label = [NDArray(v._t[sel]) for _, v in self.label]
        return DataBatch(data, label, pad=pad,
                         provide_data=self.provide_data,
                         provide_label=self.provide_label)


class ImageRecordIter(DataIter):
    """Threaded RecordIO image iterator (reference ImageRecordIter,
    src/io/iter_image_recordio_2.cc:887): C++ decode threads run the own
    baseline JPEG decoder (or raw records) + resize/crop/mirror augment
    into uint8 NHWC batches; see mxnet_amd/_imageio."""

    def __init__(self, path_imgrec, batch_size, data_shape,
                 shuffle=False, rand_crop=False, rand_mirror=False,
                 resize=0, preprocess_threads=0, seed=0, label_width=1,
                 **kwargs):
        super().__init__(batch_size)
        from .. import _imageio
        # data_shape mxnet-style (C,H,W) or NHWC (H,W,C with C last)
        if len(data_shape) == 3 and data_shape[0] in (1, 3):
            c, h, w = data_shape
        else:
            h, w, c = data_shape
        assert c == 3, 'ImageRecordIter decodes RGB'
        self._h, self._w = h, w
        self._it = _imageio.ImageRecordIter(
            path_imgrec, batch_size, h, w, preprocess_threads, shuffle,
            rand_crop, rand_mirror, resize, seed)
        self.batch_size = batch_size

    @property
    def num_records(self):
        return self._it.size

    def reset(self):
        self._it.reset()

    def next(self):
        from ..ndarray import ndarray as nd
        got, data, labels = self._it.next_batch(self.batch_size, self._h,
                                                self._w)
        if got == 0:
            raise StopIteration
        batch = DataBatch(
            data=[nd.array(data[:got])],
            label=[nd.array(labels[:got])],
            pad=self.batch_size - got)
        return batch

    def next_raw(self):
        """(got, uint8 ndarray [B,H,W,3], float32 labels) — zero-copy for
        pipelines that upload/normalize on the GPU."""
        return self._it.next_batch(self.batch_size, self._h, self._w)


class CSVIter(DataIter):
    """CSV file iterator (reference CSVIter, src/io/iter_csv.cc):
    C++ parse of data (and optional label) CSV files into float batches
    via mxnet_amd._imageio.CsvIter."""

    def __init__(self, data_csv, data_shape, label_csv=None,
                 label_shape=(1,), batch_size=1, round_batch=True,
                 **kwargs):
        super().__init__(batch_size)
        from .. import _imageio
        self._dshape = tuple(int(d) for d in data_shape)
        self._lshape = tuple(int(d) for d in label_shape) or (1,)
        dw = 1
        for d in self._dshape:
            dw *= d
        lw = 1
        for d in self._lshape:
            lw *= d
        self._dw, self._lw = dw, lw
        self._it = _imageio.CsvIter(data_csv, label_csv or '',
                                    batch_size, dw, lw)

    @property
    def num_records(self):
        return self._it.size

    def reset(self):
        self._it.reset()

    def next(self):
        from ..ndarray import ndarray as nd
        got, data, labels = self._it.next_batch(self.batch_size,
                                                self._dw, self._lw)
        if got == 0:
            raise StopIteration
        data = data[:got].reshape((got,) + self._dshape)
        labels = labels[:got].reshape((got,) + self._lshape)
        return DataBatch(data=[nd.array(data)], label=[nd.array(labels)],
                         pad=self.batch_size - got)
