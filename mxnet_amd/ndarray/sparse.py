"""Sparse NDArray storage types (reference python/mxnet/ndarray/sparse.py,
include/mxnet/ndarray.h kRowSparseStorage/kCSRStorage).

* ``RowSparseNDArray`` — a subset of rows is stored: ``data`` [nnz, ...]
  plus sorted ``indices`` [nnz].  The storage type of sparse gradients
  (Embedding/FullyConnected with sparse_grad) and sparse optimizer
  updates (lazy_update).
* ``CSRNDArray`` — classic CSR (indptr/indices/data) backed by
  torch.sparse_csr_tensor; ``dot(csr, dense)`` uses the library spmm.
"""
import numpy as _np
import torch

from .ndarray import NDArray

__all__ = ['RowSparseNDArray', 'CSRNDArray', 'row_sparse_array', 'csr_matrix',
           'zeros', 'retain', 'sparse_dot', 'add']


class BaseSparseNDArray:
    @property
    def stype(self):
        raise NotImplementedError

    def asnumpy(self):
        return self.tostype('default').asnumpy()


class RowSparseNDArray(BaseSparseNDArray):
    def __init__(self, data, indices, shape):
        self.data = data if isinstance(data, torch.Tensor) else data.handle
        self.indices = indices if isinstance(indices, torch.Tensor) \
            else indices.handle
        self.indices = self.indices.long()
        self._shape = tuple(shape)

    @property
    def stype(self):
        return 'row_sparse'

    @property
    def shape(self):
        return self._shape

    @property
    def dtype(self):
        return NDArray(self.data).dtype

    @property
    def context(self):
        return NDArray(self.data).context

    def tostype(self, stype):
        if stype == 'row_sparse':
            return self
        assert stype == 'default'
        dense = torch.zeros(self._shape, dtype=self.data.dtype,
                            device=self.data.device)
        if self.indices.numel():
            dense[self.indices] = self.data
        return NDArray(dense)

    def copyto(self, other):
        other._t.zero_()
        if self.indices.numel():
            other._t[self.indices] = self.data.to(other._t.dtype)
        return other

    def __repr__(self):
        return (f'<RowSparseNDArray {self._shape} '
                f'({self.indices.numel()} rows stored)>')


class CSRNDArray(BaseSparseNDArray):
    def __init__(self, t_csr):
        assert t_csr.layout == torch.sparse_csr
        self._t = t_csr

    @property
    def stype(self):
        return 'csr'

    @property
    def shape(self):
        return tuple(self._t.shape)

    @property
    def indptr(self):
        return NDArray(self._t.crow_indices())

    @property
    def indices(self):
        return NDArray(self._t.col_indices())

    @property
    def data(self):
        return NDArray(self._t.values())

    def tostype(self, stype):
        if stype == 'csr':
            return self
        assert stype == 'default'
        return NDArray(self._t.to_dense())

    def __repr__(self):
        return f'<CSRNDArray {self.shape}>'


def row_sparse_array(arg, shape=None, ctx=None, dtype=None):
    """Create from (data, indices) or a dense array (reference
    sparse.py:row_sparse_array)."""
    if isinstance(arg, tuple) and len(arg) == 2:
        data, indices = arg
        data = torch.as_tensor(_np.asarray(data))
        indices = torch.as_tensor(_np.asarray(indices)).long()
        if shape is None:
            shape = (int(indices.max().item()) + 1 if indices.numel() else 0,
                     ) + tuple(data.shape[1:])
        return RowSparseNDArray(data, indices, shape)
    dense = torch.as_tensor(_np.asarray(arg))
    nz_rows = (dense.reshape(dense.shape[0], -1).abs().sum(1) != 0).nonzero()
    idx = nz_rows.flatten()
    return RowSparseNDArray(dense[idx], idx, tuple(dense.shape))


def csr_matrix(arg, shape=None, ctx=None, dtype=None):
    if isinstance(arg, tuple) and len(arg) == 3:
        data, indices, indptr = arg
        t = torch.sparse_csr_tensor(
            torch.as_tensor(_np.asarray(indptr)).long(),
            torch.as_tensor(_np.asarray(indices)).long(),
            torch.as_tensor(_np.asarray(data)), size=shape)
        return CSRNDArray(t)
    dense = torch.as_tensor(_np.asarray(arg))
    return CSRNDArray(dense.to_sparse_csr())


def zeros(stype, shape, ctx=None, dtype=None):
    if stype == 'row_sparse':
        td = torch.zeros((0,) + tuple(shape[1:]))
        return RowSparseNDArray(td, torch.zeros(0, dtype=torch.long), shape)
    if stype == 'csr':
        return csr_matrix(torch.zeros(shape))
    return NDArray(torch.zeros(shape))


def retain(rsp, indices):
    """Keep only the given rows (reference sparse_retain op)."""
    idx = indices.handle.long() if isinstance(indices, NDArray) \
        else torch.as_tensor(indices).long()
    mask = torch.isin(rsp.indices, idx)
    return RowSparseNDArray(rsp.data[mask], rsp.indices[mask], rsp.shape)


def sparse_dot(lhs, rhs):
    """dot(csr, dense) via library spmm (reference dot FComputeEx)."""
    if isinstance(lhs, CSRNDArray):
        r = rhs.handle if isinstance(rhs, NDArray) else rhs
        return NDArray(torch.sparse.mm(lhs._t, r))
    raise TypeError('sparse_dot expects a CSRNDArray lhs')


def add(lhs, rhs):
    """row_sparse + row_sparse -> row_sparse (union of rows)."""
    assert isinstance(lhs, RowSparseNDArray) and \
        isinstance(rhs, RowSparseNDArray) and lhs.shape == rhs.shape
    all_idx = torch.unique(torch.cat([lhs.indices, rhs.indices]))
    data = torch.zeros((all_idx.numel(),) + tuple(lhs.data.shape[1:]),
                       dtype=lhs.data.dtype, device=lhs.data.device)
    pos_l = torch.searchsorted(all_idx, lhs.indices)
    pos_r = torch.searchsorted(all_idx, rhs.indices)
    data[pos_l] += lhs.data
    data[pos_r] += rhs.data.to(lhs.data.dtype)
    return RowSparseNDArray(data, all_idx, lhs.shape)
