"""mx.npx — numpy-extension namespace (reference python/mxnet/numpy_extension):
neural-network ops over np-style arrays, plus np-mode switches."""
import torch

from .ndarray.ndarray import NDArray
from .ndarray import ops as _ops
from .util import set_np, reset_np, is_np_array  # noqa: F401

__all__ = ['set_np', 'reset_np', 'is_np_array', 'softmax', 'log_softmax',
           'masked_softmax', 'relu', 'sigmoid', 'gelu', 'batch_norm',
           'layer_norm', 'fully_connected', 'convolution', 'pooling',
           'embedding', 'one_hot', 'pick', 'topk', 'batch_dot',
           'sequence_mask', 'gamma', 'erf']

softmax = _ops.softmax
log_softmax = _ops.log_softmax
relu = lambda x: NDArray(torch.relu(x._t))
sigmoid = lambda x: NDArray(torch.sigmoid(x._t))
gelu = lambda x: NDArray(torch.nn.functional.gelu(x._t, approximate='tanh'))
batch_norm = _ops.BatchNorm
layer_norm = _ops.LayerNorm
fully_connected = _ops.FullyConnected
convolution = _ops.Convolution
pooling = _ops.Pooling
embedding = _ops.Embedding
one_hot = _ops.one_hot
pick = _ops.pick
topk = _ops.topk
batch_dot = _ops.batch_dot
sequence_mask = _ops.sequence_mask


def masked_softmax(data, mask, axis=-1, temperature=1.0):
    from .ops import nn as _nn
    t = data._t if isinstance(data, NDArray) else data
    m = mask._t if isinstance(mask, NDArray) else mask
    return NDArray(_nn.masked_softmax(t, m, axis, temperature))


def gamma(x):
    return NDArray(torch.special.gammaln(x._t).exp())


def erf(x):
    return NDArray(torch.erf(x._t))
