"""Gluon — the imperative/hybrid model API (reference python/mxnet/gluon)."""
from .block import (Block, HybridBlock, SymbolBlock, Sequential,
                    HybridSequential)
from .parameter import Parameter, Constant, ParameterDict
from .trainer import Trainer
from . import nn, loss, metric
from . import rnn
from . import data
from . import model_zoo
from . import probability  # noqa: F401
