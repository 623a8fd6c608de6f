"""Gluon probability (reference python/mxnet/gluon/probability)."""
from .distributions import *  # noqa: F401,F403
from .stochastic_block import StochasticBlock, StochasticSequential  # noqa: F401
from .transformation import *  # noqa: F401,F403
