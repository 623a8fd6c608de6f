"""mx.nd namespace: NDArray + ops."""
from .ndarray import (NDArray, array, zeros, ones, full, empty, arange,
                      from_torch, waitall, concat, stack, save, load,
                      zeros_like, ones_like)
from .ops import *  # noqa: F401,F403
from . import ops
from .ndarray import concat, stack  # keep creation-module versions authoritative

from . import contrib  # noqa: F401
from . import random  # noqa: F401
from . import sparse  # noqa: F401


def Custom(*inputs, op_type=None, **kwargs):
    """Run a registered python CustomOp (reference nd.Custom)."""
    from ..operator import invoke
    return invoke(op_type, *inputs, **kwargs)
