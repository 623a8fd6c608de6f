"""Contrib namespace (reference python/mxnet/contrib)."""
from . import quantization  # noqa: F401
