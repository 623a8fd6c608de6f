"""Gluon contrib (reference python/mxnet/gluon/contrib)."""
from . import estimator  # noqa: F401
