"""Gluon Estimator fit-loop (reference gluon/contrib/estimator)."""
from .estimator import Estimator  # noqa: F401
from .event_handler import (  # noqa: F401
    TrainBegin, TrainEnd, EpochBegin, EpochEnd, BatchBegin, BatchEnd,
    StoppingHandler, MetricHandler, ValidationHandler, LoggingHandler,
    CheckpointHandler, EarlyStoppingHandler)
