"""Operator implementations (torch-tensor level) + HIP dispatch."""
from . import nn  # noqa: F401
from .dispatch import hipops, hip_required, use_hip  # noqa: F401
