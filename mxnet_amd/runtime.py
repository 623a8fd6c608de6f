"""Runtime feature detection (reference python/mxnet/runtime.py).

``Features`` reports what this build supports; on MI355X the compute
stack is ROCm/HIP + RCCL, so the reference's CUDA/CUDNN flags map to
their ROCm equivalents.
"""
import collections

import torch

__all__ = ['Feature', 'Features', 'feature_list']

Feature = collections.namedtuple('Feature', ['name', 'enabled'])


def _detect():
    from .ops.dispatch import hipops
    hip_ext = hipops() is not None
    feats = {
        'HIP': True,                       # ROCm compute path (reference CUDA)
        'ROCM': True,
        'MFMA': hip_ext,                   # hand-written gfx950 MFMA kernels
        'HIPOPS_EXTENSION': hip_ext,
        'GPU': torch.cuda.is_available(),
        'RCCL': torch.distributed.is_nccl_available(),  # NCCL==RCCL on ROCm
        'DIST_KVSTORE': True,
        'F16C': True,
        'OPENMP': True,
        'SSE': True,
        'BLAS_OPEN': True,
        'SIGNAL_HANDLER': False,
        'DEBUG': False,
        'TVM_OP': False,                   # no multi-backend dispatch
        'CUDA': False, 'CUDNN': False, 'NCCL': False, 'TENSORRT': False,
        'MKLDNN': False, 'ONEDNN': False,  # replaced by native CPU paths
        'INT64_TENSOR_SIZE': True,
    }
    return [Feature(k, v) for k, v in feats.items()]


class Features(dict):
    """dict of name -> Feature with is_enabled (reference runtime.py:69)."""

    instance = None

    def __init__(self):
        super().__init__([(f.name, f) for f in _detect()])

    def __repr__(self):
        return '[' + ', '.join(
            f'{"✔" if f.enabled else "✖"} {f.name}' for f in self.values()) + ']'

    def is_enabled(self, name):
        feature_name = name.upper()
        if feature_name not in self:
            raise RuntimeError(f'Feature {name} does not exist')
        return self[feature_name].enabled


def feature_list():
    return list(Features().values())
