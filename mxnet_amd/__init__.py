"""mxnet_amd — a brand-new MI355X-native deep-learning framework with the
capabilities of Apache MXNet (incubator-mxnet).

Not a port: the compute substrate is PyTorch-ROCm tensors + hand-written
gfx950 HIP kernels (MFMA/LDS-tiled) + RCCL over xGMI; asynchronous
execution rides HIP streams instead of a worker-thread engine; hybridized
blocks capture hipGraphs instead of building a CachedOp memory plan.
Checkpoint formats (.params / -symbol.json) are byte-compatible with the
reference.

Usage mirrors `import mxnet as mx`:

    import mxnet_amd as mx
    net = mx.gluon.model_zoo.vision.resnet50_v1(layout='NHWC')
    net.initialize(ctx=mx.gpu(0))
    with mx.autograd.record():
        loss = ...
    loss.backward()
"""
__version__ = '0.1.0'

from .context import Context, cpu, gpu, cpu_pinned, current_context, num_gpus
from .base import MXNetError
from . import ndarray
from . import ndarray as nd
from . import operator
from . import rtc
from . import library
from . import image
from . import image as img
from . import numpy as np  # mx.np numpy-compatible namespace
from . import symbol
from . import symbol as sym
from . import autograd
from . import initializer
from . import initializer as init
from . import optimizer
from . import gluon
from . import io
from . import parallel
from .parallel import kvstore as kv
from . import utils
from . import amp
from . import profiler
from . import lr_scheduler
from . import random
from .ndarray.ndarray import waitall
from . import engine
from .util import is_np_array, set_np, use_np

# mx.metric alias (pre-2.0 location), mirroring gluon.metric
from .gluon import metric

def __getattr__(name):
    if name == 'npx':
        import importlib
        mod = importlib.import_module('.numpy_extension', __name__)
        globals()['npx'] = mod
        return mod
    # lazily imported to avoid import cycles
    if name in ('test_utils', 'runtime', 'visualization',
                'numpy_extension'):
        import importlib
        mod = importlib.import_module('.' + name, __name__)
        globals()[name] = mod
        return mod
    raise AttributeError(name)
