"""mx.nd.random (reference python/mxnet/ndarray/random.py): thin
namespace over the sampling ops — native philox kernels under the
native runtime, torch samplers otherwise."""
from .ops import (random_uniform as uniform,          # noqa: F401
                  random_normal as normal,            # noqa: F401
                  random_randint as randint,          # noqa: F401
                  shuffle, sample_multinomial as multinomial)  # noqa: F401
