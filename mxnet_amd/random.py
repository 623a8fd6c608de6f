"""mx.random (reference python/mxnet/random.py)."""
import torch

from .ndarray.ops import (random_uniform as uniform,          # noqa: F401
                          random_normal as normal,            # noqa: F401
                          random_randint as randint,          # noqa: F401
                          shuffle, sample_multinomial as multinomial)  # noqa: F401


def seed(seed_state, ctx='all'):
    """Seed host + all HIP device RNGs (reference MXRandomSeed)."""
    torch.manual_seed(seed_state)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed_state)
    import numpy as _np
    import random as _random
    _np.random.seed(seed_state % (2 ** 32))
    _random.seed(seed_state)
