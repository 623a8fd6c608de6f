from ..block import (Block, HybridBlock, SymbolBlock, Sequential,
                     HybridSequential)
from .basic_layers import (Dense, Dropout, BatchNorm, BatchNormReLU,
                           SyncBatchNorm, Embedding, LayerNorm, GroupNorm,
                           InstanceNorm, Flatten, Activation, Lambda,
                           HybridLambda, Identity)
from .conv_layers import (Conv1D, Conv2D, MaxPool2D, AvgPool2D,
                          GlobalMaxPool2D, GlobalAvgPool2D)
from .activations import (LeakyReLU, PReLU, ELU, SELU, GELU, Swish, SiLU)
