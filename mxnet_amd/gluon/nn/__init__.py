from ..block import (Block, HybridBlock, SymbolBlock, Sequential,
                     HybridSequential)
from .basic_layers import (Dense, Dropout, BatchNorm, BatchNormReLU,
                           SyncBatchNorm, Embedding, LayerNorm, GroupNorm,
                           InstanceNorm, Flatten, Activation, Lambda,
                           HybridLambda, Identity)
from .conv_layers import (Conv1D, Conv2D, Conv3D, Conv1DTranspose,
                          Conv2DTranspose, MaxPool2D, AvgPool2D,
                          MaxPool3D, AvgPool3D, GlobalMaxPool2D,
                          GlobalAvgPool2D, PixelShuffle2D)
from .activations import (LeakyReLU, PReLU, ELU, SELU, GELU, Swish, SiLU)
