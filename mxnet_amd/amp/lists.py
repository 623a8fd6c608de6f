"""AMP op lists (reference python/mxnet/amp/lists/symbol_fp16.py).

Classification of ops by numeric safety in fp16 — consumed by
``amp.init`` for models built from generic ops; the model-zoo layers are
already dtype-safe by construction (fp32 accumulation in every kernel).
"""

# run in fp16 freely (MFMA fp32-accumulating kernels)
FP16_FUNCS = [
    'FullyConnected', 'Convolution', 'dot', 'batch_dot', 'RNN',
]

# keep in fp32 (reductions / exponentials with wide dynamic range)
FP32_FUNCS = [
    'softmax', 'log_softmax', 'SoftmaxOutput', 'norm', 'mean', 'sum',
    'exp', 'log', 'BatchNorm', 'LayerNorm', 'erf', 'erfinv',
]

# run in the widest input type
WIDEST_TYPE_CASTS = [
    'elemwise_add', 'elemwise_sub', 'elemwise_mul', 'elemwise_div',
    'broadcast_add', 'broadcast_sub', 'broadcast_mul', 'broadcast_div',
    'add_n', 'where', 'concat', 'stack',
]

# conditionally fp32 (by parameterization)
CONDITIONAL_FP32_FUNCS = [
    ('Activation', 'act_type', ['softrelu']),
]
