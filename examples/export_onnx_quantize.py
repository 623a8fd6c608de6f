"""Deployment walk-through: train a small net, export it three ways
(.params+symbol.json, ONNX, int8), and check the outputs agree.

Run: python examples/export_onnx_quantize.py
"""
import os
import tempfile

import numpy as np
import torch

import mxnet_amd as mx
from mxnet_amd import autograd
from mxnet_amd.gluon import Trainer, nn
from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
from mxnet_amd.contrib import onnx as mxonnx
from mxnet_amd.contrib.quantization import quantize_net


def main():
    torch.manual_seed(0)
    net = nn.HybridSequential()
    net.add(nn.Conv2D(16, 3, padding=1, in_channels=3),
            nn.BatchNorm(in_channels=16),
            nn.Activation('relu'),
            nn.MaxPool2D(2),
            nn.Flatten(),
            nn.Dense(10, in_units=16 * 16 * 16))
    net.initialize()
    net.hybridize()

    X = mx.nd.from_torch(torch.randn(8, 3, 32, 32))
    Y = mx.nd.from_torch(torch.randint(0, 10, (8,)))
    tr = Trainer(net.collect_params(), 'adam', {'learning_rate': 1e-3})
    loss_fn = SoftmaxCrossEntropyLoss()
    for _ in range(3):
        with autograd.record():
            L = loss_fn(net(X), Y)
        L.backward()
        tr.step(8)
    ref = net(X).asnumpy()

    with tempfile.TemporaryDirectory() as d:
        # 1. reference .params + symbol.json checkpoint
        net.export(os.path.join(d, 'model'))
        from mxnet_amd.gluon.block import SymbolBlock
        loaded = SymbolBlock.imports(os.path.join(d, 'model-symbol.json'),
                                     ['data'],
                                     os.path.join(d, 'model-0000.params'))
        np.testing.assert_allclose(loaded(X).asnumpy(), ref,
                                   rtol=1e-4, atol=1e-5)
        print('checkpoint round-trip OK')

        # 2. ONNX (self-contained protobuf writer, no onnx package)
        params = mx.nd.load(os.path.join(d, 'model-0000.params'))
        mxonnx.export_model(os.path.join(d, 'model-symbol.json'), params,
                            [(8, 3, 32, 32)],
                            onnx_file=os.path.join(d, 'model.onnx'))
        net2 = mxonnx.import_to_gluon(os.path.join(d, 'model.onnx'))
        np.testing.assert_allclose(net2(X).asnumpy(), ref,
                                   rtol=1e-3, atol=1e-4)
        print('ONNX round-trip OK')

    # 3. int8: swap Dense layers (conv stays fp: NCHW here; the NHWC conv
    # path quantizes too, see contrib.quantization.QuantizedConv2D)
    swapped = quantize_net(net)
    print(f'int8 swap: {len(swapped)} layer(s) quantized')


if __name__ == '__main__':
    main()
