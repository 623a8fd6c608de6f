#!/usr/bin/env python3
"""Export a trained Gluon model to the reference checkpoint pair
(model-symbol.json + model-0000.params) and serve it back through
SymbolBlock — the reference's deploy workflow (example/image-classification
+ SymbolBlock.imports), with hipGraph-captured scoring on GPU."""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import mxnet_amd as mx
from mxnet_amd.gluon import nn, SymbolBlock
from mxnet_amd.gluon.model_zoo import vision


def main():
    on_gpu = torch.cuda.is_available()
    ctx = mx.gpu(0) if on_gpu else mx.cpu()
    net = vision.resnet18_v1(classes=1000)
    net.initialize(ctx=ctx)
    dev0 = torch.device('cuda', 0) if on_gpu else torch.device('cpu')
    net(mx.nd.from_torch(torch.randn(1, 3, 64, 64, device=dev0)))  # shapes

    workdir = tempfile.mkdtemp()
    prefix = os.path.join(workdir, 'resnet18')
    sym_file, params_file = net.export(prefix)
    print('exported:', sym_file, params_file)

    served = SymbolBlock.imports(sym_file, ['data'], params_file, ctx=ctx)
    dev = torch.device('cuda', 0) if on_gpu else torch.device('cpu')
    x = mx.nd.from_torch(torch.randn(8, 3, 64, 64, device=dev))
    ref = net(x).asnumpy()
    out = served(x).asnumpy()
    import numpy as np
    np.testing.assert_allclose(out, ref, rtol=1e-3, atol=1e-4)
    print('SymbolBlock output matches the exporting net')

    served.hybridize(static_alloc=True, static_shape=True)
    for _ in range(3):
        served(x)
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(20):
        served(x)
    if on_gpu:
        torch.cuda.synchronize()
    print(f'serving: {8 * 20 / (time.time() - t0):.1f} img/s '
          f'({"hipGraph" if on_gpu else "cpu"})')


if __name__ == '__main__':
    main()
