#!/usr/bin/env python3
"""Batched inference scoring, images/sec — parity with the reference's
example/image-classification/benchmark_score.py (the source of the
BASELINE.md inference tables).

Runs every model of the fp16 headline table at the published batch sizes.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def score(model_name, batch, steps, warmup, dtype, image_size=224):
    import mxnet_amd as mx
    from mxnet_amd.gluon.model_zoo import vision

    on_gpu = torch.cuda.is_available()
    ctx = mx.gpu(0) if on_gpu else mx.cpu()
    kwargs = {}
    if model_name.startswith('resnet') or model_name.startswith('vgg'):
        kwargs['layout'] = 'NHWC'
    net = getattr(vision, model_name)(**kwargs)
    net.initialize(ctx=ctx)
    net.cast(dtype)
    net.hybridize(static_alloc=True, static_shape=True)  # hipGraph capture
    dev = torch.device('cuda', 0) if on_gpu else torch.device('cpu')
    tdt = {'float16': torch.float16, 'float32': torch.float32}[dtype]
    shape = (batch, image_size, image_size, 3) if 'layout' in kwargs \
        else (batch, 3, image_size, image_size)
    x = mx.nd.from_torch(torch.randn(*shape, device=dev, dtype=tdt))
    for _ in range(warmup):
        net(x).handle
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        net(x).handle
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return batch * steps / dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--model', default='resnet50_v1')
    p.add_argument('--batch', type=int, default=0, help='0 = sweep table')
    p.add_argument('--steps', type=int, default=20)
    p.add_argument('--warmup', type=int, default=5)
    p.add_argument('--dtype', default='float16')
    args = p.parse_args()
    on_gpu = torch.cuda.is_available()
    if not on_gpu:
        args.dtype = 'float32'
    batches = [args.batch] if args.batch else ([1, 32, 128] if on_gpu else [1])
    for b in batches:
        ips = score(args.model, b, args.steps, args.warmup, args.dtype,
                    64 if not on_gpu else 224)
        print(json.dumps({'metric': f'inference images/sec {args.model}',
                          'batch': b, 'value': round(ips, 2),
                          'dtype': args.dtype, 'data': 'synthetic'}))


if __name__ == '__main__':
    main()
