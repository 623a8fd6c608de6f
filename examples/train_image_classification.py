#!/usr/bin/env python3
"""Image-classification training loop (reference
example/image-classification/train_imagenet.py shape, synthetic data).

Single GPU:  python examples/train_image_classification.py --model resnet50_v1
Multi GPU:   python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
               --master-addr 127.0.0.1 examples/train_image_classification.py
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--model', default='resnet50_v1')
    p.add_argument('--batch-size', type=int, default=128)
    p.add_argument('--image-size', type=int, default=224)
    p.add_argument('--epochs', type=int, default=1)
    p.add_argument('--iters-per-epoch', type=int, default=20)
    p.add_argument('--lr', type=float, default=0.1)
    p.add_argument('--dtype', default='float16')
    p.add_argument('--save-prefix', default='')
    args = p.parse_args()

    world = int(os.environ.get('WORLD_SIZE', 1))
    rank = int(os.environ.get('RANK', 0))
    local_rank = int(os.environ.get('LOCAL_RANK', 0))
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(local_rank)
    else:
        args.batch_size, args.image_size = 4, 64
        args.dtype = 'float32'

    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer
    from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
    from mxnet_amd.gluon.metric import Accuracy
    from mxnet_amd.gluon.model_zoo import vision
    from mxnet_amd import lr_scheduler

    ctx = mx.gpu(local_rank) if on_gpu else mx.cpu()
    net = getattr(vision, args.model)(classes=1000, layout='NHWC')
    net.initialize(ctx=ctx)
    net.cast(args.dtype)

    sched = lr_scheduler.CosineScheduler(
        max_update=args.epochs * args.iters_per_epoch,
        base_lr=args.lr * world, final_lr=0.0) \
        if hasattr(lr_scheduler, 'CosineScheduler') else None
    trainer = Trainer(net.collect_params(), 'sgd',
                      {'learning_rate': args.lr * world, 'momentum': 0.9,
                       'wd': 1e-4, 'multi_precision': True,
                       'lr_scheduler': sched},
                      kvstore='dist_device_sync' if world > 1 else None)
    loss_fn = SoftmaxCrossEntropyLoss()
    metric = Accuracy()

    B, S = args.batch_size, args.image_size
    dev = torch.device('cuda', local_rank) if on_gpu else torch.device('cpu')
    tdt = {'float16': torch.float16, 'float32': torch.float32}[args.dtype]

    for epoch in range(args.epochs):
        metric.reset()
        t0 = time.time()
        for it in range(args.iters_per_epoch):
            # synthetic batch (no dataset access in this environment)
            x = mx.nd.from_torch(torch.randn(B, S, S, 3, device=dev,
                                             dtype=tdt))
            y = mx.nd.from_torch(torch.randint(0, 1000, (B,), device=dev))
            with autograd.record():
                out = net(x)
                L = loss_fn(out, y)
            L.backward()
            trainer.step(B)
            metric.update(y, out)
        if on_gpu:
            torch.cuda.synchronize()
        dt = time.time() - t0
        if rank == 0:
            name, acc = metric.get()
            print(f'epoch {epoch}: {B * world * args.iters_per_epoch / dt:.1f} '
                  f'img/s, loss {float(L.mean().asnumpy()):.3f}, '
                  f'{name} {acc:.4f}')
    if args.save_prefix and rank == 0:
        net.save_parameters(f'{args.save_prefix}-{args.model}.params')
        sym_f, par_f = net.export(f'{args.save_prefix}-{args.model}')
        print('exported', sym_f, par_f)


if __name__ == '__main__':
    main()
