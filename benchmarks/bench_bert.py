#!/usr/bin/env python3
"""BERT-base fp16 pretraining-step benchmark (BASELINE config 4).

samples/sec whole job; seq=128 synthetic tokens, random-init weights.
Launch multi-GPU exactly like bench.py (torch.distributed.run, RCCL).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--steps', type=int, default=20)
    p.add_argument('--warmup', type=int, default=5)
    p.add_argument('--batch-size', type=int, default=32)
    p.add_argument('--seq-len', type=int, default=128)
    p.add_argument('--dtype', default='float16')
    args = p.parse_args()

    world = int(os.environ.get('WORLD_SIZE', 1))
    rank = int(os.environ.get('RANK', 0))
    local_rank = int(os.environ.get('LOCAL_RANK', 0))
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(local_rank)
    else:
        args.batch_size, args.seq_len = 2, 32

    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer
    from mxnet_amd.ndarray.ndarray import NDArray
    from mxnet_amd.models.bert import bert_base, BERTModel
    from mxnet_amd.ops import nn as F

    ctx = mx.gpu(local_rank) if on_gpu else mx.cpu()
    dtype = args.dtype if on_gpu else 'float32'
    if on_gpu:
        net = bert_base()
    else:
        net = BERTModel(vocab_size=1000, units=64, hidden_size=128,
                        num_layers=2, num_heads=4)
    net.initialize(ctx=ctx)
    net.cast(dtype)
    # embeddings + LN params stay functional in fp16 (master weights in opt)
    trainer = Trainer(net.collect_params(), 'adam',
                      {'learning_rate': 1e-4, 'multi_precision': True},
                      kvstore='dist_device_sync' if world > 1 else None)

    B, S = args.batch_size, args.seq_len
    dev = torch.device('cuda', local_rank) if on_gpu else torch.device('cpu')
    vocab = 30522 if on_gpu else 1000
    torch.manual_seed(1 + rank)
    tokens = mx.nd.from_torch(torch.randint(0, vocab, (B, S), device=dev))
    types = mx.nd.from_torch(torch.zeros(B, S, dtype=torch.long, device=dev))
    mask = mx.nd.from_torch(torch.ones(B, S, dtype=torch.bool, device=dev))
    mlm_label = torch.randint(0, vocab, (B, S), device=dev)
    nsp_label = torch.randint(0, 2, (B,), device=dev)

    def step():
        with autograd.record():
            _, _, mlm, nsp = net(tokens, types, mask)
            l1 = F.softmax_cross_entropy(
                mlm.handle.reshape(-1, vocab), mlm_label.reshape(-1)).mean()
            l2 = F.softmax_cross_entropy(nsp.handle, nsp_label).mean()
            L = NDArray((l1 + l2).float())
        L.backward()
        trainer.step(B)

    if world > 1:
        import torch.distributed as dist
    for _ in range(args.warmup):
        step()

    # whole-step hipGraph capture (same pattern as bench.py)
    graph = None
    if on_gpu and world == 1 and \
            os.environ.get('MXNET_BENCH_HIPGRAPH', '1') != '0':
        try:
            torch.cuda.synchronize()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                step()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                step()
            g.replay()
            torch.cuda.synchronize()
            graph = g
        except Exception as e:
            print('# hipgraph capture unavailable:', e)
            graph = None

    if on_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        graph.replay() if graph is not None else step()
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([dt], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())
    if rank == 0:
        print(json.dumps({
            'metric': 'samples/sec BERT-base fp16 seq128 (whole node)',
            'value': round(B * world * args.steps / dt, 2),
            'unit': 'samples/sec', 'n_gpus': world, 'steps': args.steps,
            'warmup': args.warmup, 'ms_per_step': round(dt / args.steps * 1e3, 3),
            'higher_is_better': True, 'scaling': 'weak', 'vs_baseline': None,
            'dtype': dtype, 'data': 'synthetic',
            'config': {'model': 'bert_base', 'global_batch': B * world,
                       'seq_len': S, 'parallelism': f'dp{world}'}}))


if __name__ == '__main__':
    main()
