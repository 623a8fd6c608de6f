#!/usr/bin/env python3
"""2-layer LSTM (PTB-style) hidden=1024 benchmark (BASELINE config 5).

tokens/sec: embedding -> 2xLSTM -> FC decoder over vocab, fp16 on GPU.
The LSTM step runs the fused gfx950 lstm_cell kernel + MFMA gate GEMMs.
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
    p.add_argument('--batch-size', type=int, default=128)
    p.add_argument('--seq-len', type=int, default=35)
    p.add_argument('--hidden', type=int, default=1024)
    p.add_argument('--vocab', type=int, default=10000)
    args = p.parse_args()

    on_gpu = torch.cuda.is_available()
    if not on_gpu:
        args.batch_size, args.hidden, args.vocab = 4, 64, 200

    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer, nn, rnn
    from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
    from mxnet_amd.gluon.block import Block

    class PTBModel(Block):
        def __init__(self, vocab, hidden, **kw):
            super().__init__(**kw)
            self.embed = nn.Embedding(vocab, hidden)
            self.lstm = rnn.LSTM(hidden_size=hidden, num_layers=2)
            self.decoder = nn.Dense(vocab, flatten=False)

        def forward(self, x):
            e = self.embed(x)  # [T, N, H]
            out = self.lstm(e)
            return self.decoder(out)

    ctx = mx.gpu(0) if on_gpu else mx.cpu()
    dtype = 'float16' if on_gpu else 'float32'
    net = PTBModel(args.vocab, args.hidden)
    net.initialize(ctx=ctx)
    net.cast(dtype)
    trainer = Trainer(net.collect_params(), 'sgd',
                      {'learning_rate': 1.0, 'momentum': 0.0,
                       'multi_precision': True}, kvstore=None)
    loss_fn = SoftmaxCrossEntropyLoss()

    T, N = args.seq_len, args.batch_size
    dev = torch.device('cuda', 0) if on_gpu else torch.device('cpu')
    x = mx.nd.from_torch(torch.randint(0, args.vocab, (T, N), device=dev))
    y = mx.nd.from_torch(torch.randint(0, args.vocab, (T, N), device=dev))

    def step():
        with autograd.record():
            out = net(x)
            L = loss_fn(out, y)
        L.backward()
        trainer.step(N)

    for _ in range(args.warmup):
        step()

    # whole-step hipGraph capture (same pattern as bench.py): the LSTM
    # step is launch-bound, so replay removes the per-kernel gaps.
    graph = None
    if on_gpu and os.environ.get('MXNET_BENCH_HIPGRAPH', '1') != '0':
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
    t0 = time.perf_counter()
    for _ in range(args.steps):
        graph.replay() if graph is not None else step()
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        'metric': 'tokens/sec 2-layer LSTM h1024 (PTB-style)',
        'value': round(T * N * args.steps / dt, 2), 'unit': 'tokens/sec',
        'n_gpus': 1, 'steps': args.steps, 'warmup': args.warmup,
        'ms_per_step': round(dt / args.steps * 1e3, 3),
        'higher_is_better': True, 'scaling': 'weak', 'vs_baseline': None,
        'dtype': dtype, 'data': 'synthetic',
        'config': {'model': 'lstm_2x1024', 'global_batch': N,
                   'seq_len': T, 'parallelism': 'dp1'}}))


if __name__ == '__main__':
    main()
