#!/usr/bin/env python3
"""BERT-base fp16 pretraining-step benchmark (BASELINE config 4).

samples/sec whole job; seq=128 synthetic tokens, random-init weights.
Launch multi-GPU exactly like bench.py (torch.distributed.run, RCCL).
Default runtime is the native C++ engine (`--runtime torch` selects the
torch-tensor frontend over the same gfx950 kernels).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--steps', type=int, default=20)
    p.add_argument('--warmup', type=int, default=5)
    p.add_argument('--batch-size', type=int, default=32)
    p.add_argument('--seq-len', type=int, default=128)
    p.add_argument('--dtype', default='float16')
    p.add_argument('--runtime', choices=['native', 'torch'],
                   default='native')
    args = p.parse_args()

    world = int(os.environ.get('WORLD_SIZE', 1))
    rank = int(os.environ.get('RANK', 0))
    local_rank = int(os.environ.get('LOCAL_RANK', 0))
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(local_rank)
    else:
        args.batch_size, args.seq_len = 2, 32

    native = args.runtime == 'native'
    # native+distributed: preflight the RCCL bootstrap with a consensus
    # fallback (all ranks must take the same runtime — bench.py pattern)
    if native and world > 1 and on_gpu:
        ok = 1
        try:
            os.environ['MXNET_NATIVE_RUNTIME'] = '1'
            from mxnet_amd import _core as _c0
            _c0.rccl_init(world, rank, local_rank)
        except Exception as e:
            print(f'# native RCCL preflight failed ({e})', file=sys.stderr)
            ok = 0
        import torch.distributed as dist
        if not dist.is_initialized():
            dist.init_process_group('gloo')
        flag = torch.tensor([ok], dtype=torch.int64)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN)
        if int(flag.item()) == 0:
            native = False
    os.environ['MXNET_NATIVE_RUNTIME'] = '1' if native else '0'

    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer
    from mxnet_amd.ndarray.ndarray import NDArray
    from mxnet_amd.models.bert import bert_base, BERTModel
    from mxnet_amd.ndarray import ops as F

    ctx = mx.gpu(local_rank) if on_gpu else mx.cpu()
    dtype = args.dtype if on_gpu else 'float32'
    if on_gpu:
        net = bert_base()
    else:
        net = BERTModel(vocab_size=1000, units=64, hidden_size=128,
                        num_layers=2, num_heads=4)
    net.initialize(ctx=ctx)
    net.cast(dtype)
    # LN/embedding masters come from the optimizer's multi_precision
    trainer = Trainer(net.collect_params(), 'adam',
                      {'learning_rate': 1e-4, 'multi_precision': True},
                      kvstore='dist_device_sync' if world > 1 else None)

    B, S = args.batch_size, args.seq_len
    dev = torch.device('cuda', local_rank) if on_gpu else torch.device('cpu')
    vocab = 30522 if on_gpu else 1000
    rng = np.random.RandomState(1 + rank)
    torch.manual_seed(1 + rank)

    if native:
        tokens = mx.nd.array(rng.randint(0, vocab, (B, S)), ctx=ctx,
                             dtype='int64')
        types = mx.nd.array(np.zeros((B, S)), ctx=ctx, dtype='int64')
        mask = mx.nd.array(np.ones((B, S)), ctx=ctx, dtype=dtype)
        mlm_label = mx.nd.array(rng.randint(0, vocab, (B * S,)), ctx=ctx,
                                dtype='float32')
        nsp_label = mx.nd.array(rng.randint(0, 2, (B,)), ctx=ctx,
                                dtype='float32')

        def loss_fn(mlm, nsp):
            lp1 = F.log_softmax(mlm.reshape(-1, vocab))
            l1 = F.pick(lp1, mlm_label, axis=-1).mean() * -1.0
            lp2 = F.log_softmax(nsp)
            l2 = F.pick(lp2, nsp_label, axis=-1).mean() * -1.0
            return l1 + l2
    else:
        from mxnet_amd.ops import nn as Fnn
        tokens = mx.nd.from_torch(
            torch.randint(0, vocab, (B, S), device=dev))
        types = mx.nd.from_torch(
            torch.zeros(B, S, dtype=torch.long, device=dev))
        mask = mx.nd.from_torch(torch.ones(B, S, dtype=torch.bool,
                                           device=dev))
        mlm_t = torch.randint(0, vocab, (B * S,), device=dev)
        nsp_t = torch.randint(0, 2, (B,), device=dev)

        def loss_fn(mlm, nsp):
            l1 = Fnn.softmax_cross_entropy(
                mlm.handle.reshape(-1, vocab), mlm_t).mean()
            l2 = Fnn.softmax_cross_entropy(nsp.handle, nsp_t).mean()
            return NDArray((l1 + l2).float())

    def step():
        with autograd.record():
            _, _, mlm, nsp = net(tokens, types, mask)
            L = loss_fn(mlm, nsp)
        L.backward()
        trainer.step(B)

    if world > 1:
        import torch.distributed as dist
        if not dist.is_initialized():
            # host-side gloo group for the timing barrier/max-reduce —
            # the native runtime's RCCL binding owns the GPU collectives
            # and must not share a second device communicator
            dist.init_process_group('gloo')
    for _ in range(args.warmup):
        step()

    # whole-step hipGraph capture (same two-graph scheme as bench.py:
    # G1 fwd+bwd on the engine compute stream, eager RCCL between, G2
    # fused Adam updates with read-deps on the grad vars)
    graph = None
    want_graph = os.environ.get('MXNET_BENCH_HIPGRAPH', '1') != '0'
    if on_gpu and want_graph and native:
        from mxnet_amd import _core
        try:
            mx.nd.waitall()
            _core.begin_capture(local_rank)
            with autograd.record():
                _, _, mlm, nsp = net(tokens, types, mask)
                L = loss_fn(mlm, nsp)
            L.backward()
            g1 = _core.end_capture(local_rank)
            mx.nd.waitall()
            _core.begin_capture(local_rank)
            trainer._optimizer.rescale_grad = 1.0 / B
            trainer._update(False)
            g2 = _core.end_capture(local_rank)
            mx.nd.waitall()
            grad_handles = [p.list_grad()[0]._h for p in trainer._params]

            def replay_step():
                # G1 declares the grads it writes so the comm-stream
                # all-reduces order behind the REPLAYED backward (not a
                # stale pre-capture event)
                _core.launch_graph(local_rank, g1, [],
                                   grad_handles if world > 1 else [])
                if world > 1:
                    trainer._allreduce_grads()
                _core.launch_graph(local_rank, g2,
                                   grad_handles if world > 1 else [])
            replay_step()
            mx.nd.waitall()
            graph = ('native', replay_step)
        except Exception as e:
            print(f'# native hipgraph capture unavailable: {e}',
                  file=sys.stderr)
            graph = None
    elif on_gpu and want_graph and world == 1:
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
            print('# hipgraph capture unavailable:', e, file=sys.stderr)
            graph = None

    if native:
        mx.nd.waitall()
    if on_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        if graph is None:
            step()
        elif isinstance(graph, tuple):
            graph[1]()
        else:
            graph.replay()
    if native:
        mx.nd.waitall()
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([dt], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())
    # post-bench numerics check: one eager step, loss must be finite
    with autograd.record():
        _, _, mlm_f, nsp_f = net(tokens, types, mask)
        Lf = loss_fn(mlm_f, nsp_f)
    import math as _math
    final_loss = float(Lf.asnumpy())
    assert _math.isfinite(final_loss), f'non-finite loss: {final_loss}'
    if rank == 0:
        print(json.dumps({
            'metric': 'samples/sec BERT-base fp16 seq128 (whole node)',
            'value': round(B * world * args.steps / dt, 2),
            'unit': 'samples/sec', 'n_gpus': world, 'steps': args.steps,
            'warmup': args.warmup, 'ms_per_step': round(dt / args.steps * 1e3, 3),
            'higher_is_better': True, 'scaling': 'weak', 'vs_baseline': None,
            'dtype': dtype, 'data': 'synthetic', 'runtime': args.runtime,
            'final_loss': round(final_loss, 4),
            'config': {'model': 'bert_base', 'global_batch': B * world,
                       'seq_len': S, 'parallelism': f'dp{world}'}}))


if __name__ == '__main__':
    main()
