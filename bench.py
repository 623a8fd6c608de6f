#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 v1.5 fp16 training, images/sec (whole node).

Matches BASELINE.json: "images/sec (whole node) ResNet-50 v1.5 fp16 at
1/2/4/8 MI355X" — Gluon model (layout NHWC, native gfx950 MFMA kernels),
SGD momentum with fp32 master weights, synthetic ImageNet-shaped data,
random-init weights.  Weak scaling: fixed 256 images per GPU.

Single GPU:   python bench.py --gpus 1 --steps 20 --warmup 5
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                --master-addr 127.0.0.1 bench.py --gpus N ...
(one process per GPU over RCCL/xGMI; gradient sync = DistKVStore bucketed
async all-reduce overlapping backward — mxnet_amd/parallel/kvstore.py)
"""
import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=20)
    p.add_argument('--warmup', type=int, default=5)
    p.add_argument('--batch-size', type=int, default=256,
                   help='per-GPU batch size')
    p.add_argument('--image-size', type=int, default=224)
    p.add_argument('--model', default='resnet50_v1')
    p.add_argument('--dtype', default='float16')
    p.add_argument('--rec', default='',
                   help='RecordIO file: feed batches through the C++ '
                        'image pipeline (decode threads + copy-stream '
                        'upload + on-GPU cast) instead of a fixed '
                        'synthetic buffer')
    p.add_argument('--runtime', choices=['native', 'torch'],
                   default=os.environ.get('MXNET_BENCH_RUNTIME', 'native'),
                   help='native = the own C++ runtime (pooled HIP storage, '
                        'threaded engine, own tape, RCCL binding) — the '
                        'default; torch = torch-tensor frontend over the '
                        'same CDNA4 kernel library')
    p.add_argument('--native', action='store_true',
                   help='(alias for --runtime native)')
    return p.parse_args()


def main():
    args = parse_args()
    world = int(os.environ.get('WORLD_SIZE', 1))
    rank = int(os.environ.get('RANK', 0))
    local_rank = int(os.environ.get('LOCAL_RANK', 0))
    distributed = world > 1

    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(local_rank)

    # --native (or MXNET_NATIVE_RUNTIME=1): the whole step runs on the
    # own C++ runtime — pooled HIP storage, threaded engine streams, own
    # autograd tape, registry-dispatched CDNA4 kernels; torch only
    # supplies the host process (no torch tensors, no torch.autograd)
    native = args.native or args.runtime == 'native' or \
        os.environ.get('MXNET_NATIVE_RUNTIME', '0') == '1'
    # native+distributed runs the own RCCL binding (NativeDistKVStore);
    # pre-flight the communicator before building the model so a
    # bootstrap failure degrades to the torch frontend instead of dying
    if native and distributed and on_gpu:
        ok = 1
        try:
            from mxnet_amd import _core as _c0
            _c0.rccl_init(world, rank, local_rank)
        except Exception as e:
            print(f'# native RCCL preflight failed ({e})', file=sys.stderr)
            ok = 0
        # consensus over a host-side gloo group: every rank must take
        # the SAME runtime or the collectives deadlock
        import torch.distributed as dist
        if not dist.is_initialized():
            dist.init_process_group('gloo')
        flag = torch.tensor([ok], dtype=torch.int64)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN)
        if int(flag.item()) == 0:
            if ok:
                print('# a peer rank failed RCCL preflight; '
                      'falling back to the torch frontend together',
                      file=sys.stderr)
            native = False

    import mxnet_amd as mx
    if native:
        from mxnet_amd.base import set_native
        set_native(True)
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer
    from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
    from mxnet_amd.gluon.model_zoo import vision

    ctx = mx.gpu(local_rank) if on_gpu else mx.cpu()
    if not on_gpu:
        # CPU smoke config (the driver benches on MI355X; this path just
        # proves the script runs) — reported config stays truthful below.
        args.batch_size = min(args.batch_size, 2)
        args.image_size = min(args.image_size, 64)

    B, S = args.batch_size, args.image_size
    dtype = args.dtype if on_gpu else 'float32'

    net = getattr(vision, args.model)(layout='NHWC', classes=1000)
    net.initialize(ctx=ctx)
    net.cast(dtype)

    kv = 'dist_device_sync' if distributed else None
    trainer = Trainer(net.collect_params(), 'sgd',
                      {'learning_rate': 0.1 * world, 'momentum': 0.9,
                       'wd': 1e-4, 'multi_precision': True},
                      kvstore=kv)
    loss_fn = SoftmaxCrossEntropyLoss()

    torch.manual_seed(1234 + rank)
    dev = torch.device('cuda', local_rank) if on_gpu else torch.device('cpu')
    tdt = {'float16': torch.float16, 'bfloat16': torch.bfloat16,
           'float32': torch.float32}[dtype]
    if native:
        import numpy as _np
        rs = _np.random.RandomState(1234 + rank)
        gctx = mx.gpu(local_rank) if on_gpu else mx.cpu()
        x = mx.nd.array(rs.randn(B, S, S, 3).astype('float32'),
                        ctx=gctx).astype(dtype)
        label = mx.nd.array(rs.randint(0, 1000, (B,)).astype('int64'),
                            ctx=gctx)
    else:
        x = mx.nd.from_torch(torch.randn(B, S, S, 3, device=dev, dtype=tdt))
        label = mx.nd.from_torch(torch.randint(0, 1000, (B,), device=dev))

    feeder = None
    if args.rec:
        # data pipeline: C++ threads decode/augment the NEXT batch on the
        # host while the GPU runs the CURRENT step; upload rides the copy
        # stream, the cast into the (fixed) network input buffer is a
        # compute-queue op so it interleaves with graph replays in order.
        from mxnet_amd import io as mxio, _core
        rec_it = mxio.ImageRecordIter(args.rec, B, (S, S, 3), shuffle=True,
                                      rand_crop=True, rand_mirror=True,
                                      preprocess_threads=0, seed=rank)
        if native:
            # decode threads write straight into a PINNED host buffer
            # (own storage manager, hipHostMalloc pool); upload rides the
            # engine copy stream; the cast lands in the fixed network
            # input buffer on the compute queue
            host_dev = 3 if on_gpu else 1  # pinned on GPU boxes
            pinned = _core.NDArray([B, S, S, 3], host_dev, 0, 3)  # u8
            plabels = _core.NDArray([B], host_dev, 0, 0)          # f32
            staging = mx.nd.empty((B, S, S, 3), ctx=ctx, dtype='uint8') \
                if on_gpu else None

            def feeder():
                got = rec_it._it.next_into(pinned.data_ptr,
                                           plabels.data_ptr)
                if got < B:
                    rec_it.reset()
                    rec_it._it.next_into(pinned.data_ptr, plabels.data_ptr)
                if on_gpu:
                    pinned.copyto(staging._h)
                    src_h = staging._h
                else:
                    src_h = pinned
                _core.invoke_into('cast', [src_h], [x._h], {})
        else:
            def feeder():
                got, data_np, _labels = rec_it.next_raw()
                if got < B:
                    rec_it.reset()
                    got, data_np, _labels = rec_it.next_raw()
                t = torch.from_numpy(data_np)
                with torch.no_grad():
                    x._t.copy_(t.to(x._t.device, non_blocking=True)
                               .to(x._t.dtype))

    def step():
        if feeder is not None:
            feeder()
        with autograd.record():
            out = net(x)
            L = loss_fn(out, label)
        L.backward()
        trainer.step(B)
        return L

    if distributed:
        import torch.distributed as dist
        if not dist.is_initialized():
            # host-side gloo group for the timing barrier/max-reduce —
            # the native runtime's RCCL binding owns the GPU collectives
            # and must not share a second device communicator
            dist.init_process_group('gloo')

    for _ in range(args.warmup):
        step()

    # whole-step hipGraph capture (fwd+bwd+optimizer): removes the
    # launch gaps between the ~300 kernels of a step.  Same kernels,
    # same math -- lr/wd are constant during the timed window.  RCCL
    # capture on multi-rank is untested on this pool, so the default is
    # single-process only (MXNET_BENCH_HIPGRAPH=1 forces, =0 disables).
    graph = None
    env_g = os.environ.get('MXNET_BENCH_HIPGRAPH', '')
    want_graph = env_g == '1' or env_g != '0'
    if on_gpu and want_graph and native:
        # Native runtime: capture the engine compute stream.  Two graphs:
        # G1 = forward+backward, G2 = fused optimizer updates + grad
        # zeroing; the RCCL all-reduces (engine comm stream) run eagerly
        # between the replays — for ResNet-50 the whole gradient set is
        # ~50 MB, well under a millisecond over xGMI, so overlap is not
        # the constraint the launch gaps are.
        from mxnet_amd import _core
        try:
            mx.nd.waitall()
            _core.begin_capture(local_rank)
            with autograd.record():
                out = net(x)
                L = loss_fn(out, label)
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
                                   grad_handles if distributed else [])
                if distributed:
                    trainer._allreduce_grads()
                # read-deps on the grads order G2 behind the comm-stream
                # all-reduces (engine events; no host sync)
                _core.launch_graph(local_rank, g2,
                                   grad_handles if distributed else [])
            replay_step()
            mx.nd.waitall()
            graph = ('native', replay_step)
        except Exception as e:
            print(f'# native hipgraph capture unavailable: {e}',
                  file=sys.stderr)
            graph = None
    elif on_gpu and want_graph and not distributed:
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
        except Exception as e:  # fall back to eager launches
            print(f'# hipgraph capture unavailable: {e}', file=sys.stderr)
            graph = None

    if native:
        mx.nd.waitall()
    if on_gpu:
        torch.cuda.synchronize()
    if distributed:
        dist.barrier()
    t0 = time.perf_counter()
    if native and isinstance(graph, tuple):
        for _ in range(args.steps):
            if feeder is not None:
                feeder()
            graph[1]()
        mx.nd.waitall()
    else:
        for _ in range(args.steps):
            graph.replay() if graph is not None else step()
        if native:
            mx.nd.waitall()
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=dev if on_gpu else None)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # numerics check inside the bench box (round-1 verdict ask): one
    # eager step after the timed window must produce a finite loss
    final_L = step()
    import numpy as _np
    final_loss = float(_np.asarray(final_L.asnumpy()).mean()) \
        if final_L is not None else None
    if final_loss is not None:
        import math as _math
        assert _math.isfinite(final_loss), \
            f'non-finite loss after bench: {final_loss}'

    n_gpus = world if on_gpu else args.gpus
    total_images = B * world * args.steps
    ips = total_images / elapsed
    if rank == 0:
        baseline = 363.69  # reference's best published ResNet-50 train img/s
        result = {
            'metric': 'images/sec (whole node) ResNet-50 v1.5 fp16',
            'value': round(ips, 2),
            'unit': 'images/sec',
            'n_gpus': n_gpus,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(elapsed / args.steps * 1000, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': round(ips / baseline, 3),
            'dtype': dtype,
            'data': ('recordio:' + os.path.basename(args.rec))
                    if args.rec else 'synthetic',
            'runtime': 'native' if native else 'torch-frontend',
            'final_loss': round(final_loss, 4) if final_loss is not None
                          else None,
            'config': {
                'model': 'resnet50_v1.5',
                'global_batch': B * world,
                'image_size': S,
                'parallelism': f'dp{world}',
                'kvstore': kv or 'none',
                'layout': 'NHWC',
            },
        }
        print(json.dumps(result))


if __name__ == '__main__':
    main()
