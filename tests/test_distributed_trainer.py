"""Multi-process data-parallel Trainer correctness over gloo (CPU stand-in
for the RCCL path the bench uses on MI355X; world_size=2, 127.0.0.1).

Checks the DP invariant the 8-GPU bench relies on: after a step with
per-rank batches, every rank holds identical parameters, equal to a
single-process step on the concatenated batch.
"""
import os
import subprocess
import sys

import numpy as np
import torch

_WORKER = '''
import os, sys
import numpy as np
import torch
import mxnet_amd as mx
from mxnet_amd import autograd
from mxnet_amd.gluon import Trainer, nn
from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss

rank = int(os.environ['RANK'])
torch.manual_seed(7)  # same init on both ranks (broadcast also enforces)
net = nn.Dense(4, in_units=6)
net.initialize()
loss_fn = SoftmaxCrossEntropyLoss()
tr = Trainer(net.collect_params(), 'sgd',
             {'learning_rate': 0.5, 'momentum': 0.9},
             kvstore='dist_sync')

# deterministic full batch, each rank takes its shard
torch.manual_seed(123)
X = torch.randn(8, 6)
Y = torch.randint(0, 4, (8,))
xs = mx.nd.from_torch(X[rank * 4:(rank + 1) * 4])
ys = mx.nd.from_torch(Y[rank * 4:(rank + 1) * 4])
for _ in range(3):  # step 1 = sync path, steps 2-3 = overlap hooks
    with autograd.record():
        out = net(xs)
        L = loss_fn(out, ys)
    L.backward()
    tr.step(4)  # per-rank batch size; grads averaged over workers
w = net.weight.data().asnumpy()
np.save(os.environ['OUT_PREFIX'] + f'_r{rank}.npy', w)
print('STEP_OK', rank)
'''


def test_dp_trainer_matches_single_process(tmp_path):
    script = tmp_path / 'worker.py'
    script.write_text(_WORKER)
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    import socket
    sock = socket.socket()
    sock.bind(('127.0.0.1', 0))
    port = str(sock.getsockname()[1])
    sock.close()
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': port,
                'OUT_PREFIX': str(tmp_path / 'w'),
                'PYTHONPATH': repo_root + os.pathsep + env.get('PYTHONPATH', '')})
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank), WORLD_SIZE='2', LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for p in procs:
        out, _ = p.communicate(timeout=180)
        assert p.returncode == 0, out.decode()

    w0 = np.load(tmp_path / 'w_r0.npy')
    w1 = np.load(tmp_path / 'w_r1.npy')
    np.testing.assert_allclose(w0, w1, rtol=1e-6)

    # single-process oracle on the full batch
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import Trainer, nn
    from mxnet_amd.gluon.loss import SoftmaxCrossEntropyLoss
    torch.manual_seed(7)
    net = nn.Dense(4, in_units=6)
    net.initialize()
    torch.manual_seed(123)
    X = torch.randn(8, 6)
    Y = torch.randint(0, 4, (8,))
    loss_fn = SoftmaxCrossEntropyLoss()
    tr = Trainer(net.collect_params(), 'sgd',
                 {'learning_rate': 0.5, 'momentum': 0.9}, kvstore=None)
    for _ in range(3):
        with autograd.record():
            L = loss_fn(net(mx.nd.from_torch(X)), mx.nd.from_torch(Y))
        L.backward()
        tr.step(8)
    np.testing.assert_allclose(w0, net.weight.data().asnumpy(), rtol=1e-5,
                               atol=1e-6)


_SYNCBN_WORKER = '''
import os
import numpy as np
import torch
import mxnet_amd as mx
from mxnet_amd import autograd
from mxnet_amd.gluon import nn

rank = int(os.environ['RANK'])
import torch.distributed as dist
dist.init_process_group('gloo', rank=rank,
                        world_size=int(os.environ['WORLD_SIZE']))
torch.manual_seed(3)
bn = nn.SyncBatchNorm(in_channels=3)
bn.initialize()
torch.manual_seed(11)
X = torch.randn(8, 3, 5, 5)
W = torch.randn(8, 3, 5, 5)  # fixed weights make dy vary per element
x = mx.nd.from_torch(X[rank * 4:(rank + 1) * 4].clone())
with autograd.record():
    y = bn(x)
    L = mx.nd.from_torch((y.handle * W[rank * 4:(rank + 1) * 4]).sum())
L.backward()
pre = os.environ['OUT_PREFIX']
np.save(pre + f'_y{rank}.npy', y.asnumpy())
np.save(pre + f'_gg{rank}.npy', bn.gamma.grad().asnumpy())
np.save(pre + f'_gb{rank}.npy', bn.beta.grad().asnumpy())
np.save(pre + f'_rm{rank}.npy', bn.running_mean.data().asnumpy())
np.save(pre + f'_rv{rank}.npy', bn.running_var.data().asnumpy())
print('SYNCBN_OK', rank)
'''


def test_sync_batchnorm_matches_full_batch(tmp_path):
    """SyncBN over 2 ranks (half-batch each) == plain BN on the full batch:
    outputs, gamma/beta grads, and running statistics all agree."""
    script = tmp_path / 'sbn.py'
    script.write_text(_SYNCBN_WORKER)
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    import socket
    sock = socket.socket()
    sock.bind(('127.0.0.1', 0))
    port = str(sock.getsockname()[1])
    sock.close()
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': port,
                'OUT_PREFIX': str(tmp_path / 's'),
                'PYTHONPATH': repo_root + os.pathsep + env.get('PYTHONPATH', '')})
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank), WORLD_SIZE='2', LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for p in procs:
        out, _ = p.communicate(timeout=180)
        assert p.returncode == 0, out.decode()

    # single-process oracle: plain BN over the concatenated batch
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon import nn
    torch.manual_seed(3)
    bn = nn.BatchNorm(in_channels=3)
    bn.initialize()
    torch.manual_seed(11)
    X = torch.randn(8, 3, 5, 5)
    W = torch.randn(8, 3, 5, 5)
    x = mx.nd.from_torch(X.clone())
    with autograd.record():
        y = bn(x)
        L = mx.nd.from_torch((y.handle * W).sum())
    L.backward()

    y_ref = y.asnumpy()
    for rank in range(2):
        yr = np.load(tmp_path / f's_y{rank}.npy')
        np.testing.assert_allclose(yr, y_ref[rank * 4:(rank + 1) * 4],
                                   rtol=1e-4, atol=1e-4)
        np.testing.assert_allclose(np.load(tmp_path / f's_gg{rank}.npy'),
                                   bn.gamma.grad().asnumpy(),
                                   rtol=1e-4, atol=1e-3)
        np.testing.assert_allclose(np.load(tmp_path / f's_gb{rank}.npy'),
                                   bn.beta.grad().asnumpy(),
                                   rtol=1e-4, atol=1e-3)
        np.testing.assert_allclose(np.load(tmp_path / f's_rm{rank}.npy'),
                                   bn.running_mean.data().asnumpy(),
                                   rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(np.load(tmp_path / f's_rv{rank}.npy'),
                                   bn.running_var.data().asnumpy(),
                                   rtol=1e-4, atol=1e-4)


def test_native_distributed_trainer_cpu():
    """2-process data-parallel training on the NATIVE runtime (gloo
    bridge): ranks see different data, gradients all-reduce-average, and
    parameters stay bit-identical across ranks."""
    import subprocess
    import sys
    import os
    script = r'''
import os
os.environ['MXNET_NATIVE_RUNTIME'] = '1'
import numpy as np
import mxnet_amd as mx
from mxnet_amd import autograd
from mxnet_amd.gluon import nn, Trainer
rank = int(os.environ['RANK'])
net = nn.Dense(3)
net.initialize()
x = mx.nd.array(np.full((2, 4), float(rank + 1), dtype='float32'))
net(x)
for k, p in net.collect_params().items():
    rs = np.random.RandomState(sum(ord(c) for c in k) % 997)
    p.set_data(mx.nd.array(rs.randn(*p.shape).astype('float32') * 0.3))
tr = Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.1},
             kvstore='dist_sync')
for _ in range(3):
    with autograd.record():
        L = (net(x) ** 2).sum()
    L.backward()
    tr.step(2)
w = net.weight.data(mx.cpu()).asnumpy()
print('WHASH', rank, float(np.abs(w).sum()))
'''
    env_base = dict(os.environ,
                    MASTER_ADDR='127.0.0.1', MASTER_PORT='29754',
                    WORLD_SIZE='2')
    procs = []
    for r in range(2):
        env = dict(env_base, RANK=str(r), LOCAL_RANK=str(r))
        procs.append(subprocess.Popen(
            [sys.executable, '-c', script], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=240)
        outs.append(out.decode())
    hashes = {}
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f'rank {r} failed:\n{out}'
        for line in out.splitlines():
            if line.startswith('WHASH'):
                hashes[r] = float(line.split()[2])
    assert len(hashes) == 2
    assert abs(hashes[0] - hashes[1]) < 1e-6, hashes
