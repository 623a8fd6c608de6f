"""KVStore tests: local/device single-process + multi-process gloo
(reference tests/python/unittest/test_kvstore.py + nightly dist tests)."""
import os
import subprocess
import sys
import textwrap

import numpy as np
import pytest

import mxnet_amd as mx
from mxnet_amd import nd
from mxnet_amd.parallel import kvstore as kv


def test_local_push_pull():
    store = kv.create('local')
    store.init(3, nd.ones((2, 2)))
    store.push(3, nd.ones((2, 2)) * 4)
    out = nd.zeros((2, 2))
    store.pull(3, out=out)
    assert np.allclose(out.asnumpy(), 4)


def test_local_multi_value_reduce():
    store = kv.create('local')
    store.init('w', nd.zeros((3,)))
    vals = [nd.ones((3,)) * i for i in range(1, 4)]  # 1+2+3 = 6
    store.push('w', vals)
    out = nd.zeros((3,))
    store.pull('w', out=out)
    assert np.allclose(out.asnumpy(), 6)


def test_pushpull_fused():
    store = kv.create('device')
    store.init(0, nd.zeros((4,)))
    g = nd.ones((4,)) * 2
    out = nd.zeros((4,))
    store.pushpull(0, g, out=out)
    assert np.allclose(out.asnumpy(), 2)


def test_list_keys():
    store = kv.create('local')
    store.init([1, 2], [nd.ones((2,)), nd.ones((2,)) * 2])
    outs = [nd.zeros((2,)), nd.zeros((2,))]
    store.pull([1, 2], out=outs)
    assert np.allclose(outs[0].asnumpy(), 1)
    assert np.allclose(outs[1].asnumpy(), 2)


def test_optimizer_on_kvstore():
    """update_on_kvstore path (reference kvstore set_optimizer)."""
    from mxnet_amd import optimizer as opt
    store = kv.create('local')
    store.set_optimizer(opt.SGD(learning_rate=0.1))
    w = nd.ones((2,))
    store.init(0, w)
    store.push(0, nd.ones((2,)))   # grad = 1 -> w -= 0.1
    out = nd.zeros((2,))
    store.pull(0, out=out)
    assert np.allclose(out.asnumpy(), 0.9, atol=1e-6)


_DIST_SCRIPT = textwrap.dedent('''
    import os
    import numpy as np
    import mxnet_amd as mx
    from mxnet_amd import nd
    from mxnet_amd.parallel import kvstore as kv

    store = kv.create('dist_sync')
    rank, nw = store.rank, store.num_workers
    assert nw == 2, nw
    g = nd.ones((8,)) * (rank + 1)      # ranks push 1 and 2 -> sum 3
    store.pushpull('w', g)
    assert np.allclose(g.asnumpy(), 3), g.asnumpy()
    # broadcast: rank0 value wins
    b = nd.ones((4,)) * (10 if rank == 0 else -1)
    store.broadcast('b', b, b)
    assert np.allclose(b.asnumpy(), 10), b.asnumpy()
    print('RANK_OK', rank)
''')


def test_dist_kvstore_gloo_two_procs(tmp_path):
    """2-process all-reduce over gloo on localhost (the CPU stand-in for
    RCCL; reference nightly dist_sync_kvstore.py pattern)."""
    script = tmp_path / 'worker.py'
    script.write_text(_DIST_SCRIPT)
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': '29511',
                'PYTHONPATH': repo_root + os.pathsep + env.get('PYTHONPATH', '')})
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank), WORLD_SIZE='2', LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
        assert p.returncode == 0, out.decode()
    assert any('RANK_OK 0' in o for o in outs)
    assert any('RANK_OK 1' in o for o in outs)


_PS_WORKER = '''
import os
import numpy as np
import torch
import torch.distributed as dist
import mxnet_amd as mx
from mxnet_amd import autograd
from mxnet_amd.gluon import Trainer, nn
from mxnet_amd.parallel import kvstore as kvs

rank = int(os.environ['RANK'])
world = int(os.environ['WORLD_SIZE'])
kv = kvs.create('dist_async')
if kv.is_server:
    from mxnet_amd import optimizer as opt
    kv.set_optimizer(opt.create('sgd', learning_rate=0.5))
    kv.run_server()
else:
    torch.manual_seed(4)
    net = nn.Dense(3, in_units=5, use_bias=False)
    net.initialize()
    tr = Trainer(net.collect_params(), 'sgd', {'learning_rate': 0.5},
                 kvstore=kv)
    X = torch.full((2, 5), 0.1 * (rank + 1))
    x = mx.nd.from_torch(X)
    with autograd.record():
        L = mx.nd.from_torch(net(x).handle.sum())
    L.backward()
    tr.step(1)
    kv.barrier_workers()       # both pushes applied before final pull
    kv.pull(0, net.weight.data())
    np.save(os.environ['OUT_PREFIX'] + f'_w{rank}.npy',
            net.weight.data().asnumpy())
    kv.stop()
print('PS_OK', rank)
'''


def test_dist_async_parameter_server(tmp_path):
    """dist_async: 2 workers + 1 server rank; server applies SGD per
    push; after both pushes the pulled weights equal w0 - lr*(g1+g2)
    (reference KVStoreDistServer async mode)."""
    import subprocess, sys, socket
    script = tmp_path / 'ps.py'
    script.write_text(_PS_WORKER)
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sock = socket.socket(); sock.bind(('127.0.0.1', 0))
    port = str(sock.getsockname()[1]); sock.close()
    env = dict(os.environ)
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': port,
                'OUT_PREFIX': str(tmp_path / 'ps'),
                'PYTHONPATH': repo_root + os.pathsep +
                env.get('PYTHONPATH', '')})
    procs = []
    for rank in range(3):
        e = dict(env, RANK=str(rank), WORLD_SIZE='3')
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
        assert p.returncode == 0, out.decode()

    import numpy as np
    import torch
    w0 = np.load(tmp_path / 'ps_w0.npy')
    w1 = np.load(tmp_path / 'ps_w1.npy')
    np.testing.assert_allclose(w0, w1, rtol=1e-6)
    # oracle: dL/dW for sum(x @ W^T) is ones(3,1) @ x_mean... grad per
    # worker = column-broadcast of X rows summed: each row of dW =
    # sum of X rows = 2*0.1*(rank+1) per entry
    torch.manual_seed(4)
    import mxnet_amd as mx
    from mxnet_amd.gluon import nn as gnn
    net = gnn.Dense(3, in_units=5, use_bias=False)
    net.initialize()
    init_w = net.weight.data().asnumpy()
    g_total = np.full((3, 5), 2 * 0.1 * 1) + np.full((3, 5), 2 * 0.1 * 2)
    np.testing.assert_allclose(w0, init_w - 0.5 * g_total, rtol=1e-4,
                               atol=1e-5)


_SYNC_PS_WORKER = '''
import os
import numpy as np
import torch
from mxnet_amd.ndarray.ndarray import NDArray
from mxnet_amd.parallel import kvstore as kvs

rank = int(os.environ['RANK'])
kv = kvs.create('dist_sync_ps')
if kv.is_server:
    from mxnet_amd import optimizer as opt
    kv.set_optimizer(opt.create('sgd', learning_rate=0.5))
    kv.run_server()
else:
    # two keys -> sharded across the two server ranks
    w0 = NDArray(torch.full((4,), 10.0))
    w1 = NDArray(torch.full((6,), 20.0))
    kv.init(0, w0)
    kv.init(1, w1)
    g0 = NDArray(torch.full((4,), float(rank + 1)))
    g1 = NDArray(torch.full((6,), 2.0 * (rank + 1)))
    kv.push(0, g0)
    kv.push(1, g1)
    kv.pull(0, w0)   # blocks until BOTH workers pushed (sync barrier)
    kv.pull(1, w1)
    np.save(os.environ['OUT_PREFIX'] + f'_s{rank}.npy',
            np.concatenate([w0.asnumpy(), w1.asnumpy()]))
    kv.barrier_workers()
    kv.stop()   # every worker tells every server (live-count protocol)
print('SYNC_PS_OK', rank)
'''


def test_dist_sync_parameter_server(tmp_path):
    """dist_sync_ps: 2 workers + 2 servers (key-sharded).  The server
    aggregates one push per worker per key, applies SGD ONCE on the
    merged gradient, and only then releases the queued pulls — per-key
    global barrier semantics (reference ApplyUpdates,
    kvstore_dist_server.h:346-365)."""
    import subprocess, sys, socket
    script = tmp_path / 'sync_ps.py'
    script.write_text(_SYNC_PS_WORKER)
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sock = socket.socket(); sock.bind(('127.0.0.1', 0))
    port = str(sock.getsockname()[1]); sock.close()
    env = dict(os.environ)
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': port,
                'MXNET_PS_NSERVERS': '2',
                'OUT_PREFIX': str(tmp_path / 'ps'),
                'PYTHONPATH': repo_root + os.pathsep +
                env.get('PYTHONPATH', '')})
    procs = []
    for rank in range(4):
        e = dict(env, RANK=str(rank), WORLD_SIZE='4')
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
        assert p.returncode == 0, out.decode()
    import numpy as np
    s0 = np.load(tmp_path / 'ps_s0.npy')
    s1 = np.load(tmp_path / 'ps_s1.npy')
    np.testing.assert_allclose(s0, s1, rtol=1e-6)
    # merged g0 = (1+2) = 3 -> w0' = 10 - 0.5*3 = 8.5
    # merged g1 = (2+4) = 6 -> w1' = 20 - 0.5*6 = 17
    np.testing.assert_allclose(s0[:4], 8.5, rtol=1e-6)
    np.testing.assert_allclose(s0[4:], 17.0, rtol=1e-6)


def test_native_dist_kvstore_gloo_bridge():
    """NativeDistKVStore on CPU hosts bridges native arrays onto gloo
    (RCCL needs a GPU): 2 processes all-reduce-average and broadcast."""
    import subprocess
    import sys
    import os
    script = r'''
import os
os.environ['MXNET_NATIVE_RUNTIME'] = '1'
import numpy as np
import mxnet_amd as mx
from mxnet_amd.parallel.kvstore import NativeDistKVStore
rank = int(os.environ['RANK'])
kv = NativeDistKVStore('dist_sync')
v = mx.nd.array(np.full((8,), float(rank + 1), dtype='float32'))
kv.pushpull(0, v, out=v)
np.testing.assert_allclose(v.asnumpy(), np.full((8,), 1.5))
b = mx.nd.array(np.full((4,), float(10 * (rank + 1)), dtype='float32'))
kv.broadcast(1, b, b)
np.testing.assert_allclose(b.asnumpy(), np.full((4,), 10.0))
print('GLOO_BRIDGE_OK', rank)
'''
    procs = []
    env_base = dict(os.environ,
                    MASTER_ADDR='127.0.0.1', MASTER_PORT='29753',
                    WORLD_SIZE='2')
    for r in range(2):
        env = dict(env_base, RANK=str(r), LOCAL_RANK=str(r))
        procs.append(subprocess.Popen(
            [sys.executable, '-c', script], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f'rank {r} failed:\n{out}'
        assert f'GLOO_BRIDGE_OK {r}' in out
