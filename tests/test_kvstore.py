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
