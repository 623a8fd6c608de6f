"""Focused hipGraph-capture repro for the native engine (GPU box)."""
import os
import sys

sys.path.insert(0, os.getcwd())
os.environ['MXNET_ENGINE_DEBUG'] = '1'
import numpy as np
from mxnet_amd import _core

rs = np.random.RandomState(5)
an = rs.randn(64, 64).astype('float32')
bn = rs.randn(64, 64).astype('float32')
a = _core.from_numpy(an, 2, 0)
b = _core.from_numpy(bn, 2, 0)
out = _core.invoke('elemwise_add', [a, b], {})[0]
_core.wait_all()
print('pre-capture ok:', np.allclose(out.asnumpy(), an + bn))

_core.begin_capture(0)
print('begin ok')
_core.invoke_into('_grad_add', [b], [out], {})
print('op pushed')
g = _core.end_capture(0)
print('end ok, exec=', hex(g))
_core.wait_all()
v0 = out.asnumpy()
print('post-capture value == a+b (capture must NOT execute):',
      np.allclose(v0, an + bn), 'max-dev', np.abs(v0 - (an + bn)).max())
for i in range(3):
    _core.launch_graph(0, g)
    _core.wait_all()
    vi = out.asnumpy()
    k = np.round((vi - an)[0, 0] / bn[0, 0], 3)
    print(f'after replay {i+1}: b-multiples at [0,0] = {k}, '
          f'match a+{i+2}b: {np.allclose(vi, an + (i + 2) * bn, rtol=1e-5)}')
print('DONE')
