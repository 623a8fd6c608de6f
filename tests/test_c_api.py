"""C ABI over the native runtime, driven through raw ctypes — proves the
handle-based interface (reference include/mxnet/c_api.h model) works
without any Python-binding help."""
import ctypes
import glob
import os

import numpy as np
import pytest

_SO = glob.glob(os.path.join(os.path.dirname(__file__), '..', 'mxnet_amd',
                             '_core*.so'))[0]


@pytest.fixture(scope='module')
def lib():
    import mxnet_amd  # ensure the module (and its registry) is loaded
    L = ctypes.CDLL(_SO)
    L.MXGetLastError.restype = ctypes.c_char_p
    c = ctypes
    L.MXNDArrayCreate.argtypes = [c.POINTER(c.c_int64), c.c_int, c.c_int,
                                  c.c_int, c.c_int, c.POINTER(c.c_void_p)]
    L.MXNDArraySyncCopyFromCPU.argtypes = [c.c_void_p, c.c_void_p,
                                           c.c_size_t]
    L.MXNDArraySyncCopyToCPU.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t]
    L.MXNDArrayFree.argtypes = [c.c_void_p]
    L.MXNDArrayGetShape.argtypes = [c.c_void_p, c.POINTER(c.c_int),
                                    c.POINTER(c.POINTER(c.c_int64))]
    L.MXImperativeInvoke.argtypes = [
        c.c_char_p, c.c_int, c.POINTER(c.c_void_p), c.POINTER(c.c_int),
        c.POINTER(c.POINTER(c.c_void_p)), c.c_int,
        c.POINTER(c.c_char_p), c.POINTER(c.c_char_p)]
    L.MXAutogradMarkVariables.argtypes = [c.c_int, c.POINTER(c.c_void_p),
                                          c.POINTER(c.c_void_p),
                                          c.POINTER(c.c_int)]
    L.MXAutogradBackward.argtypes = [c.c_int, c.POINTER(c.c_void_p),
                                     c.POINTER(c.c_void_p), c.c_int]
    L.MXNDArraySave.argtypes = [c.c_char_p, c.c_int, c.POINTER(c.c_void_p),
                                c.POINTER(c.c_char_p)]
    L.MXNDArrayLoad.argtypes = [c.c_char_p, c.POINTER(c.c_int),
                                c.POINTER(c.POINTER(c.c_void_p)),
                                c.POINTER(c.POINTER(c.c_char_p))]
    return L


def _create(lib, arr):
    arr = np.ascontiguousarray(arr, dtype='float32')
    shape = (ctypes.c_int64 * arr.ndim)(*arr.shape)
    h = ctypes.c_void_p()
    assert lib.MXNDArrayCreate(shape, arr.ndim, 1, 0, 0,
                               ctypes.byref(h)) == 0, \
        lib.MXGetLastError()
    assert lib.MXNDArraySyncCopyFromCPU(
        h, arr.ctypes.data_as(ctypes.c_void_p), arr.nbytes) == 0
    return h, arr


def _tonumpy(lib, h, shape):
    out = np.empty(shape, dtype='float32')
    assert lib.MXNDArraySyncCopyToCPU(
        h, out.ctypes.data_as(ctypes.c_void_p), out.nbytes) == 0, \
        lib.MXGetLastError()
    return out


def test_c_api_create_invoke(lib):
    a, av = _create(lib, [[1, 2], [3, 4]])
    b, bv = _create(lib, [[10, 20], [30, 40]])
    nout = ctypes.c_int()
    outs = ctypes.POINTER(ctypes.c_void_p)()
    rc = lib.MXImperativeInvoke(b'elemwise_add', 2,
                                (ctypes.c_void_p * 2)(a, b),
                                ctypes.byref(nout), ctypes.byref(outs),
                                0, None, None)
    assert rc == 0, lib.MXGetLastError()
    assert nout.value == 1
    got = _tonumpy(lib, outs[0], (2, 2))
    np.testing.assert_array_equal(got, av + bv)
    # shape/dtype introspection
    nd = ctypes.c_int()
    sp = ctypes.POINTER(ctypes.c_int64)()
    assert lib.MXNDArrayGetShape(outs[0], ctypes.byref(nd),
                                 ctypes.byref(sp)) == 0
    assert nd.value == 2 and sp[0] == 2 and sp[1] == 2
    lib.MXNDArrayFree(a)
    lib.MXNDArrayFree(b)


def test_c_api_error_reporting(lib):
    h = ctypes.c_void_p()
    nout = ctypes.c_int()
    outs = ctypes.POINTER(ctypes.c_void_p)()
    rc = lib.MXImperativeInvoke(b'no_such_op', 0, None, ctypes.byref(nout),
                                ctypes.byref(outs), 0, None, None)
    assert rc == -1
    assert b'no_such_op' in lib.MXGetLastError()


def test_c_api_list_ops(lib):
    n = ctypes.c_int()
    names = ctypes.POINTER(ctypes.c_char_p)()
    assert lib.MXListOps(ctypes.byref(n), ctypes.byref(names)) == 0
    ops = {names[i] for i in range(n.value)}
    assert b'Convolution' in ops and b'FullyConnected' in ops
    assert n.value > 80


def test_c_api_autograd(lib):
    x, xv = _create(lib, [1.0, -2.0, 3.0])
    g, _ = _create(lib, [0.0, 0.0, 0.0])
    assert lib.MXAutogradMarkVariables(
        1, (ctypes.c_void_p * 1)(x), (ctypes.c_void_p * 1)(g),
        (ctypes.c_int * 1)(1)) == 0
    prev = ctypes.c_int()
    lib.MXAutogradSetIsRecording(1, ctypes.byref(prev))
    nout = ctypes.c_int()
    outs = ctypes.POINTER(ctypes.c_void_p)()
    lib.MXImperativeInvoke(b'relu', 1, (ctypes.c_void_p * 1)(x),
                           ctypes.byref(nout), ctypes.byref(outs), 0,
                           None, None)
    y = ctypes.c_void_p(outs[0])
    outs2 = ctypes.POINTER(ctypes.c_void_p)()
    lib.MXImperativeInvoke(b'sum', 1, (ctypes.c_void_p * 1)(y),
                           ctypes.byref(nout), ctypes.byref(outs2), 0,
                           None, None)
    L = ctypes.c_void_p(outs2[0])
    lib.MXAutogradSetIsRecording(0, ctypes.byref(prev))
    assert lib.MXAutogradBackward(1, (ctypes.c_void_p * 1)(L), None, 0) \
        == 0, lib.MXGetLastError()
    lib.MXNDArrayWaitAll()
    gv = _tonumpy(lib, g, (3,))
    np.testing.assert_array_equal(gv, [1.0, 0.0, 1.0])


def test_c_api_save_load(lib, tmp_path):
    a, av = _create(lib, np.arange(6, dtype='float32').reshape(2, 3))
    fname = str(tmp_path / 'c.params').encode()
    assert lib.MXNDArraySave(fname, 1, (ctypes.c_void_p * 1)(a),
                             (ctypes.c_char_p * 1)(b'w')) == 0, \
        lib.MXGetLastError()
    cnt = ctypes.c_int()
    arrs = ctypes.POINTER(ctypes.c_void_p)()
    names = ctypes.POINTER(ctypes.c_char_p)()
    assert lib.MXNDArrayLoad(fname, ctypes.byref(cnt), ctypes.byref(arrs),
                             ctypes.byref(names)) == 0, lib.MXGetLastError()
    assert cnt.value == 1 and names[0] == b'w'
    got = _tonumpy(lib, arrs[0], (2, 3))
    np.testing.assert_array_equal(got, av)
    # and the python serialization reader accepts the C-written file
    from mxnet_amd.utils import serialization as ser
    loaded = ser.load_ndarrays(fname.decode())
    np.testing.assert_array_equal(loaded['w'].asnumpy(), av)


def test_load_external_op_library(tmp_path):
    """MXLoadLib (reference lib_api.h): dlopen a user .so, register its
    MXCustomOpDef ops into the native registry, invoke like built-ins."""
    import subprocess
    import numpy as np
    so = str(tmp_path / 'libmy_relu.so')
    subprocess.check_call(['g++', '-shared', '-fPIC', '-I', 'include',
                           'examples/lib_custom_op/my_relu.cc', '-o', so])
    import mxnet_amd as mx
    from mxnet_amd import _core
    from mxnet_amd.ndarray.ndarray import NDArray
    mx.library.load(so)
    x = _core.from_numpy(
        np.array([[-1.0, 2.0], [3.0, -4.0]], np.float32), 1, 0)
    y = NDArray(_core.invoke('my_relu', [x], {})[0])
    np.testing.assert_array_equal(y.asnumpy(), [[0, 2], [3, 0]])
    # loading a nonexistent library raises through MXGetLastError
    import pytest
    with pytest.raises(RuntimeError):
        mx.library.load(str(tmp_path / 'nope.so'))
