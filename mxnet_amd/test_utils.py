"""Testing utilities (reference python/mxnet/test_utils.py).

The central tool is ``check_consistency``: run a Gluon block / op on
several (ctx, dtype) configurations and assert the outputs and gradients
agree with the highest-precision run — exactly how the GPU kernels here
are validated against the CPU fp32 oracle (tests/test_gpu_kernels.py).
"""
import numpy as np
import torch

from .context import cpu, gpu, current_context
from .ndarray.ndarray import NDArray, array

__all__ = ['default_context', 'assert_almost_equal', 'almost_equal',
           'same', 'rand_ndarray', 'rand_shape_2d', 'rand_shape_3d',
           'check_consistency', 'list_gpus', 'default_rtols', 'default_atols']

_DEFAULT_RTOL = {np.dtype(np.float16): 1e-2, np.dtype(np.float32): 1e-4,
                 np.dtype(np.float64): 1e-5}
_DEFAULT_ATOL = {np.dtype(np.float16): 1e-2, np.dtype(np.float32): 1e-5,
                 np.dtype(np.float64): 1e-7}


def default_rtols():
    return dict(_DEFAULT_RTOL)


def default_atols():
    return dict(_DEFAULT_ATOL)


def default_context():
    return current_context()


def list_gpus():
    if not torch.cuda.is_available():
        return []
    return list(range(torch.cuda.device_count()))


def _to_numpy(a):
    if isinstance(a, NDArray):
        return a.asnumpy()
    if isinstance(a, torch.Tensor):
        return a.detach().float().cpu().numpy()
    return np.asarray(a)


def same(a, b):
    return np.array_equal(_to_numpy(a), _to_numpy(b))


def almost_equal(a, b, rtol=None, atol=None, equal_nan=False):
    a, b = _to_numpy(a), _to_numpy(b)
    dt = np.promote_types(a.dtype, b.dtype)
    rtol = rtol if rtol is not None else _DEFAULT_RTOL.get(np.dtype(dt), 1e-5)
    atol = atol if atol is not None else _DEFAULT_ATOL.get(np.dtype(dt), 1e-7)
    return np.allclose(a, b, rtol=rtol, atol=atol, equal_nan=equal_nan)


def assert_almost_equal(a, b, rtol=None, atol=None, names=('a', 'b'),
                        equal_nan=False):
    a_np, b_np = _to_numpy(a), _to_numpy(b)
    if not almost_equal(a_np, b_np, rtol, atol, equal_nan):
        err = np.abs(a_np - b_np)
        rel = err / (np.abs(b_np) + 1e-12)
        raise AssertionError(
            f'{names[0]} != {names[1]}: max abs err {err.max():.6g}, '
            f'max rel err {rel.max():.6g}')


def rand_shape_2d(dim0=10, dim1=10):
    return (np.random.randint(1, dim0 + 1), np.random.randint(1, dim1 + 1))


def rand_shape_3d(dim0=10, dim1=10, dim2=10):
    return (np.random.randint(1, dim0 + 1), np.random.randint(1, dim1 + 1),
            np.random.randint(1, dim2 + 1))


def rand_ndarray(shape, ctx=None, dtype='float32'):
    data = torch.randn(*shape)
    nd = array(data, ctx=ctx, dtype=dtype)
    return nd


def check_consistency(fn, inputs, ctx_list=None, dtypes=None, rtol=None,
                      atol=None, grad=True):
    """Run ``fn(*inputs)`` on each (ctx, dtype) config; compare every
    output/gradient against the most precise config (reference
    test_utils.py:1490 check_consistency).

    fn: callable taking NDArrays and returning an NDArray.
    inputs: list of numpy arrays / torch tensors.
    """
    if ctx_list is None:
        ctx_list = [cpu()] + ([gpu(0)] if torch.cuda.is_available() else [])
    if dtypes is None:
        dtypes = ['float32'] + (['float16'] if torch.cuda.is_available() else [])
    configs = [(c, d) for c in ctx_list for d in dtypes
               if not (getattr(c, 'device_type', 'cpu') == 'cpu' and d == 'float16')]
    results = []
    for ctx, dtype in configs:
        nds = [array(torch.as_tensor(np.asarray(x)), ctx=ctx, dtype=dtype)
               for x in inputs]
        if grad:
            for nd in nds:
                nd.attach_grad()
            from . import autograd
            with autograd.record():
                out = fn(*nds)
            from .ndarray.ndarray import ones as _ones
            out.backward(_ones(tuple(out.shape), ctx=ctx,
                               dtype=str(out.dtype))
                         if out.is_native
                         else NDArray(torch.ones_like(out.handle)))
            grads = [nd.grad.asnumpy() for nd in nds]
        else:
            out = fn(*nds)
            grads = []
        results.append((dtype, out.asnumpy(), grads))
    # reference config = first (most precise)
    ref_dtype, ref_out, ref_grads = results[0]
    for dtype, out, grads in results[1:]:
        worst = np.dtype(np.float16) if 'float16' in (dtype, ref_dtype) \
            else np.dtype(np.float32)
        r = rtol if rtol is not None else _DEFAULT_RTOL[worst]
        a = atol if atol is not None else _DEFAULT_ATOL[worst]
        scale = max(np.abs(ref_out).max(), 1.0)
        assert np.allclose(out.astype(np.float64), ref_out.astype(np.float64),
                           rtol=r, atol=a * scale), f'output mismatch ({dtype})'
        for g, rg in zip(grads, ref_grads):
            gs = max(np.abs(rg).max(), 1.0)
            assert np.allclose(g.astype(np.float64), rg.astype(np.float64),
                               rtol=r * 4, atol=a * 4 * gs), \
                f'gradient mismatch ({dtype})'
    return results


def check_numeric_gradient(fn, inputs, eps=1e-3, rtol=2e-2, atol=2e-3,
                           ctx=None, dtype='float64'):
    """Finite-difference gradient check against autograd (reference
    test_utils.py:1043 check_numeric_gradient).

    ``fn`` takes NDArrays and returns one NDArray; its sum is the scalar
    objective.  Central differences are computed per input element and
    compared with the tape's gradients.  Works on either runtime (the
    native tape or torch.autograd, depending on the array backend).
    """
    from . import autograd
    from .ndarray import ndarray as _nd
    ctx = ctx or cpu()
    arrs = [array(np.asarray(x, dtype='float32'), ctx=ctx, dtype='float32')
            for x in inputs]
    for a in arrs:
        a.attach_grad()
    with autograd.record():
        out = fn(*arrs)
        loss = out.sum() if out.size > 1 else out
    loss.backward()
    analytic = [a.grad.asnumpy().astype('float64') for a in arrs]

    for i, x in enumerate(inputs):
        base = np.asarray(x, dtype='float64')
        num = np.zeros_like(base)
        flat = base.reshape(-1)
        nflat = num.reshape(-1)
        for j in range(flat.size):
            orig = flat[j]
            for sign in (+1, -1):
                flat[j] = orig + sign * eps
                probe = [array(np.asarray(b if k != i else base,
                                          dtype='float32'), ctx=ctx)
                         for k, b in enumerate(inputs)]
                # re-materialize the perturbed input
                probe[i] = array(base.astype('float32'), ctx=ctx)
                val = float(np.asarray(fn(*probe).asnumpy(),
                                       dtype='float64').sum())
                nflat[j] += sign * val / (2 * eps)
            flat[j] = orig
        assert_almost_equal(analytic[i], num, rtol=rtol, atol=atol,
                            names=(f'autograd_grad[{i}]',
                                   f'numeric_grad[{i}]'))
