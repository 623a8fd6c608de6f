"""mx.np.linalg (reference src/operator/numpy/linalg — rocSOLVER-class ops
run through torch.linalg which dispatches to rocSOLVER/hipBLAS on ROCm)."""
import torch

from ..ndarray.ndarray import NDArray


def _t(x):
    return x._t if isinstance(x, NDArray) else x


def norm(x, ord=None, axis=None, keepdims=False):
    return NDArray(torch.linalg.norm(_t(x).float(), ord=ord, dim=axis,
                                     keepdim=keepdims))


def svd(a):
    u, s, vh = torch.linalg.svd(_t(a), full_matrices=False)
    return NDArray(u), NDArray(s), NDArray(vh)


def qr(a):
    q, r = torch.linalg.qr(_t(a))
    return NDArray(q), NDArray(r)


def cholesky(a):
    return NDArray(torch.linalg.cholesky(_t(a)))


def inv(a):
    return NDArray(torch.linalg.inv(_t(a)))


def pinv(a, rcond=1e-15):
    return NDArray(torch.linalg.pinv(_t(a), rtol=rcond))


def det(a):
    return NDArray(torch.linalg.det(_t(a)))


def slogdet(a):
    s, l = torch.linalg.slogdet(_t(a))
    return NDArray(s), NDArray(l)


def solve(a, b):
    return NDArray(torch.linalg.solve(_t(a), _t(b)))


def lstsq(a, b, rcond='warn'):
    out = torch.linalg.lstsq(_t(a), _t(b))
    return NDArray(out.solution), NDArray(out.residuals), out.rank, NDArray(out.singular_values)


def eig(a):
    w, v = torch.linalg.eig(_t(a))
    return NDArray(w), NDArray(v)


def eigh(a, UPLO='L'):
    w, v = torch.linalg.eigh(_t(a), UPLO=UPLO)
    return NDArray(w), NDArray(v)


def eigvals(a):
    return NDArray(torch.linalg.eigvals(_t(a)))


def matrix_rank(a, tol=None):
    return NDArray(torch.linalg.matrix_rank(_t(a), tol=tol))


def tensorsolve(a, b, axes=None):
    raise NotImplementedError


def tensorinv(a, ind=2):
    return NDArray(torch.linalg.tensorinv(_t(a), ind=ind))


def eig(a):
    """(reference np_eig op)."""
    import torch
    from ..ndarray.ndarray import NDArray
    w, v = torch.linalg.eig(_t(a))
    return NDArray(w), NDArray(v)


def eigh(a, UPLO='L'):
    import torch
    from ..ndarray.ndarray import NDArray
    w, v = torch.linalg.eigh(_t(a), UPLO=UPLO)
    return NDArray(w), NDArray(v)


def eigvals(a):
    import torch
    from ..ndarray.ndarray import NDArray
    return NDArray(torch.linalg.eigvals(_t(a)))


def lstsq(a, b, rcond='warn'):
    """(reference np_lstsq)."""
    import torch
    from ..ndarray.ndarray import NDArray
    rc = None if rcond in ('warn', None) else rcond
    res = torch.linalg.lstsq(_t(a), _t(b), rcond=rc)
    return (NDArray(res.solution), NDArray(res.residuals),
            int(res.rank) if res.rank.numel() == 1 else NDArray(res.rank),
            NDArray(res.singular_values))


def matrix_rank(a, tol=None):
    import torch
    from ..ndarray.ndarray import NDArray
    return NDArray(torch.linalg.matrix_rank(_t(a), tol=tol))


def tensorinv(a, ind=2):
    import torch
    from ..ndarray.ndarray import NDArray
    return NDArray(torch.linalg.tensorinv(_t(a), ind=ind))
