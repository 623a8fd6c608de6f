from . import serialization  # noqa: F401


def split_and_load(data, ctx_list, batch_axis=0, even_split=True):
    """Split a batch across contexts (reference gluon/utils.py)."""
    from ..ndarray.ndarray import NDArray
    if not isinstance(ctx_list, (list, tuple)):
        ctx_list = [ctx_list]
    if len(ctx_list) == 1:
        return [data.as_in_context(ctx_list[0])]
    n = data.shape[batch_axis]
    k = len(ctx_list)
    assert not even_split or n % k == 0, \
        f'batch {n} not divisible by {k} contexts'
    step = n // k
    slices = [data.slice_axis(batch_axis, i * step,
                              (i + 1) * step if i < k - 1 else n)
              for i in range(k)]
    return [s.as_in_context(c) for s, c in zip(slices, ctx_list)]


def clip_global_norm(arrays, max_norm, check_isfinite=True):
    """Reference gluon/utils.py clip_global_norm."""
    import math
    import torch
    total = 0.0
    for a in arrays:
        total += float(a._t.float().norm() ** 2)
    total = math.sqrt(total)
    if check_isfinite and not math.isfinite(total):
        import warnings
        warnings.warn('nan or inf in global norm')
        return total
    scale = max_norm / (total + 1e-8)
    if scale < 1.0:
        with torch.no_grad():
            for a in arrays:
                a._t.mul_(scale)
    return total
