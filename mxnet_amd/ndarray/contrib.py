"""Control-flow operators (reference src/operator/control_flow.cc:
``_foreach``, ``_while_loop``, ``_cond`` and python wrappers
python/mxnet/ndarray/contrib.py:foreach/while_loop/cond).

Imperative implementations over NDArray; autograd flows through the
underlying torch graph, so these are differentiable like the
reference's stateful control-flow ops.
"""
import torch

from .ndarray import NDArray

__all__ = ['foreach', 'while_loop', 'cond', 'isinf', 'isnan', 'isfinite']


def _as_list(x):
    return list(x) if isinstance(x, (list, tuple)) else [x]


def foreach(body, data, init_states):
    """Iterate ``body(step_data, states) -> (out, new_states)`` over the
    leading axis of ``data``; stacks per-step outputs
    (reference contrib.py:foreach / control_flow.cc ForeachOp)."""
    states = _as_list(init_states)
    data_list = _as_list(data)
    n = data_list[0].shape[0]
    outputs = []
    for i in range(n):
        step = [d[i] for d in data_list]
        out, states = body(step[0] if len(step) == 1 else step, states)
        outputs.append(_as_list(out))
    stacked = []
    for slot in zip(*outputs):
        stacked.append(NDArray(torch.stack([o.handle for o in slot])))
    out = stacked[0] if len(stacked) == 1 else stacked
    return out, states


def while_loop(cond_fn, func, loop_vars, max_iterations=None):
    """(reference contrib.py:while_loop): run ``func`` while ``cond_fn``
    holds; per-step outputs are stacked and padded to max_iterations."""
    loop_vars = _as_list(loop_vars)
    steps = 0
    outputs = []
    while bool(_scalar(cond_fn(*loop_vars))):
        if max_iterations is not None and steps >= max_iterations:
            break
        out, loop_vars = func(*loop_vars)
        loop_vars = _as_list(loop_vars)
        outputs.append(_as_list(out))
        steps += 1
    if not outputs:
        return [], loop_vars
    stacked = []
    for slot in zip(*outputs):
        st = torch.stack([o.handle for o in slot])
        if max_iterations is not None and steps < max_iterations:
            pad = torch.zeros((max_iterations - steps,) + tuple(st.shape[1:]),
                              dtype=st.dtype, device=st.device)
            st = torch.cat([st, pad])
        stacked.append(NDArray(st))
    out = stacked[0] if len(stacked) == 1 else stacked
    return out, loop_vars


def cond(pred, then_func, else_func):
    """(reference contrib.py:cond / CondOp)."""
    return then_func() if bool(_scalar(pred)) else else_func()


def _scalar(v):
    if isinstance(v, NDArray):
        return v.handle.item()
    if isinstance(v, torch.Tensor):
        return v.item()
    return v


def isinf(data):
    return NDArray(torch.isinf(data.handle))


def isnan(data):
    return NDArray(torch.isnan(data.handle))


def isfinite(data):
    return NDArray(torch.isfinite(data.handle))
