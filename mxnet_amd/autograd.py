"""Autograd — MXNet-style recording semantics over torch.autograd.

Reference parity: python/mxnet/autograd.py + Imperative::RecordOp/Backward
(src/imperative/imperative.cc:204,387).  The MI355X design does not keep a
separate tape: every op in mxnet_amd.ops is a torch.autograd.Function whose
backward launches our HIP kernels, so ``record()`` maps to enabling grad
mode and ``backward()`` to torch.autograd.backward — the gradient graph is
executed asynchronously on the device's HIP streams exactly like forward.
"""
import threading

import torch

from . import _core
from .ndarray.ndarray import NDArray

_STATE = threading.local()


def _state():
    if not hasattr(_STATE, 'recording'):
        _STATE.recording = False
        _STATE.training = False
    return _STATE


def is_recording():
    return _state().recording


def is_training():
    return _state().training


def set_recording(is_rec):
    prev = _state().recording
    _STATE.recording = is_rec
    return prev


def set_training(train_mode):
    prev = _state().training
    _STATE.training = train_mode
    return prev


class _RecordingStateScope:
    def __init__(self, is_record, train_mode):
        self._enter_is_record = is_record
        self._enter_train_mode = train_mode
        self._prev_is_record = None
        self._prev_train_mode = None
        self._grad_ctx = None

    def __enter__(self):
        if self._enter_is_record is not None:
            self._prev_is_record = set_recording(self._enter_is_record)
            self._prev_core_record = _core.set_recording(
                self._enter_is_record)
            self._grad_ctx = torch.enable_grad() if self._enter_is_record \
                else torch.no_grad()
            self._grad_ctx.__enter__()
        if self._enter_train_mode is not None:
            self._prev_train_mode = set_training(self._enter_train_mode)
            self._prev_core_train = _core.set_training(
                self._enter_train_mode)
        return self

    def __exit__(self, *args):
        if self._enter_is_record is not None:
            set_recording(self._prev_is_record)
            _core.set_recording(self._prev_core_record)
            self._grad_ctx.__exit__(*args)
        if self._enter_train_mode is not None:
            set_training(self._prev_train_mode)
            _core.set_training(self._prev_core_train)


def record(train_mode=True):
    """Scope in which ops are recorded for gradient computation."""
    return _RecordingStateScope(True, train_mode)


def pause(train_mode=False):
    return _RecordingStateScope(False, train_mode)


def train_mode():
    return _RecordingStateScope(None, True)


def predict_mode():
    return _RecordingStateScope(None, False)


def mark_variables(variables, gradients, grad_reqs='write'):
    if isinstance(variables, NDArray):
        variables, gradients = [variables], [gradients]
    if isinstance(grad_reqs, str):
        grad_reqs = [grad_reqs] * len(variables)
    for v, g, r in zip(variables, gradients, grad_reqs):
        if v.is_native:
            v._native_grad = g
            _core.mark_variable(v._h, g._h, 2 if r == 'add' else 1)
            continue
        v._t.requires_grad_(True)
        v._t.grad = g._t


def backward(heads, head_grads=None, retain_graph=False, train_mode=True):
    """Compute gradients of heads w.r.t. marked variables.

    grad_req='add' semantics are native to torch (grads accumulate);
    'write' semantics are provided by the Trainer zeroing between steps
    (reference: grads are written, then cleared by the user/trainer).
    """
    if isinstance(heads, NDArray):
        heads = [heads]
    if heads and heads[0].is_native:
        # own C++ tape (Imperative::Backward — torch.autograd is not
        # involved on the native runtime)
        hs = [h._h for h in heads]
        if head_grads is None:
            gs = []
        else:
            if isinstance(head_grads, NDArray):
                head_grads = [head_grads]
            gs = [g._h for g in head_grads if g is not None]
        _core.backward(hs, gs, retain_graph)
        return
    tensors = [h._t for h in heads]
    if head_grads is None:
        grads = [torch.ones_like(t) for t in tensors]
    else:
        if isinstance(head_grads, NDArray):
            head_grads = [head_grads]
        grads = [g._t if g is not None else torch.ones_like(t)
                 for g, t in zip(head_grads, tensors)]
    torch.autograd.backward(tensors, grads, retain_graph=retain_graph)


def grad(heads, variables, head_grads=None, retain_graph=None,
         create_graph=False, train_mode=True):
    if isinstance(heads, NDArray):
        heads = [heads]
    if isinstance(variables, NDArray):
        variables = [variables]
    tensors = [h._t for h in heads]
    if head_grads is None:
        gr = [torch.ones_like(t) for t in tensors]
    else:
        if isinstance(head_grads, NDArray):
            head_grads = [head_grads]
        gr = [g._t for g in head_grads]
    outs = torch.autograd.grad(tensors, [v._t for v in variables], gr,
                               retain_graph=retain_graph,
                               create_graph=create_graph)
    return [NDArray(o) for o in outs]


class Function:
    """User-defined differentiable function (reference autograd.Function)."""

    def __init__(self):
        self._saved = None

    def save_for_backward(self, *args):
        self._saved = args

    @property
    def saved_tensors(self):
        return self._saved

    def forward(self, *inputs):
        raise NotImplementedError

    def backward(self, *output_grads):
        raise NotImplementedError

    def __call__(self, *inputs):
        outer = self

        class _Wrapped(torch.autograd.Function):
            @staticmethod
            def forward(fctx, *ts):
                nds = [NDArray(t) for t in ts]
                with pause():
                    out = outer.forward(*nds)
                fctx._outer = outer
                if isinstance(out, NDArray):
                    return out._t
                return tuple(o._t for o in out)

            @staticmethod
            def backward(fctx, *gts):
                gnds = [NDArray(g.contiguous()) for g in gts]
                with pause():
                    out = fctx._outer.backward(*gnds)
                if isinstance(out, NDArray):
                    return out._t
                return tuple(o._t if o is not None else None for o in out)

        ts = [i._t for i in inputs]
        out = _Wrapped.apply(*ts)
        if isinstance(out, torch.Tensor):
            return NDArray(out)
        return tuple(NDArray(o) for o in out)
