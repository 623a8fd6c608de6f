"""AMP — automatic mixed precision (reference python/mxnet/amp/amp.py).

MI355X-native: fp16/bf16 storage with fp32 MFMA accumulation is the
kernels' native mode, so "casting" means running the model parameters in
the target dtype (net.cast) plus a dynamic LossScaler whose finiteness
check is the native multi-tensor all_finite kernel
(reference all_finite.cu:33-66).
"""
import torch

from .loss_scaler import LossScaler
from ..ndarray.ndarray import NDArray
from . import lists  # noqa: F401

_amp_initialized = False
_target_dtype = 'float16'


def init(target_dtype='float16', target_precision_ops=None,
         conditional_fp32_ops=None, fp32_ops=None):
    global _amp_initialized, _target_dtype
    _amp_initialized = True
    _target_dtype = target_dtype


def init_trainer(trainer):
    """Attach a dynamic loss scaler to the trainer."""
    trainer._amp_loss_scaler = LossScaler()
    return trainer


def scale_loss(loss, trainer):
    """Context manager: scale the loss, unscale gradients at step time."""
    class _ScaleCtx:
        def __enter__(self):
            scaler = getattr(trainer, '_amp_loss_scaler', None)
            if scaler is None:
                trainer._amp_loss_scaler = scaler = LossScaler()
            self.scaler = scaler
            trainer._set_scale(1.0 / scaler.loss_scale)
            def _scale(l):
                if getattr(l, 'is_native', False):
                    return l * float(scaler.loss_scale)
                return NDArray(l._t * scaler.loss_scale)
            if isinstance(loss, (list, tuple)):
                return [_scale(l) for l in loss]
            return _scale(loss)

        def __exit__(self, *a):
            pass

    return _ScaleCtx()


def unscale(trainer):
    scaler = getattr(trainer, '_amp_loss_scaler', None)
    if scaler is None:
        return
    inv = 1.0 / scaler.loss_scale
    with torch.no_grad():
        for p in trainer._params:
            for g in p.list_grad():
                if getattr(g, 'is_native', False):
                    (g * inv).copyto(g)
                else:
                    g._t.mul_(inv)
    trainer._set_scale(1.0)


def all_finite(arrays):
    """Multi-tensor finiteness check (native kernel on GPU)."""
    arrays = [a for a in arrays if a is not None]
    if arrays and isinstance(arrays[0], NDArray) and arrays[0].is_native:
        from .. import _core
        out = _core.invoke('multi_all_finite',
                           [a._h for a in arrays], {})[0]
        return bool(int(out.asnumpy()[0]))
    ts = [a._t if isinstance(a, NDArray) else a for a in arrays]
    if not ts:
        return True
    if ts[0].is_cuda:
        from ..ops.dispatch import hipops
        ext = hipops()
        if ext is not None and hasattr(ext, 'multi_all_finite'):
            return bool(ext.multi_all_finite(ts))
    ok = True
    for t in ts:
        ok = ok and bool(torch.isfinite(t.float()).all())
    return ok


def convert_model(model, target_dtype='float16'):
    model.cast(target_dtype)
    return model


def convert_hybrid_block(block, target_dtype='float16'):
    block.cast(target_dtype)
    return block
