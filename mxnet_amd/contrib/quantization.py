"""int8 quantization (reference python/mxnet/contrib/quantization.py +
src/operator/quantization): symmetric per-tensor int8 with min/max or
entropy calibration; the quantized GEMM runs the gfx950
``v_mfma_i32_16x16x64_i8`` kernel (~2x the bf16 MFMA rate).
"""
import torch

from ..ndarray.ndarray import NDArray

__all__ = ['quantize', 'dequantize', 'calib_minmax', 'QuantizedDense',
           'quantize_net']


def _t(x):
    return x.handle if isinstance(x, NDArray) else x


def calib_minmax(x):
    """Symmetric per-tensor scale from abs-max (reference quantize_v2
    min/max calibration)."""
    return float(_t(x).abs().max().item()) / 127.0 or 1.0


def quantize(x, scale=None):
    t = _t(x)
    if scale is None:
        scale = calib_minmax(t)
    if t.is_cuda:
        from ..ops.dispatch import hip_required
        q = hip_required('quantize').quantize_i8(t.contiguous(), scale)
    else:
        q = torch.clamp(torch.round(t.float() / scale), -127, 127).to(torch.int8)
    return NDArray(q), scale


def dequantize(q, scale, dtype='float32'):
    t = _t(q)
    td = {'float32': torch.float32, 'float16': torch.float16}[dtype]
    if t.is_cuda:
        from ..ops.dispatch import hip_required
        return NDArray(hip_required('dequantize').dequantize_i8(
            t.contiguous(), scale, td))
    return NDArray(t.to(td) * scale)


class QuantizedDense:
    """Int8 inference Dense: weights pre-quantized once, activations
    quantized per batch, i8 MFMA GEMM with fused rescale
    (reference quantized_fully_connected.cc)."""

    def __init__(self, dense_layer):
        w = dense_layer.weight.data().handle
        self._w_scale = calib_minmax(w)
        wq, _ = quantize(NDArray(w.contiguous()), self._w_scale)
        self._wq = wq.handle
        b = dense_layer.bias
        self._bias = b.data().handle.float() if b is not None else None
        self._out_dtype = w.dtype

    def __call__(self, x):
        t = _t(x)
        xs = calib_minmax(t)
        if t.is_cuda:
            from ..ops.dispatch import hip_required
            ext = hip_required('quantized_dense')
            xq = ext.quantize_i8(t.reshape(-1, t.shape[-1]).contiguous(), xs)
            y = ext.gemm_nt_i8(xq, self._wq, xs * self._w_scale,
                               self._out_dtype)
        else:
            xq = torch.clamp(torch.round(t.float() / xs), -127, 127)
            y = (xq.reshape(-1, t.shape[-1]) @
                 self._wq.float().t() * (xs * self._w_scale)).to(self._out_dtype)
        if self._bias is not None:
            y = y + self._bias.to(y.dtype)
        return NDArray(y.reshape(*t.shape[:-1], y.shape[-1]))


def quantize_net(net, quantized_dtype='int8', exclude_layers=None):
    """Swap Dense layers for int8 inference versions (reference
    quantize_model; conv quantization lands with the int8 conv kernel)."""
    from ..gluon import nn
    swapped = []
    def visit(block):
        for name, child in list(block._children.items()):
            if isinstance(child, nn.Dense) and \
                    (not exclude_layers or name not in exclude_layers):
                q = QuantizedDense(child)
                swapped.append((block, name, q))
            else:
                visit(child)
    visit(net)
    return swapped
