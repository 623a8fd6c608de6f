"""int8 quantization (reference python/mxnet/contrib/quantization.py +
src/operator/quantization): symmetric per-tensor int8 with min/max or
entropy calibration; the quantized GEMM runs the gfx950
``v_mfma_i32_16x16x64_i8`` kernel (~2x the bf16 MFMA rate).
"""
import torch

from ..ndarray.ndarray import NDArray

__all__ = ['quantize', 'dequantize', 'calib_minmax', 'QuantizedDense',
           'QuantizedConv2D', 'quantize_net']


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


class QuantizedConv2D:
    """Int8 inference conv (NHWC, groups=1): weight pre-quantized to
    [K, R*S*C] int8 rows; per batch the input is im2col'd (native
    kernel), quantized, and multiplied on the i8 MFMA GEMM with fused
    rescale (reference quantized_conv.cc — cuDNN int8 there; gfx950
    ``v_mfma_i32_16x16x64_i8`` here)."""

    def __init__(self, conv_layer):
        assert conv_layer._groups == 1, 'int8 conv: groups==1 only'
        assert conv_layer._layout == 'NHWC', 'int8 conv: NHWC only'
        w = conv_layer.weight.data().handle  # [K, R, S, C]
        self._K, self._R, self._S, self._C = w.shape
        self._stride = conv_layer._strides
        self._pad = conv_layer._padding
        self._dil = conv_layer._dilation
        self._w_scale = calib_minmax(w)
        w2 = w.reshape(self._K, -1).contiguous()
        if w.is_cuda:
            from ..ops.dispatch import hip_required
            self._wq = hip_required('quantized_conv').quantize_i8(
                w2, self._w_scale)
        else:
            self._wq = torch.clamp(torch.round(w2.float() / self._w_scale),
                                   -127, 127).to(torch.int8)
        b = conv_layer.bias
        self._bias = b.data().handle.float() if b is not None else None
        self._out_dtype = w.dtype
        self._act = conv_layer._act_type

    def __call__(self, x):
        t = _t(x)
        N, H, W, C = t.shape
        (sh, sw), (ph, pw), (dh, dw) = self._stride, self._pad, self._dil
        P = (H + 2 * ph - dh * (self._R - 1) - 1) // sh + 1
        Q = (W + 2 * pw - dw * (self._S - 1) - 1) // sw + 1
        xs = calib_minmax(t)
        if t.is_cuda:
            from ..ops.dispatch import hip_required
            ext = hip_required('quantized_conv')
            col = ext.im2col_nhwc(t.contiguous(), self._R, self._S,
                                  sh, sw, ph, pw, dh, dw)
            xq = ext.quantize_i8(col, xs)
            y = ext.gemm_nt_i8(xq, self._wq, xs * self._w_scale,
                               self._out_dtype)
        else:
            xn = t.float().permute(0, 3, 1, 2)
            col = torch.nn.functional.unfold(
                xn, (self._R, self._S), dilation=(dh, dw),
                padding=(ph, pw), stride=(sh, sw))  # [N, C*R*S, P*Q]
            # unfold is c-major [C,R,S]; our layout is [R,S,C]
            col = col.reshape(N, C, self._R * self._S, -1)                      .permute(0, 3, 2, 1).reshape(-1, self._R * self._S * C)
            xq = torch.clamp(torch.round(col / xs), -127, 127)
            y = (xq @ self._wq.float().t() * (xs * self._w_scale))                 .to(self._out_dtype)
        if self._bias is not None:
            y = y + self._bias.to(y.dtype)
        y = y.reshape(N, P, Q, self._K)
        if self._act == 'relu':
            y = torch.relu(y)
        return NDArray(y)


def quantize_net(net, quantized_dtype='int8', exclude_layers=None):
    """Swap Dense / NHWC Conv2D layers for int8 inference versions
    (reference quantize_model)."""
    from ..gluon import nn
    swapped = []
    def visit(block):
        for name, child in list(block._children.items()):
            if exclude_layers and name in exclude_layers:
                continue
            if isinstance(child, nn.Dense):
                swapped.append((block, name, QuantizedDense(child)))
            elif isinstance(child, nn.Conv2D) and child._groups == 1 and \
                    child._layout == 'NHWC':
                swapped.append((block, name, QuantizedConv2D(child)))
            else:
                visit(child)
    visit(net)
    return swapped
