"""int8 quantization (reference python/mxnet/contrib/quantization.py +
src/operator/quantization): symmetric per-tensor int8 with min/max or
entropy calibration; the quantized GEMM runs the gfx950
``v_mfma_i32_16x16x64_i8`` kernel (~2x the bf16 MFMA rate).
"""
import torch

from ..ndarray.ndarray import NDArray

__all__ = ['quantize', 'dequantize', 'calib_minmax', 'calib_entropy',
           'QuantizedDense', 'QuantizedConv2D', 'quantize_net']


def _t(x):
    return x.handle if isinstance(x, NDArray) else x


def calib_minmax(x):
    """Symmetric per-tensor scale from abs-max (reference quantize_v2
    min/max calibration)."""
    return float(_t(x).abs().max().item()) / 127.0 or 1.0


def _calib_sample(t, cap=65536):
    """Subsampled |activation| values for offline calibration (keeps
    memory bounded while the KL sweep sees the value distribution)."""
    import numpy as np
    a = np.abs(t.detach().float().reshape(-1).cpu().numpy())
    if a.size > cap:
        a = a[:: (a.size + cap - 1) // cap]
    return a


def calib_entropy(x, num_bins=2048, num_quantized_bins=255):
    """KL-divergence-optimal symmetric threshold (reference
    quantization.py:_get_optimal_threshold / calibrate.cc entropy
    mode): sweep candidate clip thresholds over an abs-value histogram
    and pick the one whose 255-bin re-quantization has minimal KL
    against the clipped reference distribution."""
    import numpy as np
    t = _t(x)
    arr = np.abs(t.detach().float().cpu().numpy()).ravel()
    amax = float(arr.max()) if arr.size else 0.0
    if amax == 0.0:
        return 1.0 / 127.0
    # histogram over a robust range: a lone extreme outlier would push
    # all the real mass into a handful of bins and blind the KL sweep
    # (outliers beyond the range clip into the last candidate bin)
    hi = min(amax, 4.0 * float(np.percentile(arr, 99.9)) + 1e-12)
    hist, edges = np.histogram(np.minimum(arr, hi), bins=num_bins,
                               range=(0, hi))
    hist = hist.astype(np.float64)
    best_kl, best_t = None, amax
    # candidate thresholds: from ~1/8 of the range up to amax
    for i in range(num_quantized_bins // 2, num_bins + 1,
                   max(1, num_bins // 128)):
        threshold = edges[i]
        p = hist[:i].copy()
        p[-1] += hist[i:].sum()  # clip outliers into the last bin
        if p.sum() == 0:
            continue
        # quantize the i reference bins down to num_quantized_bins
        factor = i / num_quantized_bins
        q = np.zeros(i)
        for j in range(num_quantized_bins):
            lo = int(np.floor(j * factor))
            hi = max(lo + 1, int(np.ceil((j + 1) * factor)))
            hi = min(hi, i)
            seg = p[lo:hi]
            nz = (seg > 0).sum()
            if nz:
                q[lo:hi][seg > 0] = seg[seg > 0].sum() / nz
        pn = p / p.sum()
        qs = q.sum()
        if qs == 0:
            continue
        qn = q / qs
        mask = pn > 0
        kl = float(np.sum(pn[mask] * np.log(
            pn[mask] / np.maximum(qn[mask], 1e-12))))
        if best_kl is None or kl < best_kl:
            best_kl, best_t = kl, threshold
    return best_t / 127.0


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
        self._act = dense_layer._act_type

    # offline-calibration hooks: when `collect` is armed, record the
    # activation for scale fitting; when `_x_scale` is fixed, use the
    # static scale instead of per-batch minmax (reference calibrated
    # quantize_v2 with out_type-range attrs)
    _x_scale = None
    _collect = None

    def __call__(self, x):
        t = _t(x)
        if self._collect is not None:
            self._collect.append(_calib_sample(t))
        xs = self._x_scale if self._x_scale is not None else calib_minmax(t)
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
        if self._act == 'relu':
            y = torch.relu(y)
        elif self._act:
            from ..ops import nn as _opsnn
            y = _opsnn.activation(y, self._act)
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

    _x_scale = None
    _collect = None

    def __call__(self, x):
        t = _t(x)
        if self._collect is not None:
            self._collect.append(_calib_sample(t))
        N, H, W, C = t.shape
        (sh, sw), (ph, pw), (dh, dw) = self._stride, self._pad, self._dil
        P = (H + 2 * ph - dh * (self._R - 1) - 1) // sh + 1
        Q = (W + 2 * pw - dw * (self._S - 1) - 1) // sw + 1
        xs = self._x_scale if self._x_scale is not None else calib_minmax(t)
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


def quantize_net(net, quantized_dtype='int8', exclude_layers=None,
                 calib_data=None, calib_mode='naive', num_calib_batches=5):
    """Quantize a Gluon net for int8 inference (reference
    quantize_model / quantize_graph_pass.cc): Dense and NHWC Conv2D
    children are REPLACED in the block tree with int8 versions.

    calib_data (iterable of input batches) runs an offline calibration
    pass: activation ranges are collected per quantized layer and fixed
    as static scales — 'naive' = abs-max over the batches, 'entropy' =
    KL-optimal threshold over the collected maxima distribution.
    Without calib_data the layers fall back to per-batch dynamic
    abs-max.  Returns the (block, name, wrapper) list."""
    from ..gluon import nn
    from ..base import native_mode
    if native_mode():
        raise NotImplementedError(
            'quantize_net runs on the torch frontend '
            '(set_native(False) / MXNET_NATIVE_RUNTIME=0): the int8 '
            'inference wrappers drive the i8 MFMA kernels through '
            'torch-tensor storage')
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
    # graph surgery: the wrappers take the original layers' places
    for block, name, q in swapped:
        object.__setattr__(block, name, q)
        block._children[name] = q
    if calib_data is not None and swapped:
        import numpy as np
        for _, _, q in swapped:
            q._collect = []
        n = 0
        for batch in calib_data:
            net(batch)
            n += 1
            if n >= num_calib_batches:
                break
        for _, _, q in swapped:
            samples = q._collect
            q._collect = None
            if not samples:
                q._x_scale = 1.0 / 127.0
                continue
            pooled = np.concatenate(samples)
            if calib_mode == 'entropy':
                # KL threshold over the pooled activation-value samples
                from ..ndarray.ndarray import array as _arr
                q._x_scale = calib_entropy(
                    _arr(pooled.astype('float32')))
            else:
                q._x_scale = float(pooled.max()) / 127.0
    return swapped
