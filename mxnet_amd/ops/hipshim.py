"""Torch-tensor frontend for the native kernel library.

Presents the same API the old ``_hipops`` torch extension had, but the
kernels live in ``mxnet_amd._core`` (built by hipcc alone — no torch
headers, no hipify): torch tensors cross the boundary as
(data_ptr, shape, dtype) triples, outputs are allocated HERE with torch
so they live in the frontend's allocator, and every launch goes onto the
caller's current torch stream.
"""
import torch

from .. import _core

_raw = _core.raw

# torch dtype -> mxcore DTypeFlag (src/core/base.h)
_FLAG = {
    torch.float32: 0, torch.float64: 1, torch.float16: 2,
    torch.uint8: 3, torch.int32: 4, torch.int8: 5, torch.int64: 6,
    torch.bool: 7, torch.bfloat16: 12,
}


def _arr(t):
    if t is None:
        return None
    return (t.data_ptr(), tuple(t.shape), _FLAG[t.dtype])


def _dev_stream(t):
    dev = t.device.index or 0
    return dev, torch.cuda.current_stream(dev).cuda_stream


def _c(t):
    assert t.is_cuda, 'native kernels need GPU tensors'
    return t.contiguous()


def _f32(t):
    return t.float().contiguous() if t.dtype is not torch.float32 \
        else t.contiguous()


# ---------------------------------------------------------------------------
# gemm family
# ---------------------------------------------------------------------------
def gemm_nt(x, w, bias=None):
    x, w = _c(x), _c(w)
    dev, s = _dev_stream(x)
    shape = tuple(x.shape[:-1]) + (w.shape[-2],)
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    _raw.gemm_nt(dev, s, _arr(x), _arr(w),
                 _arr(_f32(bias)) if bias is not None and bias.numel() else None,
                 _arr(out), False, None)
    return out


def gemm_nt_fused_relu(x, w, bias=None):
    x, w = _c(x), _c(w)
    dev, s = _dev_stream(x)
    shape = tuple(x.shape[:-1]) + (w.shape[-2],)
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    _raw.gemm_nt(dev, s, _arr(x), _arr(w),
                 _arr(_f32(bias)) if bias is not None and bias.numel() else None,
                 _arr(out), True, None)
    return out


def _gemm_nt_stats_possible(x, w):
    """Mirrors the C++ routing: the fused per-channel sum/ssq epilogue only
    runs on the unsplit, unbatched MFMA path (gemm.hip gemm_nt_raw)."""
    if x.dtype not in (torch.float16, torch.bfloat16) or x.dim() == 3:
        return False
    M, K = x.shape[-2], (x.shape[-1] + 7) // 8 * 8
    N = w.shape[-2]
    nwg = ((M + 127) // 128) * ((N + 127) // 128)
    nk = (K + 63) // 64
    return not (nwg < 512 and nk > 16)


def gemm_nt_stats(x, w, bias=None):
    """gemm_nt + fused BN-forward stats ([64,2,N] fp32) when the executing
    path supports it (else stats is an empty tensor)."""
    x, w = _c(x), _c(w)
    dev, s = _dev_stream(x)
    N = w.shape[-2]
    shape = tuple(x.shape[:-1]) + (N,)
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    if _gemm_nt_stats_possible(x, w):
        stats = torch.empty((64, 2, N), dtype=torch.float32, device=x.device)
        sarr = _arr(stats)
    else:
        stats = torch.empty(0, dtype=torch.float32, device=x.device)
        sarr = None
    _raw.gemm_nt(dev, s, _arr(x), _arr(w),
                 _arr(_f32(bias)) if bias is not None and bias.numel() else None,
                 _arr(out), False, sarr)
    return out, stats


def gemm(a, b):
    a, b = _c(a), _c(b)
    dev, s = _dev_stream(a)
    out = torch.empty((a.shape[0], b.shape[1]), dtype=a.dtype,
                      device=a.device)
    _raw.gemm(dev, s, _arr(a), _arr(b), _arr(out))
    return out


def gemm_nn(dy, w):
    dy, w = _c(dy), _c(w)
    dev, s = _dev_stream(dy)
    out = torch.empty((dy.shape[0], w.shape[1]), dtype=dy.dtype,
                      device=dy.device)
    _raw.gemm_nn(dev, s, _arr(dy), _arr(w), _arr(out))
    return out


def gemm_tn(dy, x):
    dy, x = _c(dy), _c(x)
    dev, s = _dev_stream(dy)
    out = torch.empty((dy.shape[1], x.shape[1]), dtype=dy.dtype,
                      device=dy.device)
    _raw.gemm_tn(dev, s, _arr(dy), _arr(x), _arr(out))
    return out


def bgemm(a, b):
    a, b = _c(a), _c(b)
    dev, s = _dev_stream(a)
    out = torch.empty((a.shape[0], a.shape[1], b.shape[2]), dtype=a.dtype,
                      device=a.device)
    _raw.bgemm(dev, s, _arr(a), _arr(b), _arr(out))
    return out


def transpose2d(x):
    x = _c(x)
    dev, s = _dev_stream(x)
    if x.dim() == 3:
        out = torch.empty((x.shape[0], x.shape[2], x.shape[1]),
                          dtype=x.dtype, device=x.device)
    else:
        out = torch.empty((x.shape[1], x.shape[0]), dtype=x.dtype,
                          device=x.device)
    _raw.transpose2d(dev, s, _arr(x), _arr(out))
    return out


def gemm_nt_8ph(a, b):
    a, b = _c(a), _c(b)
    dev, s = _dev_stream(a)
    out = torch.empty((a.shape[-2], b.shape[-2]), dtype=a.dtype,
                      device=a.device)
    _raw.gemm_nt_8ph(dev, s, _arr(a), _arr(b), _arr(out))
    return out


def gemm_tn_fused(A, B, want_bias):
    A, B = _c(A), _c(B)
    dev, s = _dev_stream(A)
    I, J = A.shape[1], B.shape[1]
    C = torch.empty((I, J), dtype=A.dtype, device=A.device)
    db = torch.zeros((I,) if want_bias else (0,), dtype=torch.float32,
                     device=A.device)
    _raw.gemm_tn_fused(dev, s, _arr(A), _arr(B), _arr(C),
                       _arr(db) if want_bias else None)
    return [C, db]


def attention_fwd(qkv, mask=None, heads=1, temperature=1.0):
    qkv = _c(qkv)
    dev, s = _dev_stream(qkv)
    B, S, U3 = qkv.shape
    U = U3 // 3
    BH = B * heads
    out = torch.empty((B, S, U), dtype=qkv.dtype, device=qkv.device)
    att = torch.empty((BH, S, S), dtype=qkv.dtype, device=qkv.device)
    m = None
    if mask is not None and mask.numel():
        m = _arr(mask.to(torch.uint8).contiguous())
    _raw.attention_fwd(dev, s, _arr(qkv), m, heads, temperature, _arr(out),
                       _arr(att))
    return [out, att]


def attention_bwd(dout, qkv, att, heads, temperature):
    dout, qkv, att = _c(dout), _c(qkv), _c(att)
    dev, s = _dev_stream(qkv)
    dqkv = torch.empty_like(qkv)
    _raw.attention_bwd(dev, s, _arr(dout), _arr(qkv), _arr(att), heads,
                       temperature, _arr(dqkv))
    return dqkv


# ---------------------------------------------------------------------------
# softmax / colsum
# ---------------------------------------------------------------------------
def softmax_fwd(x, log, temperature, mask=None):
    x = _c(x)
    dev, s = _dev_stream(x)
    y = torch.empty_like(x)
    m = None
    if mask is not None and mask.numel():
        m = _arr(mask.to(torch.uint8).contiguous())
    _raw.softmax_fwd(dev, s, _arr(x), m, log, temperature, _arr(y))
    return y


def softmax_bwd(dy, y, log, temperature):
    dy, y = _c(dy), _c(y)
    dev, s = _dev_stream(dy)
    dx = torch.empty_like(dy)
    _raw.softmax_bwd(dev, s, _arr(dy), _arr(y), log, temperature, _arr(dx))
    return dx


def colsum(x):
    x = _c(x)
    dev, s = _dev_stream(x)
    out = torch.empty((x.shape[-1],), dtype=x.dtype, device=x.device)
    _raw.colsum(dev, s, _arr(x), _arr(out))
    return out


# ---------------------------------------------------------------------------
# conv (NHWC)
# ---------------------------------------------------------------------------
def _conv_out_hw(H, W, R, S, sh, sw, ph, pw, dh, dw):
    P = (H + 2 * ph - dh * (R - 1) - 1) // sh + 1
    Q = (W + 2 * pw - dw * (S - 1) - 1) // sw + 1
    return P, Q


def conv2d_nhwc_fwd(x, w, bias, sh, sw, ph, pw, dh, dw, groups):
    x, w = _c(x), _c(w)
    dev, s = _dev_stream(x)
    NB, H, W_, C = x.shape
    Kout, R, S = w.shape[0], w.shape[1], w.shape[2]
    P, Q = _conv_out_hw(H, W_, R, S, sh, sw, ph, pw, dh, dw)
    y = torch.empty((NB, P, Q, Kout), dtype=x.dtype, device=x.device)
    b = _arr(_f32(bias)) if bias is not None and bias.numel() else None
    _raw.conv2d_fwd(dev, s, _arr(x), _arr(w), b, sh, sw, ph, pw, dh, dw,
                    groups, _arr(y), None)
    return y


def conv2d_nhwc_fwd_stats(x, w, bias, sh, sw, ph, pw, dh, dw, groups):
    x, w = _c(x), _c(w)
    dev, s = _dev_stream(x)
    NB, H, W_, C = x.shape
    Kout, R, S = w.shape[0], w.shape[1], w.shape[2]
    Cg = C // groups
    P, Q = _conv_out_hw(H, W_, R, S, sh, sw, ph, pw, dh, dw)
    y = torch.empty((NB, P, Q, Kout), dtype=x.dtype, device=x.device)
    b = _arr(_f32(bias)) if bias is not None and bias.numel() else None
    # stats come from the MFMA igemm path or the 1x1-s1 GEMM path only
    # (mirrors conv2d_fwd_raw routing)
    mfma_ok = x.dtype in (torch.float16, torch.bfloat16) and Cg % 8 == 0
    depthwise = groups == C and Kout == C and w.shape[3] == 1
    have = mfma_ok and not depthwise and groups == 1
    if have and R == 1 and S == 1 and sh == 1 and sw == 1 and ph == 0 \
            and pw == 0:
        have = _gemm_nt_stats_possible(
            x.view(NB * P * Q, C), w.view(Kout, C))
    if have:
        stats = torch.empty((64, 2, Kout), dtype=torch.float32,
                            device=x.device)
        sarr = _arr(stats)
    else:
        stats = torch.empty(0, dtype=torch.float32, device=x.device)
        sarr = None
    _raw.conv2d_fwd(dev, s, _arr(x), _arr(w), b, sh, sw, ph, pw, dh, dw,
                    groups, _arr(y), sarr)
    return [y, stats]


def conv2d_nhwc_bwd_data(dy, w, H, W, sh, sw, ph, pw, dh, dw, groups):
    dy, w = _c(dy), _c(w)
    dev, s = _dev_stream(dy)
    NB = dy.shape[0]
    C = w.shape[3] * groups
    dx = torch.empty((NB, H, W, C), dtype=dy.dtype, device=dy.device)
    _raw.conv2d_bwd_data(dev, s, _arr(dy), _arr(w), sh, sw, ph, pw, dh, dw,
                         groups, H, W, _arr(dx))
    return dx


def conv2d_nhwc_bwd_weight(dy, x, R, S, sh, sw, ph, pw, dh, dw, groups):
    dy, x = _c(dy), _c(x)
    dev, s = _dev_stream(dy)
    Kout = dy.shape[3]
    C = x.shape[3]
    Cg = C // groups if not (groups == C and Kout == C) else 1
    if groups == C and Kout == C:
        dw_out = torch.empty((C, R, S, 1), dtype=dy.dtype, device=dy.device)
    else:
        dw_out = torch.empty((Kout, R, S, Cg if groups > 1 else C),
                             dtype=dy.dtype, device=dy.device)
    _raw.conv2d_bwd_weight(dev, s, _arr(dy), _arr(x), sh, sw, ph, pw, dh,
                           dw, groups, R, S, _arr(dw_out))
    return dw_out


def im2col_nhwc(x, R, S, sh, sw, ph, pw, dh, dw):
    x = _c(x)
    dev, s = _dev_stream(x)
    NB, H, W_, C = x.shape
    P, Q = _conv_out_hw(H, W_, R, S, sh, sw, ph, pw, dh, dw)
    col = torch.empty((NB * P * Q, R * S * C), dtype=x.dtype,
                      device=x.device)
    _raw.im2col(dev, s, _arr(x), R, S, sh, sw, ph, pw, dh, dw, _arr(col))
    return col


# ---------------------------------------------------------------------------
# norms
# ---------------------------------------------------------------------------
def bn_nhwc_fwd_train(x, gamma, beta, rmean, rvar, momentum, eps, fuse_relu,
                      residual, presums=None):
    x = _c(x)
    dev, s = _dev_stream(x)
    C = x.shape[-1]
    g32, b32 = _f32(gamma), _f32(beta)
    rm32 = rmean if rmean.dtype is torch.float32 else rmean.float()
    rv32 = rvar if rvar.dtype is torch.float32 else rvar.float()
    rm32, rv32 = rm32.contiguous(), rv32.contiguous()
    y = torch.empty_like(x)
    save_mean = torch.empty((C,), dtype=torch.float32, device=x.device)
    save_istd = torch.empty((C,), dtype=torch.float32, device=x.device)
    want_mask = fuse_relu and C % 8 == 0 and \
        x.dtype in (torch.float16, torch.bfloat16)
    mask = torch.empty((x.numel() // 8,) if want_mask else (0,),
                       dtype=torch.uint8, device=x.device)
    res = _arr(_c(residual)) if residual is not None and residual.numel() \
        else None
    pre = _arr(_c(presums)) if presums is not None and presums.numel() \
        else None
    _raw.bn_fwd_train(dev, s, _arr(x), _arr(g32), _arr(b32), _arr(rm32),
                      _arr(rv32), momentum, eps, fuse_relu, res, pre,
                      _arr(y), _arr(save_mean), _arr(save_istd),
                      _arr(mask) if want_mask else None)
    if rm32.data_ptr() != rmean.data_ptr():
        rmean.copy_(rm32.to(rmean.dtype))
    if rv32.data_ptr() != rvar.data_ptr():
        rvar.copy_(rv32.to(rvar.dtype))
    return [y, save_mean, save_istd, mask]


def bn_nhwc_fwd_infer(x, gamma, beta, rmean, rvar, eps, fuse_relu, residual):
    x = _c(x)
    dev, s = _dev_stream(x)
    y = torch.empty_like(x)
    res = _arr(_c(residual)) if residual is not None and residual.numel() \
        else None
    _raw.bn_fwd_infer(dev, s, _arr(x), _arr(_f32(gamma)), _arr(_f32(beta)),
                      _arr(_f32(rmean)), _arr(_f32(rvar)), eps, fuse_relu,
                      res, _arr(y))
    return y


def bn_nhwc_bwd(dy, x, gamma, save_mean, save_istd, fuse_relu, y, has_res,
                mask):
    dy, x = _c(dy), _c(x)
    dev, s = _dev_stream(dy)
    C = x.shape[-1]
    dx = torch.empty_like(x)
    dres = torch.empty_like(x) if has_res else \
        torch.empty(0, dtype=x.dtype, device=x.device)
    dgamma = torch.empty((C,), dtype=torch.float32, device=x.device)
    dbeta = torch.empty((C,), dtype=torch.float32, device=x.device)
    m = _arr(_c(mask)) if mask is not None and mask.numel() else None
    yarr = _arr(_c(y)) if y is not None and y.numel() else None
    _raw.bn_bwd(dev, s, _arr(dy), _arr(x), _arr(_f32(gamma)),
                _arr(save_mean), _arr(save_istd), fuse_relu, yarr, has_res,
                m, _arr(dx), _arr(dgamma), _arr(dbeta),
                _arr(dres) if has_res else None)
    return [dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype), dres]


def layernorm_fwd(x, gamma, beta, eps):
    x = _c(x)
    dev, s = _dev_stream(x)
    C = x.shape[-1]
    rows = x.numel() // C
    y = torch.empty_like(x)
    mean = torch.empty((rows,), dtype=torch.float32, device=x.device)
    istd = torch.empty((rows,), dtype=torch.float32, device=x.device)
    _raw.layernorm_fwd(dev, s, _arr(x), _arr(_f32(gamma)), _arr(_f32(beta)),
                       eps, _arr(y), _arr(mean), _arr(istd))
    return [y, mean, istd]


def layernorm_bwd(dy, x, gamma, mean, istd):
    dy, x = _c(dy), _c(x)
    dev, s = _dev_stream(dy)
    C = x.shape[-1]
    dx = torch.empty_like(x)
    dgamma = torch.empty((C,), dtype=torch.float32, device=x.device)
    dbeta = torch.empty((C,), dtype=torch.float32, device=x.device)
    _raw.layernorm_bwd(dev, s, _arr(dy), _arr(x), _arr(_f32(gamma)),
                       _arr(mean), _arr(istd), _arr(dx), _arr(dgamma),
                       _arr(dbeta))
    return [dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype)]


# ---------------------------------------------------------------------------
# pooling
# ---------------------------------------------------------------------------
def pool_nhwc_fwd(x, kind, kh, kw, sh, sw, ph, pw, cip):
    x = _c(x)
    dev, s = _dev_stream(x)
    N, H, W_, C = x.shape
    P = (H + 2 * ph - kh) // sh + 1
    Q = (W_ + 2 * pw - kw) // sw + 1
    y = torch.empty((N, P, Q, C), dtype=x.dtype, device=x.device)
    is_max = kind == 'max'
    arg = torch.empty((N, P, Q, C) if is_max else (0,), dtype=torch.int32,
                      device=x.device)
    _raw.pool_fwd(dev, s, _arr(x), kind, kh, kw, sh, sw, ph, pw, cip,
                  _arr(y), _arr(arg) if is_max else None)
    return [y, arg]


def pool_nhwc_bwd(dy, arg, kind, H, W, kh, kw, sh, sw, ph, pw, cip):
    dy = _c(dy)
    dev, s = _dev_stream(dy)
    N, P, Q, C = dy.shape
    dx = torch.empty((N, H, W, C), dtype=dy.dtype, device=dy.device)
    _raw.pool_bwd(dev, s, _arr(dy),
                  _arr(_c(arg)) if arg is not None and arg.numel() else None,
                  kind, kh, kw, sh, sw, ph, pw, H, W, cip, _arr(dx))
    return dx


# ---------------------------------------------------------------------------
# elementwise / optimizer / dropout / embedding / lstm
# ---------------------------------------------------------------------------
def act_fwd(x, kind):
    x = _c(x)
    dev, s = _dev_stream(x)
    y = torch.empty_like(x)
    _raw.act_fwd(dev, s, _arr(x), kind, _arr(y))
    return y


def act_bwd(dy, saved, kind):
    dy, saved = _c(dy), _c(saved)
    dev, s = _dev_stream(dy)
    dx = torch.empty_like(dy)
    _raw.act_bwd(dev, s, _arr(dy), _arr(saved), kind, _arr(dx))
    return dx


def sgd_update(w, master, grad, mom, lr, mu, wd, rescale, clip):
    dev, s = _dev_stream(w)
    _raw.sgd_update(dev, s, _arr(w),
                    _arr(master) if master is not None else None,
                    _arr(_c(grad)),
                    _arr(mom) if mom is not None else None,
                    lr, mu, wd, rescale, clip)


def adam_update(w, master, grad, m, v, lr_t, b1, b2, eps, wd, rescale, clip,
                adamw):
    dev, s = _dev_stream(w)
    _raw.adam_update(dev, s, _arr(w),
                     _arr(master) if master is not None else None,
                     _arr(_c(grad)), _arr(m), _arr(v), lr_t, b1, b2, eps,
                     wd, rescale, clip, adamw)


def multi_sgd_update(ws, masters, grads, moms, lrs, wds, mu, rescale, clip):
    if not ws:
        return
    dev, s = _dev_stream(ws[0])
    def opt(t):
        return _arr(t) if t is not None and t.numel() else None
    _raw.multi_sgd_update(dev, s, [_arr(t) for t in ws],
                          [opt(t) for t in masters],
                          [_arr(t) for t in grads],
                          [opt(t) for t in moms],
                          [float(v) for v in lrs], [float(v) for v in wds],
                          mu, rescale, clip)


def multi_all_finite(tensors):
    ts = [t for t in tensors if t is not None and t.numel()]
    if not ts:
        return True
    dev, s = _dev_stream(ts[0])
    flag = torch.zeros(1, dtype=torch.int32, device=ts[0].device)
    _raw.multi_all_finite(dev, s, [_arr(_c(t)) for t in ts], _arr(flag))
    return int(flag.item()) == 0


def lstm_cell_fwd(gates, c):
    gates, c = _c(gates), _c(c)
    dev, s = _dev_stream(gates)
    h_out = torch.empty_like(c)
    c_out = torch.empty_like(c)
    _raw.lstm_cell_fwd(dev, s, _arr(gates), _arr(c), _arr(h_out),
                       _arr(c_out))
    return [h_out, c_out]


def dropout_fwd(x, p, seed):
    x = _c(x)
    dev, s = _dev_stream(x)
    y = torch.empty_like(x)
    mask = torch.empty(x.shape, dtype=torch.uint8, device=x.device)
    _raw.dropout_fwd(dev, s, _arr(x), p, seed, _arr(y), _arr(mask))
    return [y, mask]


def dropout_bwd(dy, mask, p):
    dy = _c(dy)
    dev, s = _dev_stream(dy)
    dx = torch.empty_like(dy)
    _raw.dropout_bwd(dev, s, _arr(dy), _arr(_c(mask)), p, _arr(dx))
    return dx


def embedding_fwd(indices, weight):
    weight = _c(weight)
    idx = indices.to(torch.int64).contiguous()
    dev, s = _dev_stream(weight)
    D = weight.shape[1]
    out = torch.empty(tuple(idx.shape) + (D,), dtype=weight.dtype,
                      device=weight.device)
    _raw.embedding_fwd(dev, s, _arr(weight), _arr(idx), _arr(out))
    return out


def embedding_bwd(indices, dy, vocab):
    dy = _c(dy)
    idx = indices.to(torch.int64).contiguous()
    dev, s = _dev_stream(dy)
    D = dy.shape[-1]
    dw = torch.empty((vocab, D), dtype=dy.dtype, device=dy.device)
    _raw.embedding_bwd(dev, s, _arr(dy), _arr(idx), _arr(dw))
    return dw


# ---------------------------------------------------------------------------
# int8
# ---------------------------------------------------------------------------
def quantize_i8(x, scale):
    x = _c(x)
    dev, s = _dev_stream(x)
    y = torch.empty(x.shape, dtype=torch.int8, device=x.device)
    _raw.quantize_i8(dev, s, _arr(x), scale, _arr(y))
    return y


def dequantize_i8(x, scale, dtype=None):
    x = _c(x)
    dev, s = _dev_stream(x)
    dt = dtype if dtype is not None else torch.float32
    y = torch.empty(x.shape, dtype=dt, device=x.device)
    _raw.dequantize_i8(dev, s, _arr(x), scale, _arr(y))
    return y


def gemm_nt_i8(a, b, scale, out_dtype=None):
    a, b = _c(a), _c(b)
    dev, s = _dev_stream(a)
    dt = out_dtype if out_dtype is not None else torch.float16
    out = torch.empty((a.shape[0], b.shape[0]), dtype=dt, device=a.device)
    _raw.gemm_nt_i8(dev, s, _arr(a), _arr(b), scale, _arr(out))
    return out
