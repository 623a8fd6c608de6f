"""Numerics tests for the gfx950 HIP kernels against plain-PyTorch fp32
oracles (reference test strategy: check_consistency, test_utils.py:1490).

Every test here runs on a real MI355X (pytest -m gpu via gpurun).
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from mxnet_amd.ops import hipshim as ext
else:
    ext = None

DEV = 'cuda:0'


def mk(shape, dtype=torch.float16, scale=1.0, seed=None):
    if seed is not None:
        torch.manual_seed(seed)
    return (torch.randn(*shape, device=DEV) * scale).to(dtype)


def check(got, want32, tol=None):
    """got: GPU tensor; want32: fp32 oracle (any device)."""
    got32 = got.float().cpu()
    want32 = want32.float().cpu()
    assert got32.shape == want32.shape, (got32.shape, want32.shape)
    scale = want32.abs().max().item() + 1e-6
    if tol is None:
        tol = 2e-3 if got.dtype in (torch.float16, torch.bfloat16) else 1e-5
        if got.dtype is torch.bfloat16:
            tol = 1.6e-2
    err = (got32 - want32).abs().max().item()
    assert err <= tol * scale, f'max err {err:.4g} vs tol {tol * scale:.4g} (scale {scale:.4g})'


# ---------------------------------------------------------------------------
# GEMM family
# ---------------------------------------------------------------------------
@pytest.mark.parametrize('dtype', [torch.float16, torch.bfloat16, torch.float32])
@pytest.mark.parametrize('mnk', [(37, 53, 64), (128, 128, 128),
                                 (256, 1000, 2048), (130, 70, 72),
                                 (512, 256, 24)])
def test_gemm_nt(dtype, mnk):
    M, N, K = mnk
    a, b = mk((M, K), dtype, seed=0), mk((N, K), dtype, seed=1)
    bias = mk((N,), dtype, seed=2)
    y = ext.gemm_nt(a, b, bias)
    want = a.float() @ b.float().t() + bias.float()
    check(y, want)


def test_gemm_variants():
    a = mk((96, 200), seed=3)
    b = mk((200, 72), seed=4)
    check(ext.gemm(a, b), a.float() @ b.float())
    dy = mk((64, 120), seed=5)
    w = mk((120, 88), seed=6)
    check(ext.gemm_nn(dy, w), dy.float() @ w.float())
    x = mk((64, 88), seed=7)
    check(ext.gemm_tn(dy, x), dy.float().t() @ x.float())


def test_bgemm():
    a = mk((6, 33, 48), seed=8)
    b = mk((6, 48, 29), seed=9)
    check(ext.bgemm(a, b), torch.matmul(a.float(), b.float()))


def test_transpose2d():
    x = mk((123, 77), seed=10)
    assert torch.equal(ext.transpose2d(x), x.t().contiguous())
    x3 = mk((4, 50, 66), seed=11)
    assert torch.equal(ext.transpose2d(x3), x3.transpose(1, 2).contiguous())


# ---------------------------------------------------------------------------
# convolution (NHWC) vs torch fp32 NCHW oracle
# ---------------------------------------------------------------------------
CONV_CASES = [
    # (N, H, W, C, K, R, S, stride, pad, dil)  -- ResNet-50 v1.5 shapes
    (2, 56, 56, 64, 64, 1, 1, 1, 0, 1),       # 1x1 s1 (plain GEMM path)
    (2, 56, 56, 64, 64, 3, 3, 1, 1, 1),       # 3x3 s1 (implicit)
    (2, 56, 56, 256, 512, 1, 1, 2, 0, 1),     # 1x1 s2 (implicit gather)
    (2, 28, 28, 128, 128, 3, 3, 2, 1, 1),     # 3x3 s2
    (2, 32, 32, 3, 64, 7, 7, 2, 3, 1),        # stem RGB (im2col path)
    (2, 16, 16, 72, 40, 3, 3, 1, 1, 1),       # C%8==0 but not %64
    (1, 20, 20, 64, 32, 3, 3, 1, 2, 2),       # dilation 2
    (2, 9, 9, 16, 24, 2, 2, 2, 0, 1),         # even kernel, odd spatial
]


def conv_oracle(x, w, stride, pad, dil):
    xn = x.float().permute(0, 3, 1, 2)
    wn = w.float().permute(0, 3, 1, 2)
    y = torch.nn.functional.conv2d(xn, wn, None, stride=stride, padding=pad,
                                   dilation=dil)
    return y.permute(0, 2, 3, 1)


@pytest.mark.parametrize('case', CONV_CASES)
def test_conv_fwd(case):
    N, H, W, C, K, R, S, st, pd, dl = case
    x = mk((N, H, W, C), seed=20)
    w = mk((K, R, S, C), seed=21, scale=0.5)
    y = ext.conv2d_nhwc_fwd(x, w, None, st, st, pd, pd, dl, dl, 1)
    check(y, conv_oracle(x, w, st, pd, dl))


@pytest.mark.parametrize('case', CONV_CASES)
def test_conv_bwd(case):
    N, H, W, C, K, R, S, st, pd, dl = case
    x = mk((N, H, W, C), seed=22)
    w = mk((K, R, S, C), seed=23, scale=0.5)
    xn = x.float().permute(0, 3, 1, 2).requires_grad_(True)
    wn = w.float().permute(0, 3, 1, 2).requires_grad_(True)
    y = torch.nn.functional.conv2d(xn, wn, None, stride=st, padding=pd,
                                   dilation=dl)
    dy_n = torch.randn_like(y)
    y.backward(dy_n)
    dy = dy_n.permute(0, 2, 3, 1).contiguous().to(x.dtype)
    dx = ext.conv2d_nhwc_bwd_data(dy, w, H, W, st, st, pd, pd, dl, dl, 1)
    check(dx, xn.grad.permute(0, 2, 3, 1), tol=4e-3)
    dw = ext.conv2d_nhwc_bwd_weight(dy, x, R, S, st, st, pd, pd, dl, dl, 1)
    check(dw, wn.grad.permute(0, 2, 3, 1), tol=4e-3)


# ---------------------------------------------------------------------------
# batch norm
# ---------------------------------------------------------------------------
@pytest.mark.parametrize('fuse_relu', [False, True])
@pytest.mark.parametrize('with_res', [False, True])
def test_bn_train_fwd_bwd(fuse_relu, with_res):
    N, H, W, C = 4, 7, 9, 32
    x = mk((N, H, W, C), seed=30)
    res = mk((N, H, W, C), seed=31) if with_res else None
    gamma = mk((C,), torch.float32, seed=32) + 1.0
    beta = mk((C,), torch.float32, seed=33)
    rmean = torch.zeros(C, device=DEV)
    rvar = torch.ones(C, device=DEV)
    rmean0, rvar0 = rmean.clone(), rvar.clone()
    y, smean, sistd, mask = ext.bn_nhwc_fwd_train(
        x, gamma, beta, rmean, rvar, 0.9, 1e-5, fuse_relu,
        res if res is not None else x.new_empty(0))
    # oracle
    x32 = x.float().requires_grad_(True)
    res32 = res.float().requires_grad_(True) if with_res else None
    g32 = gamma.clone().requires_grad_(True)
    b32 = beta.clone().requires_grad_(True)
    rm, rv = rmean0.clone(), rvar0.clone()
    xn = x32.reshape(-1, C)
    yo = torch.nn.functional.batch_norm(
        xn.reshape(-1, C).t().reshape(1, C, -1), rm, rv, g32, b32,
        training=True, momentum=0.1, eps=1e-5)
    yo = yo.reshape(C, -1).t().reshape(N, H, W, C)
    if with_res:
        yo = yo + res32
    if fuse_relu:
        yo = torch.relu(yo)
    check(y, yo.detach(), tol=5e-3)
    check(rmean, rm, tol=1e-4)
    check(rvar, rv, tol=1e-4)
    # backward
    dy_o = torch.randn(N, H, W, C, device=DEV)
    yo.backward(dy_o)
    dy = dy_o.to(x.dtype)
    dx, dgamma, dbeta, dres = ext.bn_nhwc_bwd(dy, x, gamma, smean, sistd,
                                              fuse_relu, y, with_res, mask)
    check(dx, x32.grad, tol=6e-3)
    check(dgamma, g32.grad, tol=6e-3)
    check(dbeta, b32.grad, tol=6e-3)
    if with_res:
        check(dres, res32.grad, tol=6e-3)


def test_bn_infer():
    N, H, W, C = 3, 5, 5, 48
    x = mk((N, H, W, C), seed=35)
    gamma = mk((C,), torch.float32, seed=36) + 1.0
    beta = mk((C,), torch.float32, seed=37)
    rmean = mk((C,), torch.float32, seed=38)
    rvar = mk((C,), torch.float32, seed=39).abs() + 0.5
    y = ext.bn_nhwc_fwd_infer(x, gamma, beta, rmean, rvar, 1e-5, False,
                              x.new_empty(0))
    want = (x.float() - rmean) / torch.sqrt(rvar + 1e-5) * gamma + beta
    check(y, want, tol=5e-3)


# ---------------------------------------------------------------------------
# pooling
# ---------------------------------------------------------------------------
@pytest.mark.parametrize('kind', ['max', 'avg'])
def test_pool(kind):
    N, H, W, C = 2, 13, 13, 24
    x = mk((N, H, W, C), seed=40)
    y, arg = ext.pool_nhwc_fwd(x, kind, 3, 3, 2, 2, 1, 1, False)
    # oracle on CPU: ROCm's avg_pool2d *backward* with
    # count_include_pad=False disagrees with torch-CPU (and with the
    # mathematical vjp of its own forward); our kernels match CPU.
    xn = x.float().cpu().permute(0, 3, 1, 2).requires_grad_(True)
    if kind == 'max':
        yo = torch.nn.functional.max_pool2d(xn, 3, 2, 1)
    else:
        yo = torch.nn.functional.avg_pool2d(xn, 3, 2, 1,
                                            count_include_pad=False)
    check(y, yo.permute(0, 2, 3, 1).detach())
    torch.manual_seed(41)
    dy_o = torch.randn_like(yo)
    yo.backward(dy_o)
    dy = dy_o.permute(0, 2, 3, 1).contiguous().to(x.dtype).to(DEV)
    dx = ext.pool_nhwc_bwd(dy, arg, kind, H, W, 3, 3, 2, 2, 1, 1, False)
    check(dx, xn.grad.permute(0, 2, 3, 1), tol=4e-3)


# ---------------------------------------------------------------------------
# softmax / layernorm / colsum
# ---------------------------------------------------------------------------
@pytest.mark.parametrize('log', [False, True])
@pytest.mark.parametrize('C', [8, 1000, 4096])
def test_softmax(log, C):
    x = mk((64, C), seed=50, scale=3.0)
    y = ext.softmax_fwd(x, log, 1.0)
    want = (torch.log_softmax if log else torch.softmax)(x.float(), -1)
    check(y, want, tol=3e-3)
    x32 = x.float().requires_grad_(True)
    yo = (torch.log_softmax if log else torch.softmax)(x32, -1)
    dy_o = torch.randn_like(yo)
    yo.backward(dy_o)
    dx = ext.softmax_bwd(dy_o.to(x.dtype), y, log, 1.0)
    check(dx, x32.grad, tol=4e-3)


def test_softmax_temperature():
    x = mk((16, 100), seed=51, scale=2.0)
    y = ext.softmax_fwd(x, False, 2.5)
    check(y, torch.softmax(x.float() / 2.5, -1), tol=3e-3)


def test_layernorm():
    R, C = 128, 768
    x = mk((R, C), seed=52)
    gamma = mk((C,), torch.float32, seed=53) + 1.0
    beta = mk((C,), torch.float32, seed=54)
    y, mean, istd = ext.layernorm_fwd(x, gamma, beta, 1e-5)
    x32 = x.float().requires_grad_(True)
    g32 = gamma.clone().requires_grad_(True)
    b32 = beta.clone().requires_grad_(True)
    yo = torch.nn.functional.layer_norm(x32, (C,), g32, b32, 1e-5)
    check(y, yo.detach(), tol=4e-3)
    dy_o = torch.randn_like(yo)
    yo.backward(dy_o)
    dx, dg, db = ext.layernorm_bwd(dy_o.to(x.dtype), x, gamma, mean, istd)
    check(dx, x32.grad, tol=6e-3)
    check(dg, g32.grad, tol=6e-3)
    check(db, b32.grad, tol=6e-3)


def test_colsum():
    x = mk((4096, 1000), seed=55)
    check(ext.colsum(x), x.float().sum(0), tol=3e-3)


# ---------------------------------------------------------------------------
# elementwise
# ---------------------------------------------------------------------------
@pytest.mark.parametrize('kind', ['relu', 'sigmoid', 'tanh', 'gelu', 'silu'])
def test_activation(kind):
    x = mk((3, 1000), seed=60, scale=2.0)
    y = ext.act_fwd(x, kind)
    import torch.nn.functional as F
    oracle = {'relu': F.relu, 'sigmoid': torch.sigmoid, 'tanh': torch.tanh,
              'gelu': lambda t: F.gelu(t, approximate='tanh'),
              'silu': F.silu}[kind]
    x32 = x.float().requires_grad_(True)
    yo = oracle(x32)
    check(y, yo.detach(), tol=3e-3)
    dy_o = torch.randn_like(yo)
    yo.backward(dy_o)
    saved = y if kind in ('relu', 'sigmoid', 'tanh') else x
    dx = ext.act_bwd(dy_o.to(x.dtype), saved, kind)
    check(dx, x32.grad, tol=4e-3)


def test_sgd_update():
    n = 10007
    w = mk((n,), seed=70)
    master = w.float().clone()
    g = mk((n,), seed=71)
    mom = torch.randn(n, device=DEV)
    m2, mom2 = master.clone(), mom.clone()
    ext.sgd_update(w, master, g, mom, 0.1, 0.9, 1e-4, 0.5, 0.0)
    # oracle — reference sgd_mom_update rule: lr folded into the momentum
    # buffer (mom = mu*mom - lr*g; w += mom)
    ge = g.float() * 0.5 + 1e-4 * m2
    mom2 = mom2 * 0.9 - 0.1 * ge
    m2 = m2 + mom2
    check(master, m2, tol=1e-5)
    check(mom, mom2, tol=1e-5)
    check(w, m2, tol=2e-3)


def test_adam_update():
    n = 4099
    w = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    w2, m2, v2 = w.clone(), m.clone(), v.clone()
    ext.adam_update(w, None, g, m, v, 0.01, 0.9, 0.999, 1e-8, 0.0, 1.0, 0.0,
                    False)
    m2 = 0.9 * m2 + 0.1 * g
    v2 = 0.999 * v2 + 0.001 * g * g
    w2 = w2 - 0.01 * m2 / (v2.sqrt() + 1e-8)
    check(w, w2, tol=1e-5)


def test_multi_all_finite():
    a = mk((100,), seed=72)
    b = mk((50,), seed=73)
    assert ext.multi_all_finite([a, b])
    b[7] = float('inf')
    assert not ext.multi_all_finite([a, b])
    b[7] = float('nan')
    assert not ext.multi_all_finite([a, b])


def test_lstm_cell():
    N, H = 8, 64
    gates = mk((N, 4 * H), seed=74)
    c = mk((N, H), seed=75)
    h2, c2 = ext.lstm_cell_fwd(gates, c)
    i, f, g, o = gates.float().split(H, dim=-1)
    i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
    g = torch.tanh(g)
    cn = f * c.float() + i * g
    check(c2, cn, tol=3e-3)
    check(h2, o * torch.tanh(cn), tol=3e-3)


def test_dropout():
    x = torch.ones(100000, device=DEV, dtype=torch.float16)
    y, mask = ext.dropout_fwd(x, 0.3, 1234)
    keep = mask.float().mean().item()
    assert abs(keep - 0.7) < 0.02
    got = y.float()
    assert torch.allclose(got[mask.bool()],
                          torch.full_like(got[mask.bool()], 1 / 0.7),
                          atol=1e-3)
    dy = torch.ones_like(x)
    dx = ext.dropout_bwd(dy, mask, 0.3)
    assert torch.allclose(dx.float(), mask.float() / 0.7, atol=1e-3)


def test_embedding():
    V, D = 1000, 96
    w = mk((V, D), seed=76)
    idx = torch.randint(0, V, (4, 37), device=DEV)
    y = ext.embedding_fwd(idx, w)
    check(y, w.float()[idx])
    dy = mk((4, 37, D), seed=77)
    dw = ext.embedding_bwd(idx, dy, V)
    want = torch.zeros(V, D, device=DEV)
    want.index_add_(0, idx.reshape(-1), dy.float().reshape(-1, D))
    check(dw, want, tol=4e-3)


# ---------------------------------------------------------------------------
# integration: one ResNet bottleneck through the framework, GPU fp16 vs
# CPU fp32 oracle
# ---------------------------------------------------------------------------
def test_resnet_block_integration():
    import mxnet_amd as mx
    from mxnet_amd import autograd
    from mxnet_amd.gluon.model_zoo.vision.resnet import BottleneckV1

    torch.manual_seed(99)
    blk = BottleneckV1(64, stride=2, downsample=True, in_channels=32,
                       layout='NHWC')
    blk.initialize(ctx=mx.cpu())
    x_cpu = mx.nd.array(torch.randn(2, 16, 16, 32))
    with autograd.record():
        y_cpu = blk(x_cpu)
    y_cpu.backward()
    g_cpu = {k: p.grad().asnumpy() for k, p in blk.collect_params().items()
             if p.grad_req != 'null'}

    blk.reset_ctx(mx.gpu(0))
    blk.cast('float16')
    x_gpu = mx.nd.from_torch(x_cpu.handle.to(DEV).half())
    with autograd.record():
        y_gpu = blk(x_gpu)
    y_gpu.backward()

    np.testing.assert_allclose(y_gpu.asnumpy().astype(np.float32),
                               y_cpu.asnumpy(), rtol=0.1, atol=0.05)
    for k, p in blk.collect_params().items():
        if p.grad_req == 'null':
            continue
        got = p.grad().asnumpy().astype(np.float32)
        want = g_cpu[k]
        scale = np.abs(want).max() + 1e-6
        err = np.abs(got - want)
        # fp16 whole-block chain: demand tight agreement in bulk and
        # bounded single-element outliers; tiny per-channel params (BN
        # gamma/beta are 128-element reductions here) get a loose bound
        if want.size < 256:
            assert err.max() < 0.3 * scale, k
        else:
            assert np.percentile(err, 99.5) < 0.08 * scale, k
            assert err.max() < 0.25 * scale, k


GROUP_CASES = [
    # (N, H, W, C, K, R, S, stride, pad, dil, groups)
    (2, 14, 14, 64, 128, 3, 3, 1, 1, 1, 4),    # grouped igemm (ResNeXt-ish)
    (2, 14, 14, 32, 32, 3, 3, 1, 1, 1, 32),    # depthwise (MobileNet)
    (2, 15, 15, 48, 48, 3, 3, 2, 1, 1, 48),    # depthwise stride 2
]


def group_oracle(x, w, st, pd, dl, g):
    xn = x.float().permute(0, 3, 1, 2)
    wn = w.float().permute(0, 3, 1, 2)
    y = torch.nn.functional.conv2d(xn, wn, None, stride=st, padding=pd,
                                   dilation=dl, groups=g)
    return y.permute(0, 2, 3, 1)


@pytest.mark.parametrize('case', GROUP_CASES)
def test_grouped_conv(case):
    N, H, W, C, K, R, S, st, pd, dl, g = case
    x = mk((N, H, W, C), seed=25)
    w = mk((K, R, S, C // g), seed=26, scale=0.5)
    y = ext.conv2d_nhwc_fwd(x, w, None, st, st, pd, pd, dl, dl, g)
    check(y, group_oracle(x, w, st, pd, dl, g))
    # backward
    xn = x.float().permute(0, 3, 1, 2).requires_grad_(True)
    wn = w.float().permute(0, 3, 1, 2).requires_grad_(True)
    yo = torch.nn.functional.conv2d(xn, wn, None, stride=st, padding=pd,
                                    dilation=dl, groups=g)
    dy_n = torch.randn_like(yo)
    yo.backward(dy_n)
    dy = dy_n.permute(0, 2, 3, 1).contiguous().to(x.dtype)
    dx = ext.conv2d_nhwc_bwd_data(dy, w, H, W, st, st, pd, pd, dl, dl, g)
    check(dx, xn.grad.permute(0, 2, 3, 1), tol=4e-3)
    dw = ext.conv2d_nhwc_bwd_weight(dy, x, R, S, st, st, pd, pd, dl, dl, g)
    check(dw, wn.grad.permute(0, 2, 3, 1), tol=4e-3)


def test_hybridize_hipgraph_inference():
    """hybridize(static_alloc) captures the forward in a hipGraph; replay
    must match eager output (reference CachedOp static execution)."""
    import mxnet_amd as mx
    from mxnet_amd.gluon.model_zoo.vision import resnet18_v1
    torch.manual_seed(3)
    net = resnet18_v1(classes=10, layout='NHWC')
    net.initialize(ctx=mx.gpu(0))
    net.cast('float16')
    x = mx.nd.from_torch(torch.randn(2, 64, 64, 3, device=DEV).half())
    y_eager = net(x).asnumpy()
    net.hybridize(static_alloc=True, static_shape=True)
    y_graph1 = net(x).asnumpy()
    np.testing.assert_allclose(y_graph1, y_eager, rtol=2e-2, atol=2e-2)
    # second input through the captured graph
    x2 = mx.nd.from_torch(torch.randn(2, 64, 64, 3, device=DEV).half())
    net.hybridize(False)
    y2_eager = net(x2).asnumpy()
    net.hybridize(static_alloc=True, static_shape=True)
    net(x)  # capture
    y2_graph = net(x2).asnumpy()
    np.testing.assert_allclose(y2_graph, y2_eager, rtol=2e-2, atol=2e-2)


def test_int8_gemm_and_quantize():
    """int8 MFMA GEMM vs fp32 oracle (reference quantized FC path)."""
    from mxnet_amd.ops import hipshim as hx
    M, N, K = 128, 96, 256
    a = (torch.randn(M, K, device=DEV) * 20).clamp(-127, 127).round().to(torch.int8)
    b = (torch.randn(N, K, device=DEV) * 20).clamp(-127, 127).round().to(torch.int8)
    y = hx.gemm_nt_i8(a, b, 0.5, torch.float32)
    want = (a.float() @ b.float().t()) * 0.5
    check(y, want, tol=1e-5)
    x = torch.randn(1000, device=DEV).half()
    scale = x.float().abs().max().item() / 127.0
    q = hx.quantize_i8(x, scale)
    d = hx.dequantize_i8(q, scale, torch.float32)
    assert (d - x.float()).abs().max().item() <= scale * 0.51


@pytest.mark.parametrize('dtype', [torch.float16, torch.bfloat16])
@pytest.mark.parametrize('mnk', [(640, 512, 512), (2048, 1024, 1024),
                                 (513, 300, 264), (768, 768, 768)])
def test_gemm_nt_8phase_path(dtype, mnk):
    """256^2 8-phase kernel (raw barriers, st_16x32 swizzle) — refcheck
    against fp32 (guide two-lane discipline)."""
    M, N, K = mnk
    a, b = mk((M, K), dtype, seed=90), mk((N, K), dtype, seed=91)
    y = ext.gemm_nt_8ph(a, b)
    want = a.float() @ b.float().t()
    check(y, want)


def test_multi_sgd_matches_single():
    """multi-tensor SGD == per-tensor fused SGD (reference multi_sgd)."""
    from mxnet_amd.ops import hipshim as hx
    torch.manual_seed(5)
    shapes = [(1000,), (64, 32), (7,), (128, 3, 3, 8)]
    ws = [torch.randn(*s, device=DEV).half() for s in shapes]
    masters = [w.float().clone() for w in ws]
    grads = [torch.randn_like(w) for w in ws]
    moms = [torch.randn(w.numel(), device=DEV).reshape(w.shape) for w in ws]
    # reference: per-tensor kernel
    ws_r = [w.clone() for w in ws]
    ms_r = [m.clone() for m in masters]
    mo_r = [m.clone() for m in moms]
    for i in range(len(ws)):
        hx.sgd_update(ws_r[i], ms_r[i], grads[i], mo_r[i], 0.1 + 0.01 * i,
                      0.9, 1e-4, 0.5, 0.0)
    hx.multi_sgd_update(ws, masters, grads, moms,
                        [0.1 + 0.01 * i for i in range(len(ws))],
                        [1e-4] * len(ws), 0.9, 0.5, 0.0)
    for a, b in zip(masters, ms_r):
        assert torch.allclose(a, b, atol=1e-6), 'master mismatch'
    for a, b in zip(moms, mo_r):
        assert torch.allclose(a, b, atol=1e-6), 'momentum mismatch'


# ---------------------------------------------------------------------------
# transposed convolution (deconv2d: conv kernels with roles swapped)
# ---------------------------------------------------------------------------
DECONV_CASES = [
    # (N, H, W, Cin, Cout, R, S, stride, pad, opad, groups)
    (2, 14, 14, 64, 32, 2, 2, 2, 0, 0, 1),     # classic 2x upsample
    (2, 10, 10, 48, 24, 3, 3, 2, 1, 1, 1),     # stride 2 + output_padding
    (2, 16, 16, 32, 32, 3, 3, 1, 1, 0, 1),     # same-size
    (2, 8, 8, 64, 64, 4, 4, 2, 1, 0, 2),       # grouped
]


@pytest.mark.parametrize('case', DECONV_CASES)
def test_deconv_fwd_bwd(case):
    from mxnet_amd.ops import nn as onn
    N, H, W, Ci, Co, R, S, st, pd, op, g = case
    x = mk((N, H, W, Ci), seed=60)
    w = mk((Ci, R, S, Co // g), seed=61, scale=0.5)
    b = mk((Co,), seed=62)
    xg = x.clone().requires_grad_(True)
    wg = w.clone().requires_grad_(True)
    bg = b.clone().requires_grad_(True)
    y = onn.deconv2d(xg, wg, bg, (st, st), (pd, pd), (op, op), (1, 1), g,
                     layout='NHWC')
    # fp32 oracle on CPU through torch conv_transpose2d
    xo = x.float().cpu().permute(0, 3, 1, 2).requires_grad_(True)
    wo = w.float().cpu().permute(0, 3, 1, 2).requires_grad_(True)
    bo = b.float().cpu().requires_grad_(True)
    yo = torch.nn.functional.conv_transpose2d(
        xo, wo, bo, stride=st, padding=pd, output_padding=op, groups=g)
    check(y, yo.permute(0, 2, 3, 1).to(y.device))
    dy = torch.randn_like(yo)
    yo.backward(dy)
    y.backward(dy.permute(0, 2, 3, 1).contiguous().to(y.dtype).to(y.device))
    check(xg.grad, xo.grad.permute(0, 2, 3, 1).to(y.device), tol=4e-3)
    check(wg.grad, wo.grad.permute(0, 2, 3, 1).to(y.device), tol=4e-3)
    check(bg.grad, bo.grad.to(y.device), tol=4e-3)


def test_conv2dtranspose_layer_gpu():
    """Gluon Conv2DTranspose NHWC fp16 layer runs the native igemm path."""
    from mxnet_amd import autograd
    from mxnet_amd.gluon import nn as gnn
    import mxnet_amd as mx
    net = gnn.Conv2DTranspose(32, 4, strides=2, padding=1, layout='NHWC',
                              in_channels=64)
    net.initialize(ctx=mx.gpu(0))
    net.cast('float16')
    x = mx.nd.from_torch(torch.randn(2, 7, 7, 64, device='cuda',
                                     dtype=torch.float16))
    with autograd.record():
        y = net(x)
        L = mx.nd.from_torch(y.handle.float().square().mean())
    L.backward()
    assert y.shape == (2, 14, 14, 32)
    g = net.weight.grad().handle
    assert g is not None and torch.isfinite(g.float()).all()


def test_quantized_conv_gpu():
    """Int8 conv (im2col + i8 MFMA GEMM) tracks the fp16 conv within
    quantization error."""
    import mxnet_amd as mx
    from mxnet_amd.gluon import nn as gnn
    from mxnet_amd.contrib.quantization import QuantizedConv2D
    torch.manual_seed(1)
    c = gnn.Conv2D(32, 3, strides=1, padding=1, layout='NHWC',
                   in_channels=64, use_bias=True)
    c.initialize(ctx=mx.gpu(0))
    c.cast('float16')
    x = mx.nd.from_torch(torch.randn(2, 14, 14, 64, device='cuda',
                                     dtype=torch.float16))
    y_fp = c(x).handle.float()
    q = QuantizedConv2D(c)
    y_q = q(x).handle.float()
    assert y_q.shape == y_fp.shape
    rel = (y_q - y_fp).abs().max() / (y_fp.abs().max() + 1e-9)
    assert float(rel) < 0.06, float(rel)


# ---------------------------------------------------------------------------
# fused attention core (strided NT GEMM, no head reshapes)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize('masked', [False, True])
def test_attention_core(masked):
    from mxnet_amd.ops import nn as onn
    torch.manual_seed(33)
    B, S, H, D = 2, 64, 4, 16
    U = H * D
    temp = float(D) ** 0.5
    qkv = torch.randn(B, S, 3 * U, device='cuda', dtype=torch.float16) \
        .requires_grad_(True)
    if masked:
        valid = torch.ones(B, S, device='cuda', dtype=torch.uint8)
        valid[:, S // 2:] = 0
        m2 = valid[:, None, None, :].expand(B, H, S, S) \
            .reshape(B * H, S, S).contiguous()
    else:
        m2 = None
    out = onn.attention_core(qkv, m2, H, temp)
    # fp32 oracle: explicit head-split composition on CPU
    qkv32 = qkv.detach().float().cpu().requires_grad_(True)
    q, k, v = qkv32.split(U, dim=-1)
    def heads(z):
        return z.reshape(B, S, H, D).permute(0, 2, 1, 3) \
                .reshape(B * H, S, D)
    qh, kh, vh = heads(q), heads(k), heads(v)
    sc = qh @ kh.transpose(1, 2) / temp
    if masked:
        sc = sc.masked_fill(m2.cpu() == 0, float('-inf'))
    att = torch.softmax(sc, dim=-1)
    if masked:
        att = torch.nan_to_num(att)
    ref = (att @ vh).reshape(B, H, S, D).permute(0, 2, 1, 3) \
        .reshape(B, S, U)
    check(out, ref.to(out.device), tol=3e-3)
    # backward parity
    dy = torch.randn_like(ref)
    ref.backward(dy)
    out.backward(dy.to(out.device).to(out.dtype))
    check(qkv.grad, qkv32.grad.to(out.device), tol=6e-3)


def test_rtc_launch_gpu():
    """mx.rtc kernel launch on the MI355X via hipModuleLaunchKernel."""
    import mxnet_amd as mx
    mod = mx.rtc.HipModule(
        'extern "C" __global__ void axpy(const float* x, float* y,'
        ' float a, int n) { int i = blockIdx.x * blockDim.x + threadIdx.x;'
        ' if (i < n) y[i] += a * x[i]; }')
    k = mod.get_kernel('axpy', 'const float *x, float *y, float a, int n')
    x = torch.randn(1000, device='cuda')
    y = torch.randn(1000, device='cuda')
    y0 = y.clone()
    k.launch((x, y, 3.0, 1000), mx.gpu(0), ((1000 + 255) // 256, 1, 1),
             (256, 1, 1))
    torch.cuda.synchronize()
    assert torch.allclose(y, y0 + 3.0 * x, atol=1e-5)


def test_conv_bn_fused_presums(monkeypatch, request):
    """Conv epilogue's fused per-channel {sum, ssq} matches a direct
    reduction of the conv output (BN forward-reduce fusion; default-off
    -- measured slower end-to-end -- so enabled explicitly here)."""
    from mxnet_amd.ops import nn as onn
    monkeypatch.setenv('MXNET_BN_FUSED_STATS', '1')
    onn._bn_fused_stats.cache_clear()
    request.addfinalizer(onn._bn_fused_stats.cache_clear)
    torch.manual_seed(44)
    for shape in [((4, 16, 16, 64), (64, 3, 3, 64), (1, 1)),   # igemm
                  ((4, 14, 14, 128), (256, 1, 1, 128), (1, 0))]:  # gemm 1x1
        xs, ws, (st_, pd_) = shape
        x = torch.randn(*xs, device='cuda', dtype=torch.float16)
        w = (torch.randn(*ws, device='cuda', dtype=torch.float16) * 0.1) \
            .requires_grad_(True)
        y = onn.conv2d(x, w, None, (st_, st_), (pd_, pd_), (1, 1), 1,
                       layout='NHWC')
        st = getattr(y, '_bn_presums', None)
        assert st is not None and st.numel(), 'fused stats missing'
        folded = st.sum(0)
        yf = y.detach().float().reshape(-1, y.shape[-1])
        ref_sum = yf.sum(0)
        ref_ssq = (yf * yf).sum(0)
        assert torch.allclose(folded[0], ref_sum, rtol=2e-2, atol=2.0), \
            (folded[0] - ref_sum).abs().max()
        assert torch.allclose(folded[1], ref_ssq, rtol=2e-2, atol=2.0), \
            (folded[1] - ref_ssq).abs().max()


def test_bn_with_presums_matches_plain():
    """bn_nhwc_fwd_train(presums) == bn_nhwc_fwd_train() (same stats)."""
    torch.manual_seed(45)
    M, C = 512, 64
    x = torch.randn(M, 1, 1, C, device='cuda', dtype=torch.float16)
    g = torch.ones(C, device='cuda')
    b = torch.zeros(C, device='cuda')
    rm = torch.zeros(C, device='cuda')
    rv = torch.ones(C, device='cuda')
    y0, m0, i0, _ = ext.bn_nhwc_fwd_train(x, g, b, rm.clone(), rv.clone(),
                                          0.9, 1e-5, False, x.new_empty(0))
    xf = x.float().reshape(-1, C)
    pre = torch.zeros(64, 2, C, device='cuda')
    pre[0, 0] = xf.sum(0)
    pre[0, 1] = (xf * xf).sum(0)
    y1, m1, i1, _ = ext.bn_nhwc_fwd_train(x, g, b, rm.clone(), rv.clone(),
                                          0.9, 1e-5, False, x.new_empty(0),
                                          pre)
    assert torch.allclose(m0, m1, rtol=1e-3, atol=1e-3)
    assert torch.allclose(i0, i1, rtol=1e-3, atol=1e-3)
    assert torch.allclose(y0.float(), y1.float(), rtol=2e-2, atol=2e-2)


def test_rtc_fusion_gpu():
    """Fused elementwise chain runs as ONE hiprtc kernel on GPU and
    matches the eager composition."""
    import json
    import mxnet_amd as mx
    from mxnet_amd import symbol as S
    from mxnet_amd.symbol.subgraph import partition_graph
    x = S.var('x')
    y = S.var('y')
    z = S.Activation(x * 2.0 + y, act_type='relu') * 0.5
    ps = partition_graph(z)
    for dt in (torch.float32, torch.float16):
        xt = torch.randn(1000, device='cuda', dtype=dt)
        yt = torch.randn(1000, device='cuda', dtype=dt)
        with torch.no_grad():
            out = ps.eval(x=mx.nd.from_torch(xt), y=mx.nd.from_torch(yt))[0]
        torch.cuda.synchronize()
        ref = torch.relu(xt.float() * 2 + yt.float()) * 0.5
        assert torch.allclose(out.handle.float(), ref, atol=2e-3)


def test_adam_optimizer_fused_path():
    """Adam.update_multi_precision routes to the fused kernel and tracks
    the eager fp32 composition across steps (incl. master weights)."""
    import mxnet_amd as mx
    from mxnet_amd import optimizer as opt
    from mxnet_amd.ndarray.ndarray import NDArray
    torch.manual_seed(9)
    w0 = torch.randn(1000)
    g_seq = [torch.randn(1000) * 0.1 for _ in range(4)]

    # GPU fused, fp16 weight + fp32 master
    o = opt.create('adam', learning_rate=0.01, wd=0.01, multi_precision=True)
    w = NDArray(w0.clone().half().cuda())
    st = o.create_state_multi_precision(0, w)
    for g in g_seq:
        o.update_multi_precision(0, w, NDArray(g.half().cuda()), st)
    torch.cuda.synchronize()

    # CPU fp32 oracle through the eager update
    o2 = opt.create('adam', learning_rate=0.01, wd=0.01)
    w2 = NDArray(w0.clone())
    st2 = o2.create_state(0, w2)
    for g in g_seq:
        o2.update(0, w2, NDArray(g.clone()), st2)

    err = (st[0].cpu() - w2.handle).abs().max()
    assert float(err) < 2e-3, float(err)


@pytest.mark.parametrize('shape', [(512, 64, 72), (4096, 768, 768),
                                   (1000, 96, 40), (130, 64, 64)])
def test_gemm_tn_fused(shape):
    """Direct TN GEMM (both operands via tr_b16 reads) + fused bias
    gradient vs fp32 oracle."""
    M, I, J = shape
    a = mk((M, I), seed=70, scale=0.5)
    b = mk((M, J), seed=71, scale=0.5)
    C, db = ext.gemm_tn_fused(a, b, True)
    ref = a.float().t() @ b.float()
    check(C, ref, tol=4e-3)
    ref_db = a.float().sum(0)
    assert torch.allclose(db, ref_db, rtol=2e-2, atol=1.5), \
        (db - ref_db).abs().max()


def test_fc_backward_tn_path():
    """FullyConnected backward produces dw/db through the TN kernel."""
    from mxnet_amd.ops import nn as onn
    x = mk((256, 128), seed=72).requires_grad_(True)
    w = mk((64, 128), seed=73, scale=0.5).requires_grad_(True)
    bb = mk((64,), seed=74).requires_grad_(True)
    y = onn.fully_connected(x, w, bb)
    dy = torch.randn_like(y)
    y.backward(dy)
    xo = x.detach().float().cpu().requires_grad_(True)
    wo = w.detach().float().cpu().requires_grad_(True)
    bo = bb.detach().float().cpu().requires_grad_(True)
    yo = torch.nn.functional.linear(xo, wo, bo)
    yo.backward(dy.float().cpu())
    check(w.grad, wo.grad.to(w.device), tol=4e-3)
    check(bb.grad, bo.grad.to(w.device), tol=4e-3)
    check(x.grad, xo.grad.to(w.device), tol=4e-3)


@pytest.mark.parametrize('C', [64, 128, 200])
def test_softmax_rowreg_masked(C):
    """Register-resident small-C softmax (one-read path) with mask."""
    rows = 512
    x = mk((rows, C), seed=75, scale=3.0)
    mask = (torch.rand(rows, C, device='cuda') > 0.3).to(torch.uint8)
    mask[:, 0] = 1  # never fully masked
    y = ext.softmax_fwd(x, False, 1.3, mask)
    xm = x.float().masked_fill(mask == 0, float('-inf'))
    want = torch.softmax(xm / 1.3, -1)
    check(y, want, tol=3e-3)
    dy = torch.randn_like(want)
    x32 = xm.clone().requires_grad_(True)
    yo = torch.softmax(x32 / 1.3, -1)
    yo.backward(dy)
    dx = ext.softmax_bwd(dy.to(x.dtype), y, False, 1.3)
    check(dx, torch.nan_to_num(x32.grad), tol=4e-3)
