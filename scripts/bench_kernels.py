"""Per-kernel microbenchmarks on ResNet-50 batch-256 shapes.

Prints op, shape, time, and effective TFLOP/s / TB/s so kernel work can
be prioritized (run on MI355X via gpurun)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from mxnet_amd.ops import hipshim as ext

DEV = 'cuda:0'
B = int(os.environ.get('BENCH_BATCH', 256))

# (name, H, W, C, K, R, S, stride)  -- one per distinct ResNet-50 conv shape
CONVS = [
    ('stem7x7', 224, 224, 3, 64, 7, 7, 2),
    ('l1_1x1a', 56, 56, 64, 64, 1, 1, 1),
    ('l1_3x3', 56, 56, 64, 64, 3, 3, 1),
    ('l1_1x1b', 56, 56, 64, 256, 1, 1, 1),
    ('l1_1x1c', 56, 56, 256, 64, 1, 1, 1),
    ('l2_ds', 56, 56, 256, 512, 1, 1, 2),
    ('l2_3x3s2', 56, 56, 128, 128, 3, 3, 2),
    ('l2_3x3', 28, 28, 128, 128, 3, 3, 1),
    ('l2_1x1b', 28, 28, 128, 512, 1, 1, 1),
    ('l3_3x3', 14, 14, 256, 256, 3, 3, 1),
    ('l3_1x1b', 14, 14, 256, 1024, 1, 1, 1),
    ('l4_3x3', 7, 7, 512, 512, 3, 3, 1),
    ('l4_1x1b', 7, 7, 512, 2048, 1, 1, 1),
]


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    print(f'batch={B}')
    print(f'{"op":26s} {"us":>9s} {"TF/s":>7s} {"TB/s":>6s}')
    for name, H, W, C, K, R, S, st in CONVS:
        pd = (R - 1) // 2
        P, Q = (H + 2 * pd - R) // st + 1, (W + 2 * pd - S) // st + 1
        x = torch.randn(B, H, W, C, device=DEV).half()
        w = (torch.randn(K, R, S, C, device=DEV) * 0.05).half()
        dy = torch.randn(B, P, Q, K, device=DEV).half()
        flops = 2.0 * B * P * Q * K * R * S * C
        bytes_fwd = 2.0 * (B * H * W * C + B * P * Q * K + K * R * S * C)

        t = timeit(lambda: ext.conv2d_nhwc_fwd(x, w, None, st, st, pd, pd, 1, 1, 1))
        print(f'{name+"/fwd":26s} {t*1e6:9.1f} {flops/t/1e12:7.1f} {bytes_fwd/t/1e12:6.2f}')
        t = timeit(lambda: ext.conv2d_nhwc_bwd_data(dy, w, H, W, st, st, pd, pd, 1, 1, 1))
        print(f'{name+"/bwd_data":26s} {t*1e6:9.1f} {flops/t/1e12:7.1f} {bytes_fwd/t/1e12:6.2f}')
        t = timeit(lambda: ext.conv2d_nhwc_bwd_weight(dy, x, R, S, st, st, pd, pd, 1, 1, 1))
        print(f'{name+"/bwd_w":26s} {t*1e6:9.1f} {flops/t/1e12:7.1f} {bytes_fwd/t/1e12:6.2f}')
        del x, w, dy

    # BN shapes
    for name, HW, C in [('bn_l1', 56, 256), ('bn_l2', 28, 512),
                        ('bn_l3', 14, 1024), ('bn_s1', 56, 64)]:
        x = torch.randn(B, HW, HW, C, device=DEV).half()
        res = torch.randn_like(x)
        dy = torch.randn_like(x)
        gamma = torch.randn(C, device=DEV).abs() + 0.5
        beta = torch.randn(C, device=DEV)
        rm = torch.zeros(C, device=DEV)
        rv = torch.ones(C, device=DEV)
        nbytes = x.numel() * 2
        t = timeit(lambda: ext.bn_nhwc_fwd_train(x, gamma, beta, rm, rv, 0.9,
                                                 1e-5, True, res))
        print(f'{name+"/fwd(3x)":26s} {t*1e6:9.1f} {"":>7s} {3*nbytes/t/1e12:6.2f}')
        y, sm, si, _m = ext.bn_nhwc_fwd_train(x, gamma, beta, rm, rv, 0.9, 1e-5,
                                          True, res)
        t = timeit(lambda: ext.bn_nhwc_bwd(dy, x, gamma, sm, si, True, y, True))
        print(f'{name+"/bwd(8x)":26s} {t*1e6:9.1f} {"":>7s} {8*nbytes/t/1e12:6.2f}')
        del x, res, dy, y

    # GEMM sanity: big square
    for MNK in [(4096, 4096, 4096), (8192, 8192, 8192)]:
        M, N, K = MNK
        a = torch.randn(M, K, device=DEV).half()
        b = torch.randn(N, K, device=DEV).half()
        t = timeit(lambda: ext.gemm_nt(a, b, None))
        print(f'gemm_nt {M}x{N}x{K}: {t*1e6:9.1f}us {2.0*M*N*K/t/1e12:7.1f} TF/s')
        del a, b


if __name__ == '__main__':
    main()
