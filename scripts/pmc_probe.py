"""Minimal workload for PMC counter collection on the hot kernels
(conv bwd_weight, conv fwd, BN bwd): a few launches each, big shapes."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from mxnet_amd import _hipops as ext

DEV = 'cuda:0'
B = 256
x = torch.randn(B, 28, 28, 128, device=DEV).half()
w = (torch.randn(128, 3, 3, 128, device=DEV) * 0.05).half()
dy = torch.randn(B, 28, 28, 128, device=DEV).half()
for _ in range(5):
    ext.conv2d_nhwc_bwd_weight(dy, x, 3, 3, 1, 1, 1, 1, 1, 1, 1)
    ext.conv2d_nhwc_fwd(x, w, None, 1, 1, 1, 1, 1, 1, 1)
gamma = torch.randn(128, device=DEV).abs() + 0.5
beta = torch.randn(128, device=DEV)
rm, rv = torch.zeros(128, device=DEV), torch.ones(128, device=DEV)
y, sm, si, mask = ext.bn_nhwc_fwd_train(x, gamma, beta, rm, rv, 0.9, 1e-5,
                                        True, dy)
for _ in range(5):
    ext.bn_nhwc_bwd(dy, x, gamma, sm, si, True, y, True, mask)
a = torch.randn(4096, 4096, device=DEV).half()
b = torch.randn(4096, 4096, device=DEV).half()
for _ in range(5):
    ext.gemm_nt(a, b, None)
torch.cuda.synchronize()
print('pmc probe done')
