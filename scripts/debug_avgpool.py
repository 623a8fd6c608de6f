"""Locate the avg-pool backward mismatch pattern on device."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from mxnet_amd import _hipops as ext

N, H, W, C = 2, 13, 13, 24
torch.manual_seed(40)
x = torch.randn(N, H, W, C, device='cuda:0').half()
xn = x.float().permute(0, 3, 1, 2).requires_grad_(True)
yo = torch.nn.functional.avg_pool2d(xn, 3, 2, 1, count_include_pad=False)
dy_o = torch.randn_like(yo)
yo.backward(dy_o)
dy = dy_o.permute(0, 2, 3, 1).contiguous().half()
arg = torch.empty(0, dtype=torch.int32, device='cuda:0')
dx = ext.pool_nhwc_bwd(dy, arg, 'avg', H, W, 3, 3, 2, 2, 1, 1, False)
want = xn.grad.permute(0, 2, 3, 1)
err = (dx.float() - want).abs()
bad = (err > 0.02).nonzero()
print('num bad', bad.shape[0], 'of', err.numel())
for row in bad[:20]:
    n, h, w, c = [int(v) for v in row]
    print(f'n{n} h{h} w{w} c{c}: got {dx[n,h,w,c].item():.4f} '
          f'want {want[n,h,w,c].item():.4f}')
# which (h,w) positions are bad, aggregated
if bad.shape[0]:
    hw = {}
    for row in bad:
        key = (int(row[1]), int(row[2]))
        hw[key] = hw.get(key, 0) + 1
    print('bad (h,w) histogram:', sorted(hw.items())[:30])
