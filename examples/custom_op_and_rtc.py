"""Extension-point walk-through: a python CustomOp inside autograd, a
runtime-compiled HIP kernel (mx.rtc), and the pointwise-fusion pass.

Run: PYTHONPATH=. python examples/custom_op_and_rtc.py
(the rtc kernel launch needs a GPU; compilation works anywhere)
"""
import json

import torch

import mxnet_amd as mx
from mxnet_amd import autograd, operator as op
from mxnet_amd import symbol as S
from mxnet_amd.symbol.subgraph import partition_graph
from mxnet_amd.contrib.fusion import codegen


@op.register("swish")
class SwishProp(op.CustomOpProp):
    def create_operator(self, ctx, shapes, dtypes):
        class Swish(op.CustomOp):
            def forward(self, is_train, req, in_data, out_data, aux):
                x = in_data[0].handle
                self.assign(out_data[0], req[0], x * torch.sigmoid(x))

            def backward(self, req, out_grad, in_data, out_data, in_grad,
                         aux):
                x = in_data[0].handle
                s = torch.sigmoid(x)
                self.assign(in_grad[0], req[0],
                            out_grad[0].handle * (s + x * s * (1 - s)))
        return Swish()


def main():
    # 1. CustomOp with autograd
    x = mx.nd.from_torch(torch.randn(4, 5))
    x.attach_grad()
    with autograd.record():
        y = mx.nd.Custom(x, op_type="swish")
        loss = mx.nd.from_torch(y.handle.sum())
    loss.backward()
    print('CustomOp swish: out mean %.4f, grad mean %.4f'
          % (float(y.handle.mean()), float(x.grad.handle.mean())))

    # 2. runtime HIP compilation (hiprtc) — compiles on any box
    mod = mx.rtc.HipModule(
        'extern "C" __global__ void saxpy(const float* x, float* y, '
        'float a, int n) { int i = blockIdx.x * blockDim.x + threadIdx.x; '
        'if (i < n) y[i] += a * x[i]; }')
    kern = mod.get_kernel('saxpy', 'const float* x, float* y, float a, int n')
    print('hiprtc compile OK; kernel arg types:', kern._types)
    if torch.cuda.is_available():
        xs = torch.ones(1024, device='cuda')
        ys = torch.zeros(1024, device='cuda')
        kern.launch((xs, ys, 2.0, 1024), mx.gpu(0), (4, 1, 1), (256, 1, 1))
        torch.cuda.synchronize()
        print('launch OK, y[0] =', float(ys[0]))

    # 3. pointwise fusion: elementwise chain -> ONE generated kernel
    a, b = S.var('a'), S.var('b')
    expr = S.Activation(a * 2.0 + b, act_type='relu') * 0.5
    fused = partition_graph(expr)
    node = [n for n in json.loads(fused.tojson())['nodes']
            if n['op'] == '_fused_subgraph'][0]
    src, n_in = codegen(node['attrs']['subgraph'])
    print(f'fused {node["attrs"]["ops"]} -> 1 kernel ({n_in} inputs)')
    out = fused.eval(a=mx.nd.from_torch(torch.ones(3)),
                     b=mx.nd.from_torch(torch.ones(3)))[0]
    print('fused eval:', out.handle.tolist())


if __name__ == '__main__':
    main()
