"""Runtime HIP kernel compilation (reference mx.rtc.CudaModule,
include/mxnet/rtc.h:39 + src/common/rtc.cc:49 — NVRTC there, hiprtc
here).

``HipModule`` (alias ``CudaModule`` for API parity) compiles user HIP
source with hiprtc for gfx950 and launches kernels on the current torch
HIP stream via hipModuleLaunchKernel — both through ctypes, no build
step.
"""
import ctypes
import re

import torch

__all__ = ['HipModule', 'CudaModule']

_hiprtc = None
_hip = None


def _libs():
    global _hiprtc, _hip
    if _hiprtc is None:
        _hiprtc = ctypes.CDLL('libhiprtc.so')
        _hip = ctypes.CDLL('libamdhip64.so')
    return _hiprtc, _hip


def _check(rc, what):
    if rc != 0:
        raise RuntimeError(f'{what} failed with code {rc}')


class HipModule:
    """Compile HIP source once; ``get_kernel(name, signature)`` returns a
    launchable kernel.

    Example::

        mod = mx.rtc.HipModule(r'''
            extern "C" __global__ void axpy(const float *x, float *y,
                                            float alpha, int n) {
                int i = blockIdx.x * blockDim.x + threadIdx.x;
                if (i < n) y[i] += alpha * x[i];
            }''')
        k = mod.get_kernel("axpy", "const float *x, float *y, float alpha, int n")
        k.launch((x, y, 3.0, x.size), mx.gpu(0), grid, block)
    """

    def __init__(self, source, options=(), exports=()):
        rtc, hip = _libs()
        prog = ctypes.c_void_p()
        _check(rtc.hiprtcCreateProgram(
            ctypes.byref(prog), source.encode(), b'mxnet_rtc.cu', 0, None,
            None), 'hiprtcCreateProgram')
        opts = [b'--offload-arch=gfx950'] + [o.encode() for o in options]
        arr = (ctypes.c_char_p * len(opts))(*opts)
        rc = rtc.hiprtcCompileProgram(prog, len(opts), arr)
        if rc != 0:
            sz = ctypes.c_size_t()
            rtc.hiprtcGetProgramLogSize(prog, ctypes.byref(sz))
            buf = ctypes.create_string_buffer(sz.value)
            rtc.hiprtcGetProgramLog(prog, buf)
            raise RuntimeError('hiprtc compile failed:\n' +
                               buf.value.decode(errors='replace'))
        sz = ctypes.c_size_t()
        _check(rtc.hiprtcGetCodeSize(prog, ctypes.byref(sz)),
               'hiprtcGetCodeSize')
        code = ctypes.create_string_buffer(sz.value)
        _check(rtc.hiprtcGetCode(prog, code), 'hiprtcGetCode')
        rtc.hiprtcDestroyProgram(ctypes.byref(prog))
        self._code = code
        self._module = None  # loaded lazily on first launch (needs a GPU)

    def _load(self):
        if self._module is None:
            _, hip = _libs()
            mod = ctypes.c_void_p()
            _check(hip.hipModuleLoadData(ctypes.byref(mod), self._code),
                   'hipModuleLoadData')
            self._module = mod
        return self._module

    def get_kernel(self, name, signature):
        """signature: C parameter list, e.g. "const float *x, float *y,
        float alpha, int n" (reference CudaModule.get_kernel)."""
        types = []
        for part in signature.split(','):
            part = part.strip()
            if not part:
                continue
            if '*' in part:
                types.append('ptr')
            else:
                base = re.sub(r'\b(const|__restrict__|restrict)\b', '',
                              part).strip().split()
                t = ' '.join(base[:-1]) if len(base) > 1 else base[0]
                types.append({'float': 'f32', 'double': 'f64',
                              'int': 'i32', 'unsigned': 'u32',
                              'unsigned int': 'u32', 'long': 'i64',
                              'long long': 'i64', 'size_t': 'u64'}
                             .get(t, 'i64'))
        return HipKernel(self, name, types)


CudaModule = HipModule  # reference-compatible alias


class HipKernel:
    def __init__(self, module, name, types):
        self._module = module
        self._name = name
        self._types = types
        self._func = None

    def _get_func(self):
        if self._func is None:
            _, hip = _libs()
            fn = ctypes.c_void_p()
            _check(hip.hipModuleGetFunction(
                ctypes.byref(fn), self._module._load(),
                self._name.encode()), 'hipModuleGetFunction')
            self._func = fn
        return self._func

    def launch(self, args, ctx=None, grid_dims=(1, 1, 1),
               block_dims=(256, 1, 1), shared_mem=0):
        """args: NDArrays/torch tensors (device pointers) and python
        scalars, matching the signature order."""
        _, hip = _libs()
        fn = self._get_func()
        c_args = []
        for a, t in zip(args, self._types):
            if t == 'ptr':
                tt = a.handle if hasattr(a, 'handle') else a
                c_args.append(ctypes.c_void_p(tt.data_ptr()))
            elif t == 'f32':
                c_args.append(ctypes.c_float(a))
            elif t == 'f64':
                c_args.append(ctypes.c_double(a))
            elif t == 'i32':
                c_args.append(ctypes.c_int(a))
            elif t == 'u32':
                c_args.append(ctypes.c_uint(a))
            elif t == 'u64':
                c_args.append(ctypes.c_size_t(a))
            else:
                c_args.append(ctypes.c_longlong(a))
        ptrs = (ctypes.c_void_p * len(c_args))(
            *[ctypes.cast(ctypes.byref(x), ctypes.c_void_p)
              for x in c_args])
        stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
        _check(hip.hipModuleLaunchKernel(
            fn, grid_dims[0], grid_dims[1], grid_dims[2],
            block_dims[0], block_dims[1], block_dims[2],
            shared_mem, stream, ptrs, None), 'hipModuleLaunchKernel')
