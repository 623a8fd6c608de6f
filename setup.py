"""Build the in-tree gfx950 HIP extension ``mxnet_amd._hipops``.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Every kernel is hand-written CDNA4 HIP (MFMA / LDS / wave64) — no hipify,
no CUDA sources.  torch.utils.cpp_extension drives hipcc and links
against the PyTorch-ROCm runtime so tensors pass straight through.
"""
import os
import glob

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402
from pybind11.setup_helpers import Pybind11Extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
SRC = sorted(glob.glob(os.path.join(ROOT, 'mxnet_amd', 'ops', 'hip', '*.hip'))) \
    + [os.path.join(ROOT, 'mxnet_amd', 'ops', 'hip', 'bind.cpp')]

setup(
    name='mxnet_amd_hipops',
    ext_modules=[
        CUDAExtension(
            name='mxnet_amd._hipops',
            sources=SRC,
            extra_compile_args={
                'cxx': ['-O3', '-std=c++17'],
                'nvcc': ['-O3', '-std=c++17', '--offload-arch=gfx950'],
            },
        ),
        Pybind11Extension(
            'mxnet_amd._engine',
            [os.path.join(ROOT, 'src', 'engine.cc')],
            cxx_std=17,
            extra_compile_args=['-O2', '-pthread'],
        ),
        Pybind11Extension(
            'mxnet_amd._dataloader',
            [os.path.join(ROOT, 'src', 'dataloader.cc')],
            cxx_std=17,
            extra_compile_args=['-O2', '-pthread'],
        ),
    ],
    cmdclass={'build_ext': BuildExtension.with_options(use_ninja=True)},
)
