"""Build the host-side C++ helper extensions (engine/dataloader pybind).

The GPU kernel library ``mxnet_amd._core`` is built by the top-level
Makefile with hipcc alone (``make`` — no torch toolchain, no hipify);
this setup.py only covers the two pure-CPU pybind11 modules.
"""
import os

from setuptools import setup
from pybind11.setup_helpers import Pybind11Extension

ROOT = os.path.dirname(os.path.abspath(__file__))

setup(
    name='mxnet_amd_host',
    ext_modules=[
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
        Pybind11Extension(
            'mxnet_amd._imageio',
            [os.path.join(ROOT, 'src', 'imageio.cc')],
            cxx_std=17,
            extra_compile_args=['-O3', '-pthread'],
        ),
    ],
)
