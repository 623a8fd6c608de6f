"""Runtime-loadable external operator libraries (reference
python/mxnet/library.py mx.library.load -> MXLoadLib,
include/mxnet/lib_api.h).  The .so exports ``mxnet_amd_lib_init`` and
registers MXCustomOpDef entries; loaded ops are invokable through the
native registry like built-ins."""
import ctypes
import os

__all__ = ['load']


def load(path):
    from . import _core
    lib = ctypes.CDLL(_core.__file__, mode=ctypes.RTLD_GLOBAL)
    lib.MXLoadLib.argtypes = [ctypes.c_char_p]
    lib.MXLoadLib.restype = ctypes.c_int
    if lib.MXLoadLib(os.fspath(path).encode()) != 0:
        lib.MXGetLastError.restype = ctypes.c_char_p
        raise RuntimeError('MXLoadLib failed: %s'
                           % lib.MXGetLastError().decode())
