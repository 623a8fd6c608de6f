"""HIP kernel loader / dispatch policy.

The hot operator set (SURVEY.md §2.2 "NN core") runs hand-written gfx950
HIP kernels from the native in-tree extension ``mxnet_amd/_core*.so``
(built by hipcc alone — no torch toolchain), fronted for torch tensors by
``mxnet_amd.ops.hipshim``.  On a GPU box the extension is REQUIRED: ops
raise instead of silently falling back to eager PyTorch, so a passing GPU
test means the native path ran.  CPU tensors use plain PyTorch fp32 ops —
they are the numerics oracle (reference test strategy: check_consistency,
test_utils.py:1490).

Set MXNET_FORCE_EAGER=1 to bypass HIP kernels (debugging only).
"""
import os

_hipops = None
_tried = False


def hipops():
    """Return the native extension module, importing it on first use."""
    global _hipops, _tried
    if not _tried:
        _tried = True
        try:
            from mxnet_amd.ops import hipshim as ext  # over _core (in-tree)
            _hipops = ext
        except ImportError:
            _hipops = None
    return _hipops


def hip_required(opname):
    """Fetch the extension for a GPU op; raise loudly if missing."""
    ext = hipops()
    if ext is None:
        raise RuntimeError(
            f"mxnet_amd: op '{opname}' needs the native HIP extension "
            f"(mxnet_amd/_core*.so) but it is not built. Run "
            f"`make` (hipcc, gfx950).")
    return ext


def use_hip(tensor):
    if os.environ.get('MXNET_FORCE_EAGER', '0') == '1':
        return False
    return tensor.is_cuda
