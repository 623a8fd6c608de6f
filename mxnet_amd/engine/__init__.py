"""Execution engine (reference include/mxnet/engine.h + src/engine/).

MI355X-native split of responsibilities:

* GPU ops are asynchronous by construction — every kernel launch goes to
  a HIP stream; ordering between ops on one device comes from stream
  program order, cross-stream/cross-device ordering from HIP events.
  That replaces the reference ThreadedEnginePerDevice's per-GPU worker
  threads (threaded_engine_perdevice.cc:49): the hardware queue IS the
  worker.
* The C++ dependency engine (``mxnet_amd._engine``, src/engine.cc)
  provides the reference's Var/Opr dependency-tracking semantics for
  HOST-side async work: data-pipeline stages, checkpoint IO, KVStore
  CPU reduces.  It exposes NewVariable/PushAsync/WaitForVar/WaitForAll
  with read/write sets, per-device worker pools, and exception
  propagation exactly like threaded_engine.cc.

``waitall()``/``wait_for_var`` bridge both worlds.
"""
import os

import torch

try:
    from mxnet_amd import _engine as _cpp  # built by setup.py
except ImportError:
    _cpp = None


class _StreamPool:
    """Per-device HIP streams: compute (default), copy, priority/comm.

    Mirrors ThreadedEnginePerDevice's queues (normal / copy / priority,
    threaded_engine_perdevice.cc:97-106) as streams instead of threads.
    """

    def __init__(self):
        self._copy = {}
        self._comm = {}

    def copy_stream(self, dev):
        if dev not in self._copy:
            self._copy[dev] = torch.cuda.Stream(device=dev)
        return self._copy[dev]

    def comm_stream(self, dev):
        if dev not in self._comm:
            self._comm[dev] = torch.cuda.Stream(device=dev, priority=-1)
        return self._comm[dev]


streams = _StreamPool()

_py_engine = None


def get():
    """The host-side dependency engine (C++ if built, python fallback)."""
    global _py_engine
    if _cpp is not None:
        return _cpp.get()
    if _py_engine is None:
        _py_engine = _PyNaiveEngine()
    return _py_engine


class _PyNaiveEngine:
    """Synchronous fallback mirroring NaiveEngine (naive_engine.cc:53)."""

    def new_variable(self):
        return object()

    def push(self, fn, const_vars=(), mutable_vars=()):
        fn()

    def wait_for_var(self, var):
        pass

    def wait_for_all(self):
        pass

    def stop(self):
        pass

    def start(self):
        pass


def wait_for_all():
    get().wait_for_all()
    if torch.cuda.is_available():
        for i in range(torch.cuda.device_count()):
            torch.cuda.synchronize(i)


def engine_type():
    if _cpp is not None:
        return os.environ.get('MXNET_ENGINE_TYPE', 'ThreadedEnginePerDevice')
    return 'NaiveEngine'


# -- fork safety (reference LibraryInitializer pthread_atfork handlers,
# src/initialize.cc:71-83: stop engine threads around fork so DataLoader
# worker processes never inherit held queue mutexes) -------------------
def _before_fork():
    try:
        if _cpp is not None:
            eng = _cpp.get()
            eng.wait_for_all()
            eng.stop()
    except Exception:
        pass


def _after_fork_parent():
    try:
        if _cpp is not None:
            _cpp.get().start()
    except Exception:
        pass


def _after_fork_child():
    # child gets a fresh worker pool; inherited thread state is gone
    try:
        if _cpp is not None:
            _cpp.get().start()
    except Exception:
        pass


os.register_at_fork(before=_before_fork,
                    after_in_parent=_after_fork_parent,
                    after_in_child=_after_fork_child)
