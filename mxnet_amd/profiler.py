"""Profiler (reference python/mxnet/profiler.py + src/profiler/).

MI355X-native: op/kernel timing comes from HIP events via the torch ROCm
profiler (kineto -> roctracer), dumped in the same Chrome-tracing JSON the
reference emits (profiler.h:256 Profiler::DumpProfile).  For kernel-level
counters use rocprofv3 on the process (see profiles/ in the repo).
"""
import atexit
import json
import time

import torch

_state = {'config': {}, 'running': False, 'prof': None, 'records': []}


def set_config(**kwargs):
    """profile_all / profile_symbolic / profile_imperative / filename ..."""
    _state['config'].update(kwargs)


def set_state(state='stop', profile_process='worker'):
    # native-runtime engine profiler rides along (per-op aggregate stats,
    # reference src/profiler AggregateStats)
    try:
        from . import _core
        _core.profiler_set_state(state == 'run')
    except Exception:
        pass
    if state == 'run' and not _state['running']:
        activities = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(torch.profiler.ProfilerActivity.CUDA)
        _state['prof'] = torch.profiler.profile(activities=activities)
        _state['prof'].__enter__()
        _state['running'] = True
    elif state == 'stop' and _state['running']:
        _state['prof'].__exit__(None, None, None)
        _state['running'] = False


def start():
    set_state('run')


def stop():
    set_state('stop')


def pause(profile_process='worker'):
    pass


def resume(profile_process='worker'):
    pass


def dump(finished=True, profile_process='worker'):
    """Write Chrome-tracing JSON to the configured filename."""
    fname = _state['config'].get('filename', 'profile.json')
    prof = _state['prof']
    if prof is not None:
        prof.export_chrome_trace(fname)
    else:
        with open(fname, 'w') as f:
            json.dump({'traceEvents': []}, f)
    return fname


def native_summary(top=30):
    """Aggregate per-op table from the native engine profiler
    (op name, calls, total ms) — reference profiler aggregate stats."""
    from . import _core
    rows = _core.profiler_summary()
    lines = ['%-32s %8s %12s' % ('op', 'calls', 'total_ms')]
    for name, calls, ms in rows[:top]:
        lines.append('%-32s %8d %12.3f' % (name or '<unnamed>', calls, ms))
    return '\n'.join(lines)


def dumps(reset=False):
    """Aggregate stats table as a string (reference aggregate_stats.cc):
    the native engine's per-op table when the native runtime is active,
    else the kineto key-averages table."""
    from .base import native_mode
    if native_mode():
        return native_summary()
    prof = _state['prof']
    if prof is None:
        return ''
    return prof.key_averages().table(sort_by='self_cuda_time_total'
                                     if torch.cuda.is_available()
                                     else 'self_cpu_time_total')


class Scope:
    """Named profiling scope -> roctx-style range."""

    def __init__(self, name='<unk>'):
        self.name = name

    def __enter__(self):
        if torch.cuda.is_available():
            torch.cuda.nvtx.range_push(self.name)
        return self

    def __exit__(self, *a):
        if torch.cuda.is_available():
            torch.cuda.nvtx.range_pop()


scope = Scope


class Task:
    def __init__(self, domain=None, name='task'):
        self.name = name
        self._t0 = None

    def start(self):
        self._t0 = time.time()

    def stop(self):
        _state['records'].append((self.name, time.time() - self._t0))


Frame = Task
Event = Task


class Counter:
    def __init__(self, domain=None, name='counter', value=0):
        self.name = name
        self.value = value

    def set_value(self, v):
        self.value = v

    def increment(self, d=1):
        self.value += d

    def decrement(self, d=1):
        self.value -= d


class Domain:
    def __init__(self, name='domain'):
        self.name = name


# ---------------------------------------------------------------------------
# GPU memory profiler (reference src/profiler/storage_profiler.{h,cc}:
# per-scope Alloc/Free accounting dumped as a usage table)
# ---------------------------------------------------------------------------
_mem_scopes = {}


class profile_scope:
    """Tag allocations in a region (reference AssignStorageInfo)."""

    def __init__(self, name):
        self.name = name

    def __enter__(self):
        if torch.cuda.is_available():
            self._start = torch.cuda.memory_allocated()
        return self

    def __exit__(self, *a):
        if torch.cuda.is_available():
            delta = torch.cuda.memory_allocated() - self._start
            cur = _mem_scopes.get(self.name, 0)
            _mem_scopes[self.name] = cur + delta


def memory_stats(device=None):
    """Allocator counters (reference storage_profiler dump)."""
    if not torch.cuda.is_available():
        return {}
    s = torch.cuda.memory_stats(device)
    out = {
        'allocated_bytes': s.get('allocated_bytes.all.current', 0),
        'allocated_peak': s.get('allocated_bytes.all.peak', 0),
        'reserved_bytes': s.get('reserved_bytes.all.current', 0),
        'reserved_peak': s.get('reserved_bytes.all.peak', 0),
        'num_allocs': s.get('allocation.all.allocated', 0),
    }
    out['scopes'] = dict(_mem_scopes)
    return out


def dump_memory_profile(fname='gpu_memory_profile.csv', device=None):
    stats = memory_stats(device)
    with open(fname, 'w') as f:
        f.write('entry,bytes\n')
        for k, v in stats.items():
            if k != 'scopes':
                f.write(f'{k},{v}\n')
        for k, v in stats.get('scopes', {}).items():
            f.write(f'scope:{k},{v}\n')
    return fname
