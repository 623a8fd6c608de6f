"""Device context, modeled on the reference's mxnet.context
(/root/reference/python/mxnet/context.py) but mapped onto ROCm/HIP devices:
``gpu(i)`` is HIP device i on an MI355X node; there is no separate
"cpu_pinned" device type — pinned host memory is an allocation flag.
"""
import threading

import torch

_CTX_STACK = threading.local()


class Context:
    """A device context (cpu / gpu).

    Reference parity: mxnet.context.Context (context.py:55).  devtype ids
    keep the reference numbering so serialized NDArrays (Appendix A of
    SURVEY.md) stay byte-compatible: cpu=1, gpu=2, cpu_pinned=3, cpu_shared=5.
    """

    devtype2str = {1: 'cpu', 2: 'gpu', 3: 'cpu_pinned', 5: 'cpu_shared'}
    devstr2type = {v: k for k, v in devtype2str.items()}

    def __init__(self, device_type, device_id=0):
        if isinstance(device_type, Context):
            self.device_typeid = device_type.device_typeid
            self.device_id = device_type.device_id
        else:
            self.device_typeid = Context.devstr2type[device_type]
            self.device_id = device_id

    @property
    def device_type(self):
        return Context.devtype2str[self.device_typeid]

    def __hash__(self):
        return hash((self.device_typeid, self.device_id))

    def __eq__(self, other):
        return (isinstance(other, Context)
                and self.device_typeid == other.device_typeid
                and self.device_id == other.device_id)

    def __str__(self):
        return '%s(%d)' % (self.device_type, self.device_id)

    __repr__ = __str__

    def __enter__(self):
        if not hasattr(_CTX_STACK, 'stack'):
            _CTX_STACK.stack = []
        _CTX_STACK.stack.append(self)
        return self

    def __exit__(self, *args):
        _CTX_STACK.stack.pop()

    # --- torch interop -------------------------------------------------
    @property
    def torch_device(self):
        if self.device_type == 'gpu':
            return torch.device('cuda', self.device_id)
        return torch.device('cpu')

    @classmethod
    def from_torch(cls, dev):
        if dev.type == 'cuda':
            return cls('gpu', dev.index if dev.index is not None else 0)
        return cls('cpu', 0)


def cpu(device_id=0):
    return Context('cpu', device_id)


def cpu_pinned(device_id=0):
    return Context('cpu_pinned', device_id)


def gpu(device_id=0):
    return Context('gpu', device_id)


def current_context():
    if getattr(_CTX_STACK, 'stack', None):
        return _CTX_STACK.stack[-1]
    return Context('cpu', 0)


def num_gpus():
    return torch.cuda.device_count() if torch.cuda.is_available() else 0


def gpu_memory_info(device_id=0):
    free, total = torch.cuda.mem_get_info(device_id)
    return (free, total)
