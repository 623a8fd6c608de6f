"""KVStore — data-parallel gradient store.

Reference parity: include/mxnet/kvstore.h + src/kvstore/* (SURVEY.md §2.1).
MI355X-native mapping:

* ``local``   — CPU-side reduce (reference CommCPU, comm.h:104).
* ``device``  — single-process multi-GPU reduce with D2D copies over xGMI
  + fused sum, broadcast back (reference CommDevice, comm.h:452).  On a
  single GPU this degenerates to in-place accumulate.
* ``nccl`` / ``dist_sync`` / ``dist_device_sync`` / ``horovod`` — one
  process per GPU over torch.distributed, backend "nccl" (= RCCL over
  xGMI on ROCm, gloo on CPU): fused ``pushpull`` = bucketed async
  all-reduce on a dedicated comm stream so gradient communication
  overlaps backward (reference KVStoreNCCL kvstore_nccl.h:62; xGMI ring
  per-link bound ≈153 GB/s drives the ~50 MB bucket default).

Use ``create(name)`` exactly like ``mx.kv.create``.
"""
import os

import torch
import torch.distributed as dist

from ..ndarray.ndarray import NDArray

__all__ = ['KVStore', 'KVStoreBase', 'create']


class KVStoreBase:
    """Registry base (reference python/mxnet/kvstore/base.py:74)."""

    _registry = {}

    @classmethod
    def register(cls, klass):
        cls._registry[klass.__name__.lower()] = klass
        return klass

    OPTIMIZER = 'optimizer'

    def broadcast(self, key, value, out):
        raise NotImplementedError

    def pushpull(self, key, value, out=None, priority=0):
        raise NotImplementedError

    @property
    def type(self):
        return self._type

    @property
    def rank(self):
        return 0

    @property
    def num_workers(self):
        return 1


def create(name='local'):
    from ..base import native_mode
    if isinstance(name, str) and name == 'dist_sync_ps':
        if native_mode():
            raise ValueError(
                "kvstore 'dist_sync_ps' runs on the torch frontend "
                "(gloo send/recv internals); with MXNET_NATIVE_RUNTIME "
                "use 'dist_sync' (own RCCL / gloo-bridged all-reduce)")
        return SyncPSKVStore(name)
    if native_mode() and isinstance(name, str) and (
            name.startswith('dist') or name in ('nccl', 'device')):
        import os as _os
        if int(_os.environ.get('WORLD_SIZE', '1')) > 1:
            return NativeDistKVStore(name)
        return KVStore('local')  # single process: no collective needed
    """Factory (reference KVStore::Create kvstore.cc:42-80)."""
    name = name.lower()
    if name == 'dist_async':
        return AsyncPSKVStore(name)
    if name in ('dist_sync', 'dist_device_sync', 'nccl',
                'dist', 'horovod', 'byteps'):
        return DistKVStore(name)
    if name in ('local', 'device', 'local_allreduce_cpu',
                'local_allreduce_device'):
        return KVStore(name)
    raise ValueError(f'unknown kvstore type {name}')


class KVStore(KVStoreBase):
    """Single-process store: 'local' (CPU reduce) or 'device' (GPU reduce)."""

    def __init__(self, kind='local'):
        self._type = kind
        self._data = {}           # key -> merged NDArray (on merge ctx)
        self._updater = None
        self._optimizer = None
        self._compression = None

    def set_gradient_compression(self, compression_params):
        """2-bit/1-bit gradient compression (reference
        kvstore.cc SetGradientCompression)."""
        from .gradient_compression import GradientCompression
        self._compression = GradientCompression(**compression_params)

    # -- init / push / pull ---------------------------------------------
    def init(self, key, value):
        if isinstance(key, (list, tuple)):
            for k, v in zip(key, value):
                self.init(k, v)
            return
        v = value[0] if isinstance(value, (list, tuple)) else value
        self._data[key] = v.copy()

    def _reduce(self, values):
        """CommDevice::Reduce — copy to merge device, ElementwiseSum."""
        if len(values) == 1:
            return values[0].copy()
        merge = values[0]._t
        acc = merge.clone().float() if merge.dtype in (torch.float16, torch.bfloat16) \
            else merge.clone()
        for v in values[1:]:
            acc += v._t.to(acc.device, non_blocking=True).to(acc.dtype)
        return NDArray(acc.to(merge.dtype))

    def push(self, key, value, priority=0):
        if isinstance(key, (list, tuple)):
            for k, v in zip(key, value):
                self.push(k, v, priority)
            return
        values = value if isinstance(value, (list, tuple)) else [value]
        if self._compression is not None:
            for i, v in enumerate(values):
                self._compression.compress_decompress((key, i), v._t)
        merged = self._reduce(values)
        if self._updater is not None:
            self._updater(key, merged, self._data[key])
        else:
            self._data[key] = merged

    def pull(self, key, out=None, priority=0, ignore_sparse=True):
        if isinstance(key, (list, tuple)) and isinstance(out, (list, tuple)) \
                and len(key) > 1:
            for k, o in zip(key, out):
                self.pull(k, o, priority)
            return
        if isinstance(key, (list, tuple)):
            key = key[0]
        merged = self._data[key]
        outs = out if isinstance(out, (list, tuple)) else [out]
        for o in outs:
            with torch.no_grad():
                o._t.copy_(merged._t.to(o._t.device, non_blocking=True)
                           .to(o._t.dtype))

    def pushpull(self, key, value, out=None, priority=0):
        vals = value if isinstance(value, (list, tuple)) else [value]
        outs = out if isinstance(out, (list, tuple)) else [out]
        if getattr(vals[0], 'is_native', False) and len(vals) == 1 and \
                (out is None or (len(outs) == 1 and outs[0] is vals[0])):
            # single-process native runtime: one device per process, so
            # reducing one value into itself is the identity
            return
        self.push(key, value, priority)
        if out is not None:
            self.pull(key, out, priority)

    def broadcast(self, key, value, out, priority=0):
        self.init(key, value)
        self.pull(key, out, priority)

    # -- optimizer-on-kvstore (reference: set_updater / set_optimizer) ---
    def set_updater(self, updater):
        self._updater = updater

    def set_optimizer(self, optimizer):
        from .. import optimizer as opt
        self._optimizer = optimizer
        self.set_updater(opt.get_updater(optimizer))

    @property
    def rank(self):
        return 0

    @property
    def num_workers(self):
        return 1

    def save_optimizer_states(self, fname, dump_optimizer=False):
        import pickle
        with open(fname, 'wb') as f:
            pickle.dump({k: v for k, v in
                         (self._updater.get_states() if self._updater else {}).items()}, f)

    def load_optimizer_states(self, fname):
        import pickle
        with open(fname, 'rb') as f:
            states = pickle.load(f)
        if self._updater:
            self._updater.states.update(states)


class NativeDistKVStore(KVStoreBase):
    """Own RCCL binding, engine-sequenced (native runtime path).

    Reference counterpart: KVStoreNCCL (kvstore_nccl.h:62) — the grouped
    launch + stream-sync 3-op pattern (:267-443) collapses into ONE engine
    op per collective on the device's dedicated comm stream
    (FnProperty::kGPUPrioritized): the comm worker waits the producing
    compute kernels via hipStreamWaitEvent, launches rcclAllReduce over
    xGMI, and consumers wait its completion event — overlap with backward
    falls out of the engine's dependency tracking.
    """

    def __init__(self, kind='dist_sync'):
        import os as _os
        from .. import _core
        self._type = kind
        world = int(_os.environ.get('WORLD_SIZE', 1))
        rank = int(_os.environ.get('RANK', 0))
        dev = int(_os.environ.get('LOCAL_RANK', 0))
        _os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
        _os.environ.setdefault('MASTER_PORT', '29741')
        self._core = _core
        self._gloo = not torch.cuda.is_available()
        if self._gloo:
            # CPU hosts (CI, multi-process tests): RCCL needs a GPU, so
            # the native arrays bridge through numpy onto gloo
            import torch.distributed as dist
            if not dist.is_initialized():
                dist.init_process_group('gloo', rank=rank, world_size=world)
            self._world, self._rank = world, rank
            return
        _core.rccl_init(world, rank, dev)

    def set_gradient_compression(self, compression_params):
        raise NotImplementedError(
            'gradient compression on the native RCCL path: pending')

    @property
    def rank(self):
        return self._rank if self._gloo else self._core.rccl_rank()

    @property
    def num_workers(self):
        return self._world if self._gloo else self._core.rccl_world()

    def _gloo_allreduce(self, v, average):
        import torch.distributed as dist
        t = torch.from_numpy(v.asnumpy().copy())
        dist.all_reduce(t)
        if average:
            t /= self._world
        from ..ndarray.ndarray import array as _arr
        _arr(t.numpy(), ctx=v.context, dtype=str(v.dtype)).copyto(v)

    def _gloo_broadcast(self, v, root):
        import torch.distributed as dist
        t = torch.from_numpy(v.asnumpy().copy())
        dist.broadcast(t, src=root)
        from ..ndarray.ndarray import array as _arr
        _arr(t.numpy(), ctx=v.context, dtype=str(v.dtype)).copyto(v)

    def init(self, key, value):
        v = value[0] if isinstance(value, (list, tuple)) else value
        if self._gloo:
            self._gloo_broadcast(v, 0)
        else:
            self._core.rccl_broadcast(v._h, 0)

    def broadcast(self, key, value, out, priority=0):
        v = value[0] if isinstance(value, (list, tuple)) else value
        if self._gloo:
            self._gloo_broadcast(v, 0)
        else:
            self._core.rccl_broadcast(v._h, 0)
        outs = out if isinstance(out, (list, tuple)) else [out]
        for o in outs:
            if o is not v:
                v.copyto(o)

    def pushpull(self, key, value, out=None, priority=0, async_op=False):
        """All-reduce-average in place; async by construction (the engine
        orders it against producers/consumers via vars)."""
        v = value[0] if isinstance(value, (list, tuple)) else value
        if self._gloo:
            self._gloo_allreduce(v, True)
        else:
            self._core.rccl_allreduce(v._h, True)
        if out is not None:
            outs = out if isinstance(out, (list, tuple)) else [out]
            for o in outs:
                if o is not v:
                    v.copyto(o)
        return None


class DistKVStore(KVStoreBase):
    """Multi-process collective store over torch.distributed (RCCL/gloo).

    Reference counterpart: KVStoreNCCL (kvstore_nccl.h) / KVStoreDist.
    pushpull(key, grads, out) = all-reduce over ranks, launched async on a
    dedicated HIP comm stream; ``priority`` keeps the reference semantics
    (lower numbers = later layers; RCCL executes in issue order which the
    Trainer arranges back-to-front exactly like the reference's
    priority queue).
    """

    def __init__(self, kind='dist_sync'):
        self._type = kind
        self._handles = []
        self._pg = None  # non-default group when the default is gloo
        if not dist.is_initialized():
            # composite (cuda:nccl + cpu:gloo) so CPU tensors on a GPU
            # box still have a backend (RCCL handles the cuda ones)
            backend = 'cuda:nccl,cpu:gloo' if torch.cuda.is_available() \
                else 'gloo'
            if 'RANK' in os.environ:
                dist.init_process_group(backend=backend)
            else:
                # single-process fallback so dist code runs un-launched
                os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
                os.environ.setdefault('MASTER_PORT', '29741')
                dist.init_process_group(backend=backend, rank=0, world_size=1)
        elif torch.cuda.is_available() and \
                dist.get_backend() == 'gloo':
            # a host-side gloo group was created earlier (bench timing /
            # preflight consensus); GPU gradients need an RCCL group
            self._pg = dist.new_group(backend='nccl')
        if torch.cuda.is_available():
            # clamp for CPU-tensor tests running world>1 on a 1-GPU box
            torch.cuda.set_device(int(os.environ.get('LOCAL_RANK', 0)) %
                                  max(torch.cuda.device_count(), 1))
        self._comm_stream = (torch.cuda.Stream()
                             if torch.cuda.is_available() else None)
        self._compression = None

    def set_gradient_compression(self, compression_params):
        from .gradient_compression import GradientCompression
        self._compression = GradientCompression(**compression_params)

    @property
    def rank(self):
        return dist.get_rank()

    @property
    def num_workers(self):
        return dist.get_world_size()

    def _grp(self, t):
        # GPU tensors ride the RCCL subgroup when the default is gloo
        return self._pg if (self._pg is not None and t.is_cuda) else None

    def init(self, key, value):
        # rank-0 value wins (reference: init broadcasts from root)
        v = value[0] if isinstance(value, (list, tuple)) else value
        dist.broadcast(v._t.data, src=0, group=self._grp(v._t))

    def broadcast(self, key, value, out, priority=0):
        v = value[0] if isinstance(value, (list, tuple)) else value
        dist.broadcast(v._t.data, src=0, group=self._grp(v._t))
        outs = out if isinstance(out, (list, tuple)) else [out]
        for o in outs:
            if o is not v:
                with torch.no_grad():
                    o._t.copy_(v._t)

    def pushpull(self, key, value, out=None, priority=0, async_op=False):
        """Fused all-reduce (mean is NOT applied: caller rescales like the
        reference, Trainer divides by batch size)."""
        v = value[0] if isinstance(value, (list, tuple)) else value
        t = v._t.grad if isinstance(v._t, torch.Tensor) and v._t.grad is not None \
            and out is None else v._t
        if self._compression is not None:
            # lossy quantize-with-error-feedback before the collective
            self._compression.compress_decompress(key, t.data)
        work = dist.all_reduce(t.data, op=dist.ReduceOp.SUM,
                               group=self._grp(t), async_op=async_op)
        if async_op:
            self._handles.append(work)
        if out is not None:
            outs = out if isinstance(out, (list, tuple)) else [out]
            for o in outs:
                if o._t.data_ptr() != t.data_ptr():
                    with torch.no_grad():
                        o._t.copy_(t)
        return work if async_op else None

    def push(self, key, value, priority=0):
        self.pushpull(key, value, None, priority)

    def pull(self, key, out=None, priority=0, ignore_sparse=True):
        pass  # pushpull already materialized the reduced value in-place

    def wait_all(self):
        for h in self._handles:
            h.wait()
        self._handles.clear()

    def set_optimizer(self, optimizer):
        raise NotImplementedError(
            'server-side optimizer: use update_on_kvstore=False with the '
            'distributed Trainer')


# ---------------------------------------------------------------------------
# dist_async: parameter server with a dedicated server rank
# ---------------------------------------------------------------------------

class AsyncPSKVStore(KVStoreBase):
    """Asynchronous parameter server (reference KVStoreDist +
    KVStoreDistServer, kvstore_dist_server.h:155: workers push grads,
    the server applies the optimizer immediately — no global barrier —
    and serves pulls of the current weights).

    Roles: the LAST rank of the torch.distributed world is the server
    (reference: DMLC_ROLE=server process); ranks 0..world-2 are
    workers.  Transport is gloo point-to-point (send/recv) — weights
    live on the server in fp32; GPU workers stage through CPU.  The
    server process calls ``run_server()`` (blocks until every worker
    sent the stop command); workers use push/pull, and the Trainer
    drives update-on-kvstore semantics.
    """

    _OP_INIT, _OP_PUSH, _OP_PULL, _OP_STOP = 0, 1, 2, 3
    _sync = False

    def __init__(self, kind='dist_async'):
        import os as _os
        self._type = kind
        if not dist.is_initialized():
            dist.init_process_group(backend='gloo')
        self._world = dist.get_world_size()
        # key sharding across S servers (reference EncodeDefaultKey
        # distributes keys over ps-lite servers); the LAST S ranks serve
        self._nservers = int(_os.environ.get('MXNET_PS_NSERVERS', '1'))
        assert self._world >= 1 + self._nservers,             'dist PS needs >= 1 worker + MXNET_PS_NSERVERS servers'
        self._rank = dist.get_rank()
        self._optimizer = None
        self._store = {}        # server side: key -> fp32 tensor
        self._states = {}       # server side optimizer states
        self._pending = {}      # sync mode: key -> (accum, count)
        self._waiting = {}      # sync mode: key -> queued pull srcs
        self._workers_group = dist.new_group(
            ranks=list(range(self._world - self._nservers)))

    def _server_of(self, key):
        return self._world - 1 - (key % self._nservers)

    @property
    def rank(self):
        return self._rank

    @property
    def num_workers(self):
        return self._world - self._nservers

    @property
    def is_server(self):
        return self._rank >= self._world - self._nservers

    def set_optimizer(self, optimizer):
        """Server-side updater (reference sync/async server optimizer,
        kvstore_dist_server.h:346-365); call on the server process."""
        self._optimizer = optimizer

    # -- worker protocol -------------------------------------------------
    def _send_header(self, op, key, numel, dst=None):
        h = torch.tensor([op, key, numel, 0], dtype=torch.long)
        dist.send(h, dst=self._server_of(key) if dst is None else dst)

    def init(self, key, value):
        v = value[0] if isinstance(value, (list, tuple)) else value
        t = v._t.detach().float().cpu().contiguous()
        dist.broadcast(t, src=0, group=self._workers_group)
        with torch.no_grad():
            v._t.copy_(t.to(v._t.device, v._t.dtype))
        if self._rank == 0:
            self._send_header(self._OP_INIT, key, t.numel())
            dist.send(t, dst=self._server_of(key))
        dist.barrier(group=self._workers_group)

    def push(self, key, value, priority=0):
        v = value[0] if isinstance(value, (list, tuple)) else value
        g = v._t.detach().float().cpu().contiguous()
        self._send_header(self._OP_PUSH, key, g.numel())
        dist.send(g, dst=self._server_of(key))

    def pull(self, key, out=None, priority=0, ignore_sparse=True):
        outs = out if isinstance(out, (list, tuple)) else [out]
        n = outs[0]._t.numel()
        self._send_header(self._OP_PULL, key, n)
        buf = torch.empty(n, dtype=torch.float32)
        dist.recv(buf, src=self._server_of(key))
        with torch.no_grad():
            for o in outs:
                o._t.copy_(buf.view(o._t.shape).to(o._t.device, o._t.dtype))

    def pushpull(self, key, value, out=None, priority=0, async_op=False):
        self.push(key, value, priority)
        if out is not None:
            self.pull(key, out, priority)

    def stop(self):
        for s in range(self._nservers):
            self._send_header(self._OP_STOP, 0, 0,
                              dst=self._world - 1 - s)

    def barrier_workers(self):
        dist.barrier(group=self._workers_group)

    def _apply(self, key, g):
        w = self._store[key]
        if self._optimizer is not None:
            if key not in self._states:
                self._states[key] = \
                    self._optimizer.create_state_multi_precision(
                        key, NDArray(w))
            self._optimizer.update_multi_precision(
                key, NDArray(w), NDArray(g), self._states[key])
        else:
            w.sub_(g)  # plain accumulate (reference default)

    # -- server loop -----------------------------------------------------
    def run_server(self):
        """Service requests until every worker sent STOP (reference
        DataHandleEx dispatch, kvstore_dist_server.h:325)."""
        assert self.is_server
        live = self.num_workers
        hdr = torch.empty(4, dtype=torch.long)
        while live > 0:
            src = dist.recv(hdr, src=None)
            op, key, numel = int(hdr[0]), int(hdr[1]), int(hdr[2])
            if op == self._OP_STOP:
                live -= 1
                continue
            if op == self._OP_INIT:
                t = torch.empty(numel, dtype=torch.float32)
                dist.recv(t, src=src)
                self._store[key] = t
            elif op == self._OP_PUSH:
                g = torch.empty(numel, dtype=torch.float32)
                dist.recv(g, src=src)
                if self._sync:
                    # sync mode (reference kvstore_dist_server.h:346-365
                    # ApplyUpdates): aggregate NumWorkers pushes into
                    # `merged`, run the optimizer ONCE, then answer the
                    # pulls that queued while waiting
                    acc, cnt = self._pending.get(key, (None, 0))
                    acc = g if acc is None else acc.add_(g)
                    cnt += 1
                    if cnt == self.num_workers:
                        self._apply(key, acc)
                        self._pending[key] = (None, 0)
                        for wsrc in self._waiting.pop(key, []):
                            dist.send(self._store[key], dst=wsrc)
                    else:
                        self._pending[key] = (acc, cnt)
                else:
                    self._apply(key, g)
            elif op == self._OP_PULL:
                if self._sync and self._pending.get(key, (None, 0))[1] > 0:
                    # update still aggregating: defer the reply
                    self._waiting.setdefault(key, []).append(src)
                else:
                    dist.send(self._store[key], dst=src)


class SyncPSKVStore(AsyncPSKVStore):
    """Synchronous parameter server (reference dist_sync over ps-lite):
    the server aggregates one push per worker per key, applies the
    (server-side) optimizer once, and releases the queued pulls — a
    per-key global barrier.  Key sharding across MXNET_PS_NSERVERS
    server ranks as in the async store."""
    _sync = True

    def __init__(self, kind='dist_sync_ps'):
        super().__init__(kind)
