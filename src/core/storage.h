// Storage — pooled HIP / CPU / pinned allocators behind one interface.
//
// Reference parity: include/mxnet/storage.h (Alloc/Free/DirectFree/
// ReleaseAll), src/storage/storage.cc:38 (per-(dev_type,dev_id) lazily
// created managers), src/storage/pooled_storage_manager.h:78-183 (pow2
// rounding pool, OOM -> ReleaseAll -> retry).
//
// MI355X sizing: 288 GB HBM3E per GPU — the pool keeps everything it ever
// allocated by default (reserve watermark MXNET_GPU_MEM_POOL_RESERVE % of
// free memory preserved for others, default 5).  Small allocations round
// to powers of two; beyond the linear cutoff (2^24 = 16 MiB) they round to
// 2 MiB multiples so 100+ MiB activations don't waste half their bucket.
#pragma once

#include <mutex>
#include <unordered_map>
#include <vector>

#include "base.h"

namespace mxcore {

class Storage {
 public:
  struct Handle {
    void* dptr = nullptr;
    size_t size = 0;  // requested bytes
    Context ctx;
  };

  static Storage* Get();

  Handle Alloc(size_t size, Context ctx);
  void Free(const Handle& h);        // returns to pool
  void DirectFree(const Handle& h);  // bypasses pool
  void ReleaseAll(Context ctx);      // drop all pooled blocks on ctx

  // hipGraph-capture keepalive (the role torch's graph-private memory
  // pools play): between BeginCapture/EndCapture the engine arms this;
  // frees of GPU blocks are PARKED instead of pooled, because the
  // captured graph references their addresses on every replay.  The
  // parked set is returned at capture end and pooled only when the
  // graph is released.
  void BeginCaptureKeepalive(int dev_id);
  std::vector<Handle> EndCaptureKeepalive(int dev_id);
  void ReleaseHandles(const std::vector<Handle>& hs);  // pool them now
  // bytes currently cached in the pool for ctx (testing/telemetry)
  size_t PoolSize(Context ctx);
  // bytes handed out and not yet freed (testing/telemetry)
  size_t UsedSize(Context ctx);

 private:
  class Manager;
  Manager* GetManager(const Context& ctx);

  std::mutex mu_;
  std::unordered_map<int64_t, Manager*> managers_;

  // capture keepalive state (see BeginCaptureKeepalive)
  std::mutex cap_mu_;
  std::unordered_map<int, std::vector<Handle>> capture_parked_;
  std::unordered_map<int, bool> capturing_;
};

}  // namespace mxcore
