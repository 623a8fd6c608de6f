#include "storage.h"

#include <cstdlib>
#include <map>

namespace mxcore {

namespace {

// round sizes <= 16 MiB to the next power of two, larger ones to 2 MiB
// multiples (reference RoundPower2 + the linear-cutoff behaviour of
// MXNET_GPU_MEM_POOL_ROUND_LINEAR_CUTOFF=24)
inline size_t RoundSize(size_t s) {
  if (s == 0) s = 1;
  const size_t kCutoff = (size_t)1 << 24;  // 16 MiB
  if (s <= kCutoff) {
    size_t r = 1;
    while (r < s) r <<= 1;
    return r < 4096 ? 4096 : r;  // page floor: tiny allocs share buckets
  }
  const size_t kChunk = (size_t)2 << 20;  // 2 MiB
  return (s + kChunk - 1) / kChunk * kChunk;
}

}  // namespace

class Storage::Manager {
 public:
  explicit Manager(Context ctx) : ctx_(ctx) {}

  Handle Alloc(size_t size) {
    size_t rounded = RoundSize(size);
    {
      std::lock_guard<std::mutex> g(mu_);
      auto it = pool_.find(rounded);
      if (it != pool_.end() && !it->second.empty()) {
        void* p = it->second.back();
        it->second.pop_back();
        pooled_bytes_ -= rounded;
        used_bytes_ += rounded;
        return Handle{p, size, ctx_};
      }
    }
    void* p = RawAlloc(rounded);
    if (p == nullptr && ctx_.is_gpu()) {
      // OOM: release every cached block, then retry once
      // (reference pooled_storage_manager.h:139-183)
      ReleaseAll();
      p = RawAlloc(rounded);
    }
    MX_CHECK(p != nullptr, "out of memory allocating " << rounded
                           << " bytes on " << ctx_.str());
    std::lock_guard<std::mutex> g(mu_);
    used_bytes_ += rounded;
    return Handle{p, size, ctx_};
  }

  void Free(const Handle& h) {
    size_t rounded = RoundSize(h.size);
    std::lock_guard<std::mutex> g(mu_);
    pool_[rounded].push_back(h.dptr);
    pooled_bytes_ += rounded;
    used_bytes_ -= rounded;
  }

  void DirectFree(const Handle& h) {
    RawFree(h.dptr);
    std::lock_guard<std::mutex> g(mu_);
    used_bytes_ -= RoundSize(h.size);
  }

  void ReleaseAll() {
    std::lock_guard<std::mutex> g(mu_);
    for (auto& kv : pool_)
      for (void* p : kv.second) RawFree(p);
    pool_.clear();
    pooled_bytes_ = 0;
  }

  size_t pooled_bytes() {
    std::lock_guard<std::mutex> g(mu_);
    return pooled_bytes_;
  }
  size_t used_bytes() {
    std::lock_guard<std::mutex> g(mu_);
    return used_bytes_;
  }

 private:
  void* RawAlloc(size_t n) {
    if (ctx_.dev_type == Context::kGPU) {
      hipError_t e = hipSetDevice(ctx_.dev_id);
      if (e != hipSuccess) return nullptr;
      void* p = nullptr;
      e = hipMalloc(&p, n);
      if (e != hipSuccess) {
        (void)hipGetLastError();  // clear sticky error
        return nullptr;
      }
      return p;
    }
    if (ctx_.dev_type == Context::kCPUPinned) {
      void* p = nullptr;
      if (hipHostMalloc(&p, n, hipHostMallocDefault) != hipSuccess) {
        (void)hipGetLastError();
        return nullptr;
      }
      return p;
    }
    void* p = nullptr;
    if (posix_memalign(&p, 64, n) != 0) return nullptr;
    return p;
  }

  void RawFree(void* p) {
    if (ctx_.dev_type == Context::kGPU) {
      (void)hipFree(p);
    } else if (ctx_.dev_type == Context::kCPUPinned) {
      (void)hipHostFree(p);
    } else {
      free(p);
    }
  }

  Context ctx_;
  std::mutex mu_;
  std::unordered_map<size_t, std::vector<void*>> pool_;
  size_t pooled_bytes_ = 0;
  size_t used_bytes_ = 0;
};

Storage* Storage::Get() {
  static Storage inst;
  return &inst;
}

Storage::Manager* Storage::GetManager(const Context& ctx) {
  int64_t key = (int64_t)ctx.dev_type << 32 | (uint32_t)ctx.dev_id;
  std::lock_guard<std::mutex> g(mu_);
  auto it = managers_.find(key);
  if (it == managers_.end())
    it = managers_.emplace(key, new Manager(ctx)).first;
  return it->second;
}

Storage::Handle Storage::Alloc(size_t size, Context ctx) {
  return GetManager(ctx)->Alloc(size);
}

void Storage::Free(const Handle& h) {
  // during hipGraph capture on this device, park GPU frees: the graph
  // bakes these addresses into its nodes, so recycling them under a
  // later replay would alias live graph buffers
  if (h.ctx.dev_type == Context::kGPU) {
    std::lock_guard<std::mutex> g(cap_mu_);
    auto it = capturing_.find(h.ctx.dev_id);
    if (it != capturing_.end() && it->second) {
      capture_parked_[h.ctx.dev_id].push_back(h);
      return;
    }
  }
  GetManager(h.ctx)->Free(h);
}

void Storage::BeginCaptureKeepalive(int dev_id) {
  std::lock_guard<std::mutex> g(cap_mu_);
  capturing_[dev_id] = true;
}

std::vector<Storage::Handle> Storage::EndCaptureKeepalive(int dev_id) {
  std::lock_guard<std::mutex> g(cap_mu_);
  capturing_[dev_id] = false;
  std::vector<Handle> out;
  out.swap(capture_parked_[dev_id]);
  return out;
}

void Storage::ReleaseHandles(const std::vector<Handle>& hs) {
  for (auto& h : hs) GetManager(h.ctx)->Free(h);
}

void Storage::DirectFree(const Handle& h) { GetManager(h.ctx)->DirectFree(h); }

void Storage::ReleaseAll(Context ctx) { GetManager(ctx)->ReleaseAll(); }

size_t Storage::PoolSize(Context ctx) { return GetManager(ctx)->pooled_bytes(); }

size_t Storage::UsedSize(Context ctx) { return GetManager(ctx)->used_bytes(); }

}  // namespace mxcore
