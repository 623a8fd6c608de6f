#include "engine.h"

#include <algorithm>
#include <atomic>
#include <chrono>
#include <unordered_map>
#include <cassert>
#include <future>
#include <new>

#include "storage.h"

namespace mxcore {

namespace {

// pooled hipEvents (one pool per device)
struct EventPool {
  std::mutex mu;
  std::vector<hipEvent_t> free_list;
  hipEvent_t Take(int dev) {
    {
      std::lock_guard<std::mutex> g(mu);
      if (!free_list.empty()) {
        hipEvent_t e = free_list.back();
        free_list.pop_back();
        return e;
      }
    }
    hipEvent_t e;
    MX_HIP_CALL(hipSetDevice(dev));
    MX_HIP_CALL(hipEventCreateWithFlags(&e, hipEventDisableTiming));
    return e;
  }
  void Put(hipEvent_t e) {
    std::lock_guard<std::mutex> g(mu);
    free_list.push_back(e);
  }
};

EventPool& GetEventPool(int dev) {
  static EventPool pools[64];
  return pools[dev & 63];
}

// refcounted event: recycled to the pool when the last holder drops it
struct EventRef {
  hipEvent_t ev = nullptr;
  int dev = 0;
  hipStream_t stream = nullptr;  // stream it was recorded on
  ~EventRef() {
    if (ev) GetEventPool(dev).Put(ev);
  }
};
using EventPtr = std::shared_ptr<EventRef>;

}  // namespace

struct Opr;

struct Block {
  Opr* opr;
  bool write;
  bool granted = false;
};

struct Var {
  uint64_t version = 0;
  std::deque<Block> queue;
  int running_reads = 0;
  std::exception_ptr exc;
  EventPtr last_event;  // execution-completion of the last write
  // execution-completion of reads since that write: a later writer (or the
  // storage-freeing delete op) must order behind these, not just behind
  // the producing write (write-after-read across streams)
  std::vector<EventPtr> read_events;
  bool to_delete = false;
};

struct Opr {
  OpFn fn;
  std::vector<Var*> const_vars;
  std::vector<Var*> mutable_vars;
  int wait = 0;
  FnProperty prop = FnProperty::kNormal;
  Context ctx;
  const char* name = "";
  VarId delete_var = 0;           // kDeleteVar: var to erase afterwards
  std::vector<EventPtr> waits;    // foreign events to wait before fn
};

// one FIFO queue + worker thread(s)
struct WorkQueue {
  std::mutex mu;
  std::condition_variable cv;
  std::queue<Opr*> q;
  bool shutdown = false;
  void Push(Opr* o) {
    {
      std::lock_guard<std::mutex> g(mu);
      q.push(o);
    }
    cv.notify_one();
  }
  Opr* Pop() {
    std::unique_lock<std::mutex> lk(mu);
    cv.wait(lk, [&] { return shutdown || !q.empty(); });
    if (q.empty()) return nullptr;
    Opr* o = q.front();
    q.pop();
    return o;
  }
  void Shutdown() {
    {
      std::lock_guard<std::mutex> g(mu);
      shutdown = true;
    }
    cv.notify_all();
  }
  // post-fork: worker threads are gone in the child; drop queued work
  // and reopen
  void Reset() {
    std::lock_guard<std::mutex> g(mu);
    while (!q.empty()) q.pop();
    shutdown = false;
  }
  // child-side rebuild: the glibc condvar keeps internal waiter counts
  // for the PARENT's blocked workers — a notify in the child can be
  // consumed by those ghost slots and the new worker never wakes
  // (observed as a flaky post-fork hang).  No thread of ours exists in
  // the child yet, so placement-reinit is safe.
  void ResetAfterFork() {
    new (&mu) std::mutex();
    new (&cv) std::condition_variable();
    q = {};
    shutdown = false;
  }
};

// profiler record: timed event pair (GPU) or wall ns (CPU)
struct ProfRec {
  std::string name;
  hipEvent_t ev0 = nullptr, ev1 = nullptr;
  int dev = 0;
  double cpu_ms = 0;
};

struct DeviceWorkers {
  hipStream_t compute = nullptr, copy = nullptr, comm = nullptr;
  WorkQueue compute_q, copy_q, comm_q;
  std::thread compute_t, copy_t, comm_t;
  std::atomic<bool> capturing{false};
  hipGraph_t captured = nullptr;
};

struct Engine::Impl {
  std::mutex mu_;                       // protects vars/inflight
  std::condition_variable done_cv_;
  std::unordered_map<VarId, std::unique_ptr<Var>> vars_;
  VarId next_var_ = 1;
  std::atomic<long> inflight_{0};
  std::exception_ptr global_exc_;

  WorkQueue cpu_q_, cpu_prio_q_;
  std::vector<std::thread> cpu_workers_;

  std::atomic<bool> profiling_{false};
  std::mutex prof_mu_;
  std::vector<ProfRec> prof_;

  std::mutex dev_mu_;
  std::unordered_map<int, std::unique_ptr<DeviceWorkers>> devices_;

  std::atomic<bool> forked_{false};
  void ReinitAfterFork();

  // blocks freed during a capture, alive for the captured graph's life
  std::unordered_map<uintptr_t, std::vector<Storage::Handle>>
      graph_keepalive_;

  Engine* owner_ = nullptr;

  Var* GetVar(VarId id) {
    auto it = vars_.find(id);
    MX_CHECK(it != vars_.end(), "unknown engine var " << id);
    return it->second.get();
  }

  DeviceWorkers* GetDevice(int dev) {
    std::lock_guard<std::mutex> g(dev_mu_);
    auto it = devices_.find(dev);
    if (it != devices_.end()) return it->second.get();
    auto dw = std::make_unique<DeviceWorkers>();
    MX_HIP_CALL(hipSetDevice(dev));
    MX_HIP_CALL(hipStreamCreateWithFlags(&dw->compute, hipStreamNonBlocking));
    MX_HIP_CALL(hipStreamCreateWithFlags(&dw->copy, hipStreamNonBlocking));
    MX_HIP_CALL(hipStreamCreateWithFlags(&dw->comm, hipStreamNonBlocking));
    DeviceWorkers* p = dw.get();
    p->compute_t = std::thread([this, p, dev] {
      WorkerLoop(&p->compute_q, RunContext{Context::GPU(dev), p->compute}, p);
    });
    p->copy_t = std::thread([this, p, dev] {
      WorkerLoop(&p->copy_q, RunContext{Context::GPU(dev), p->copy}, p);
    });
    p->comm_t = std::thread([this, p, dev] {
      WorkerLoop(&p->comm_q, RunContext{Context::GPU(dev), p->comm}, p);
    });
    devices_.emplace(dev, std::move(dw));
    return p;
  }

  void StartCPUWorkers() {
    int n = env_int("MXNET_CPU_WORKER_NTHREADS", 4);
    RunContext cpu_rc{Context::CPU(), nullptr};
    for (int i = 0; i < n; ++i)
      cpu_workers_.emplace_back(
          [this, cpu_rc] { WorkerLoop(&cpu_q_, cpu_rc, nullptr); });
    for (int i = 0; i < 2; ++i)
      cpu_workers_.emplace_back(
          [this, cpu_rc] { WorkerLoop(&cpu_prio_q_, cpu_rc, nullptr); });
  }

  // ---- dependency tracking (reference threaded_engine.h:120-229) ------
  void Push(Opr* opr) {
    if (forked_.load(std::memory_order_acquire) &&
        forked_.exchange(false))
      ReinitAfterFork();
    bool ready;
    {
      std::lock_guard<std::mutex> g(mu_);
      ++inflight_;
      opr->wait = (int)(opr->const_vars.size() + opr->mutable_vars.size());
      for (Var* v : opr->const_vars) {
        bool write_pending = false;
        for (auto& b : v->queue)
          if (b.write) {
            write_pending = true;
            break;
          }
        v->queue.push_back({opr, false, !write_pending});
        if (!write_pending) {
          ++v->running_reads;
          --opr->wait;
        }
      }
      for (Var* v : opr->mutable_vars) {
        bool idle = v->queue.empty() && v->running_reads == 0;
        v->queue.push_back({opr, true, idle});
        if (idle) --opr->wait;
      }
      ready = opr->wait == 0;
    }
    if (ready) Dispatch(opr);
  }

  // snapshot foreign completion events, then hand to the right queue
  void Dispatch(Opr* opr) {
    {
      std::lock_guard<std::mutex> g(mu_);
      for (Var* v : opr->const_vars) {
        if (v->last_event) opr->waits.push_back(v->last_event);
      }
      for (Var* v : opr->mutable_vars) {
        if (v->last_event) opr->waits.push_back(v->last_event);
        for (auto& e : v->read_events) opr->waits.push_back(e);
      }
    }
    switch (opr->prop) {
      case FnProperty::kDeleteVar:
      case FnProperty::kAsync:
      case FnProperty::kCPUPrioritized:
        cpu_prio_q_.Push(opr);
        return;
      default:
        break;
    }
    if (!opr->ctx.is_gpu()) {
      cpu_q_.Push(opr);
      return;
    }
    DeviceWorkers* dw = GetDevice(opr->ctx.dev_id);
    switch (opr->prop) {
      case FnProperty::kCopyFromGPU:
      case FnProperty::kCopyToGPU:
        dw->copy_q.Push(opr);
        break;
      case FnProperty::kGPUPrioritized:
        dw->comm_q.Push(opr);
        break;
      default:
        dw->compute_q.Push(opr);
    }
  }

  void WorkerLoop(WorkQueue* q, RunContext rc, DeviceWorkers* dw) {
    if (rc.ctx.is_gpu()) (void)hipSetDevice(rc.ctx.dev_id);
    for (;;) {
      Opr* opr = q->Pop();
      if (opr == nullptr) return;
      std::exception_ptr exc;
      bool capturing = dw && dw->capturing.load(std::memory_order_relaxed);
      bool prof = profiling_.load(std::memory_order_relaxed) && !capturing;
      ProfRec rec;
      auto prof_t0 = std::chrono::steady_clock::now();
      if (prof && rc.ctx.is_gpu()) {
        rec.dev = rc.ctx.dev_id;
        MX_HIP_CALL(hipEventCreate(&rec.ev0));
        MX_HIP_CALL(hipEventCreate(&rec.ev1));
        (void)hipEventRecord(rec.ev0, rc.stream);
      }
      // dependency-failure propagation (reference threaded_engine.cc
      // OnCompleteStatic): if any input/output var carries a pending
      // exception, skip fn and forward it — downstream consumers see the
      // ORIGINAL error at their sync point instead of computing on
      // uninitialized buffers.  Deleters still run (storage must free).
      std::exception_ptr in_exc;
      if (!opr->delete_var) {
        std::lock_guard<std::mutex> g(mu_);
        for (Var* v : opr->const_vars)
          if (v->exc) { in_exc = v->exc; break; }
        if (!in_exc)
          for (Var* v : opr->mutable_vars)
            if (v->exc) { in_exc = v->exc; break; }
      }
      try {
        // order execution behind producers on other streams
        for (auto& w : opr->waits) {
          if (rc.ctx.is_gpu() && !capturing) {
            if (w->stream != rc.stream || w->dev != rc.ctx.dev_id)
              MX_HIP_CALL(hipStreamWaitEvent(rc.stream, w->ev, 0));
          } else if (!rc.ctx.is_gpu()) {
            MX_HIP_CALL(hipEventSynchronize(w->ev));
          }
        }
        if (in_exc)
          exc = in_exc;
        else if (opr->fn)
          opr->fn(rc);
      } catch (...) {
        exc = std::current_exception();
      }
      if (prof) {
        rec.name = opr->name;
        if (rc.ctx.is_gpu()) {
          (void)hipEventRecord(rec.ev1, rc.stream);
        } else {
          rec.cpu_ms = std::chrono::duration<double, std::milli>(
                           std::chrono::steady_clock::now() - prof_t0)
                           .count();
        }
        std::lock_guard<std::mutex> g(prof_mu_);
        prof_.push_back(std::move(rec));
      }
      EventPtr done;
      // re-read AFTER fn: the BeginCapture op flips capturing on inside
      // its fn — recording an event then would inject a graph node whose
      // pooled hipEvent outlives the exec (heap corruption on destroy)
      bool capturing_now = dw && dw->capturing.load(std::memory_order_relaxed);
      if (rc.ctx.is_gpu() && !exc && !capturing_now) {
        // execution-completion marker for cross-stream/CPU consumers
        done = std::make_shared<EventRef>();
        done->dev = rc.ctx.dev_id;
        done->stream = rc.stream;
        done->ev = GetEventPool(done->dev).Take(done->dev);
        hipError_t e = hipEventRecord(done->ev, rc.stream);
        if (e != hipSuccess) {
          (void)hipGetLastError();
          done.reset();
        }
      }
      OnComplete(opr, exc, done);
    }
  }

  void GrantHead(Var* v, std::vector<Opr*>* now_ready) {
    for (auto it = v->queue.begin(); it != v->queue.end() && !it->write; ++it) {
      if (!it->granted) {
        it->granted = true;
        ++v->running_reads;
        if (--it->opr->wait == 0) now_ready->push_back(it->opr);
      }
    }
    if (!v->queue.empty() && v->queue.front().write && v->running_reads == 0 &&
        !v->queue.front().granted) {
      v->queue.front().granted = true;
      if (--v->queue.front().opr->wait == 0)
        now_ready->push_back(v->queue.front().opr);
    }
  }

  void OnComplete(Opr* opr, std::exception_ptr exc, EventPtr done) {
    std::vector<Opr*> now_ready;
    VarId erase_var = 0;
    {
      std::lock_guard<std::mutex> g(mu_);
      if (exc) {
        global_exc_ = exc;
        for (Var* v : opr->mutable_vars) v->exc = exc;
      }
      for (Var* v : opr->const_vars) {
        for (auto it = v->queue.begin(); it != v->queue.end(); ++it)
          if (it->opr == opr && !it->write) {
            v->queue.erase(it);
            break;
          }
        --v->running_reads;
        if (done) v->read_events.push_back(done);
        GrantHead(v, &now_ready);
      }
      for (Var* v : opr->mutable_vars) {
        ++v->version;
        // this write's execution is ordered behind the reads it waited on,
        // so its event supersedes them
        v->read_events.clear();
        if (done) v->last_event = done;
        for (auto it = v->queue.begin(); it != v->queue.end(); ++it)
          if (it->opr == opr && it->write) {
            v->queue.erase(it);
            break;
          }
        GrantHead(v, &now_ready);
      }
      if (opr->delete_var) erase_var = opr->delete_var;
      --inflight_;
    }
    done_cv_.notify_all();
    for (Opr* o : now_ready) Dispatch(o);
    if (erase_var) {
      std::lock_guard<std::mutex> g(mu_);
      auto it = vars_.find(erase_var);
      if (it != vars_.end() && it->second->queue.empty() &&
          it->second->running_reads == 0)
        vars_.erase(it);
    }
    delete opr;
  }
};

Engine::Engine() : impl_(new Impl()) {
  impl_->owner_ = this;
  impl_->StartCPUWorkers();
}

Engine* Engine::Get() {
  static Engine* e = new Engine();
  return e;
}

VarId Engine::NewVariable() {
  std::lock_guard<std::mutex> g(impl_->mu_);
  VarId id = impl_->next_var_++;
  impl_->vars_.emplace(id, std::make_unique<Var>());
  return id;
}

void Engine::PushDeleteVariable(VarId v, OpFn on_delete) {
  Var* var;
  {
    std::lock_guard<std::mutex> g(impl_->mu_);
    auto it = impl_->vars_.find(v);
    if (it == impl_->vars_.end()) return;
    var = it->second.get();
    var->to_delete = true;
  }
  auto* opr = new Opr();
  // the generic CPU-op path synchronizes every pending execution event of
  // a mutable var before fn runs, so the storage is free to release here
  opr->fn = std::move(on_delete);
  opr->mutable_vars = {var};
  opr->prop = FnProperty::kDeleteVar;
  opr->ctx = Context::CPU();
  opr->delete_var = v;
  impl_->Push(opr);
}

void Engine::PushAsync(OpFn fn, Context ctx, const std::vector<VarId>& cv,
                       const std::vector<VarId>& mv, FnProperty prop,
                       const char* name) {
  auto* opr = new Opr();
  opr->fn = std::move(fn);
  opr->prop = prop;
  opr->ctx = ctx;
  if (name) opr->name = name;
  {
    std::lock_guard<std::mutex> g(impl_->mu_);
    for (VarId id : cv) opr->const_vars.push_back(impl_->GetVar(id));
    for (VarId id : mv) opr->mutable_vars.push_back(impl_->GetVar(id));
  }
  impl_->Push(opr);
}

void Engine::WaitForVar(VarId v) {
  std::exception_ptr exc;
  EventPtr ev;
  {
    std::unique_lock<std::mutex> lk(impl_->mu_);
    auto it = impl_->vars_.find(v);
    if (it == impl_->vars_.end()) return;
    Var* var = it->second.get();
    impl_->done_cv_.wait(
        lk, [&] { return var->queue.empty() && var->running_reads == 0; });
    exc = var->exc;
    var->exc = nullptr;
    // an exception delivered here is CONSUMED: WaitForAll must not
    // re-throw the same failure into unrelated later sync points
    if (exc && impl_->global_exc_ == exc) impl_->global_exc_ = nullptr;
    ev = var->last_event;
  }
  if (ev && ev->ev) MX_HIP_CALL(hipEventSynchronize(ev->ev));
  if (exc) std::rethrow_exception(exc);
}

void Engine::WaitForAll() {
  std::exception_ptr exc;
  {
    std::unique_lock<std::mutex> lk(impl_->mu_);
    impl_->done_cv_.wait(lk, [&] { return impl_->inflight_.load() == 0; });
    exc = impl_->global_exc_;
    impl_->global_exc_ = nullptr;
  }
  // drain execution on every known device stream
  {
    std::lock_guard<std::mutex> g(impl_->dev_mu_);
    for (auto& kv : impl_->devices_) {
      (void)hipStreamSynchronize(kv.second->compute);
      (void)hipStreamSynchronize(kv.second->copy);
      (void)hipStreamSynchronize(kv.second->comm);
    }
  }
  if (exc) std::rethrow_exception(exc);
}

uint64_t Engine::Version(VarId v) {
  std::lock_guard<std::mutex> g(impl_->mu_);
  auto it = impl_->vars_.find(v);
  return it == impl_->vars_.end() ? 0 : it->second->version;
}

void Engine::Throw(VarId v) {
  std::exception_ptr exc;
  {
    std::lock_guard<std::mutex> g(impl_->mu_);
    auto it = impl_->vars_.find(v);
    if (it == impl_->vars_.end()) return;
    exc = it->second->exc;
    it->second->exc = nullptr;
    if (exc && impl_->global_exc_ == exc) impl_->global_exc_ = nullptr;
  }
  if (exc) std::rethrow_exception(exc);
}

hipEvent_t Engine::LastEvent(VarId v) {
  std::lock_guard<std::mutex> g(impl_->mu_);
  auto it = impl_->vars_.find(v);
  if (it == impl_->vars_.end() || !it->second->last_event) return nullptr;
  return it->second->last_event->ev;
}

hipStream_t Engine::ComputeStream(int dev) {
  return impl_->GetDevice(dev)->compute;
}
hipStream_t Engine::CopyStream(int dev) { return impl_->GetDevice(dev)->copy; }
hipStream_t Engine::CommStream(int dev) { return impl_->GetDevice(dev)->comm; }

// ---- hipGraph capture -----------------------------------------------------
void Engine::BeginCapture(int dev) {
  // park pool frees on this device for the capture's duration — the
  // graph bakes buffer addresses into its nodes (see Storage keepalive)
  Storage::Get()->BeginCaptureKeepalive(dev);
  DeviceWorkers* dw = impl_->GetDevice(dev);
  std::promise<void> p;
  auto fut = p.get_future();
  PushAsync(
      [dw, &p](const RunContext& rc) {
        MX_HIP_CALL(
            hipStreamBeginCapture(rc.stream, hipStreamCaptureModeRelaxed));
        dw->capturing.store(true);
        p.set_value();
      },
      Context::GPU(dev), {}, {}, FnProperty::kNormal, "BeginCapture");
  fut.wait();
}

uintptr_t Engine::EndCapture(int dev) {
  DeviceWorkers* dw = impl_->GetDevice(dev);
  std::promise<uintptr_t> p;
  auto fut = p.get_future();
  PushAsync(
      [dw, &p](const RunContext& rc) {
        dw->capturing.store(false);
        hipGraph_t g = nullptr;
        MX_HIP_CALL(hipStreamEndCapture(rc.stream, &g));
        size_t nnodes = 0;
        (void)hipGraphGetNodes(g, nullptr, &nnodes);
        if (env_int("MXNET_ENGINE_DEBUG", 0))
          fprintf(stderr, "[engine] captured graph: %zu nodes\n", nnodes);
        hipGraphExec_t exec = nullptr;
        MX_HIP_CALL(hipGraphInstantiate(&exec, g, nullptr, nullptr, 0));
        (void)hipGraphDestroy(g);
        p.set_value((uintptr_t)exec);
      },
      Context::GPU(dev), {}, {}, FnProperty::kNormal, "EndCapture");
  uintptr_t exec = fut.get();
  // drain async deleters pushed during the capture window, then move
  // every parked block into the graph's keepalive set — they are
  // pooled again only when the graph is released
  WaitForAll();
  auto parked = Storage::Get()->EndCaptureKeepalive(dev);
  {
    std::lock_guard<std::mutex> g(impl_->dev_mu_);
    impl_->graph_keepalive_[exec] = std::move(parked);
  }
  return exec;
}

void Engine::ReleaseGraph(uintptr_t exec) {
  std::vector<Storage::Handle> parked;
  {
    std::lock_guard<std::mutex> g(impl_->dev_mu_);
    auto it = impl_->graph_keepalive_.find(exec);
    if (it != impl_->graph_keepalive_.end()) {
      parked = std::move(it->second);
      impl_->graph_keepalive_.erase(it);
    }
  }
  WaitForAll();
  (void)hipGraphExecDestroy((hipGraphExec_t)exec);
  Storage::Get()->ReleaseHandles(parked);
}

void Engine::LaunchGraph(int dev, uintptr_t exec,
                         const std::vector<VarId>& after,
                         const std::vector<VarId>& mutate) {
  // `mutate` declares the buffers the captured graph WRITES (e.g. the
  // gradient set of a fwd+bwd graph): consumers on other streams (the
  // RCCL all-reduces on the comm stream) then order behind the graph's
  // completion event instead of a stale pre-capture event.
  PushAsync(
      [exec](const RunContext& rc) {
        MX_HIP_CALL(hipGraphLaunch((hipGraphExec_t)exec, rc.stream));
      },
      Context::GPU(dev), after, mutate, FnProperty::kNormal, "GraphLaunch");
}

void Engine::SetProfiling(bool on) {
  impl_->profiling_.store(on);
}

std::vector<std::tuple<std::string, long, double>> Engine::ProfilerSummary() {
  WaitForAll();
  std::vector<ProfRec> recs;
  {
    std::lock_guard<std::mutex> g(impl_->prof_mu_);
    recs.swap(impl_->prof_);
  }
  std::unordered_map<std::string, std::pair<long, double>> agg;
  for (auto& r : recs) {
    double ms = r.cpu_ms;
    if (r.ev0) {
      float f = 0;
      (void)hipEventSynchronize(r.ev1);
      (void)hipEventElapsedTime(&f, r.ev0, r.ev1);
      ms = f;
      (void)hipEventDestroy(r.ev0);
      (void)hipEventDestroy(r.ev1);
    }
    auto& a = agg[r.name];
    a.first += 1;
    a.second += ms;
  }
  std::vector<std::tuple<std::string, long, double>> out;
  for (auto& kv : agg)
    out.emplace_back(kv.first, kv.second.first, kv.second.second);
  std::sort(out.begin(), out.end(), [](auto& a, auto& b) {
    return std::get<2>(a) > std::get<2>(b);
  });
  return out;
}

// fork support (reference LibraryInitializer::install_pthread_atfork
// _handlers): prepare drains every queue so no op is mid-flight and no
// engine lock is held across fork; the child inherits no threads, so it
// detaches the dead std::thread handles, drops GPU state (HIP contexts
// do not survive fork) and restarts the CPU workers lazily.
// prepare: drain, then take EVERY engine mutex so none is mid-acquire
// in a worker when fork() snapshots the address space (a worker grabs
// the queue mutex briefly between Pops — fork landing in that window
// would leave the child's mutex locked forever).  parent/child
// handlers release them again; the forking thread is the same thread
// in the child, so the unlocks are well-defined.
void Engine::AtForkPrepare() {
  WaitForAll();
  impl_->mu_.lock();
  impl_->prof_mu_.lock();
  impl_->dev_mu_.lock();
  impl_->cpu_q_.mu.lock();
  impl_->cpu_prio_q_.mu.lock();
  for (auto& kv : impl_->devices_) {
    kv.second->compute_q.mu.lock();
    kv.second->copy_q.mu.lock();
    kv.second->comm_q.mu.lock();
  }
}

void Engine::AtForkParent() {
  for (auto& kv : impl_->devices_) {
    kv.second->comm_q.mu.unlock();
    kv.second->copy_q.mu.unlock();
    kv.second->compute_q.mu.unlock();
  }
  impl_->cpu_prio_q_.mu.unlock();
  impl_->cpu_q_.mu.unlock();
  impl_->dev_mu_.unlock();
  impl_->prof_mu_.unlock();
  impl_->mu_.unlock();
}

// ONLY sets a flag: the handler also fires on every fork()+exec (python
// subprocess), and allocating/freeing or starting threads in a freshly
// forked child can deadlock on malloc locks held by OTHER libraries'
// threads at fork time.  The real rebuild happens lazily on the child's
// first engine use.
void Engine::AtForkChild() {
  AtForkParent();  // release the mutexes taken by prepare
  impl_->forked_.store(true, std::memory_order_release);
}

void Engine::Impl::ReinitAfterFork() {
  for (auto& t : cpu_workers_)
    if (t.joinable()) t.detach();  // handles of threads that died in fork
  cpu_workers_.clear();
  for (auto& kv : devices_) {
    if (kv.second->compute_t.joinable()) kv.second->compute_t.detach();
    if (kv.second->copy_t.joinable()) kv.second->copy_t.detach();
    if (kv.second->comm_t.joinable()) kv.second->comm_t.detach();
  }
  devices_.clear();  // HIP contexts do not survive fork
  cpu_q_.ResetAfterFork();
  cpu_prio_q_.ResetAfterFork();
  new (&done_cv_) std::condition_variable();
  inflight_ = 0;
  StartCPUWorkers();
}

void Engine::StopWorkers() {
  impl_->cpu_q_.Shutdown();
  impl_->cpu_prio_q_.Shutdown();
  for (auto& t : impl_->cpu_workers_)
    if (t.joinable()) t.join();
  impl_->cpu_workers_.clear();
  std::lock_guard<std::mutex> g(impl_->dev_mu_);
  for (auto& kv : impl_->devices_) {
    kv.second->compute_q.Shutdown();
    kv.second->copy_q.Shutdown();
    kv.second->comm_q.Shutdown();
    if (kv.second->compute_t.joinable()) kv.second->compute_t.join();
    if (kv.second->copy_t.joinable()) kv.second->copy_t.join();
    if (kv.second->comm_t.joinable()) kv.second->comm_t.join();
  }
}

}  // namespace mxcore
