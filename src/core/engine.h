// ThreadedEngine — the dependency scheduler that orders every operation
// (GPU kernels, copies, collectives, CPU work) over versioned variables.
//
// Reference parity: include/mxnet/engine.h:117 (NewVariable/PushAsync/
// WaitForVar/WaitForAll, FnProperty), src/engine/threaded_engine.h:120-229
// (per-var FIFO of read/write accesses), threaded_engine_perdevice.cc
// (per-device worker pools with dedicated streams).
//
// MI355X-first redesign — "submission-complete" scheduling:
// the reference's GPU workers synchronize their stream after every op and
// paper over the resulting launch gaps with op bulking.  Here an op is
// engine-complete when its kernels are SUBMITTED: every GPU op for a
// device is launched on that device's compute stream in dependency order,
// so stream order == topological order and no host sync is needed between
// ops.  Execution-completion is tracked by a hipEvent recorded per op into
// each written var; consumers on a different stream insert
// hipStreamWaitEvent, CPU consumers / WaitForVar do hipEventSynchronize.
// The host therefore only blocks at explicit sync points — zero launch
// gaps by construction instead of bulk-sized ones.
#pragma once

#include <condition_variable>
#include <tuple>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <queue>
#include <thread>
#include <unordered_map>
#include <vector>

#include "base.h"

namespace mxcore {

// reference engine.h:95
enum class FnProperty {
  kNormal,
  kCopyFromGPU,
  kCopyToGPU,
  kCPUPrioritized,
  kGPUPrioritized,  // comm stream (RCCL)
  kAsync,
  kDeleteVar,
};

struct RunContext {
  Context ctx;
  hipStream_t stream = nullptr;  // null on CPU
};

using VarId = int64_t;
using OpFn = std::function<void(const RunContext&)>;

class Engine {
 public:
  static Engine* Get();

  VarId NewVariable();
  // free the variable once all pending ops on it completed; on_delete (may
  // be null) runs right before the var dies — used to free NDArray storage
  void PushDeleteVariable(VarId v, OpFn on_delete);

  void PushAsync(OpFn fn, Context ctx, const std::vector<VarId>& const_vars,
                 const std::vector<VarId>& mutable_vars,
                 FnProperty prop = FnProperty::kNormal,
                 const char* name = nullptr);

  void WaitForVar(VarId v);   // queue drained + execution finished
  void WaitForAll();
  uint64_t Version(VarId v);
  void Throw(VarId v);        // rethrow a var's captured exception, if any

  // event of the last write to v (null if none / already retired);
  // callers use it for cross-stream hipStreamWaitEvent
  hipEvent_t LastEvent(VarId v);

  // streams (created lazily per device)
  hipStream_t ComputeStream(int dev);
  hipStream_t CopyStream(int dev);
  hipStream_t CommStream(int dev);

  // ---- hipGraph capture on the compute path -------------------------
  // While capturing, GPU compute ops still flow through the var tracking
  // but their launches are captured into a graph instead of executing.
  void BeginCapture(int dev);
  uintptr_t EndCapture(int dev);      // returns hipGraphExec_t
  // optional read-deps order the launch behind e.g. comm-stream work
  void ReleaseGraph(uintptr_t exec);  // destroy exec + pool its keepalive
  void LaunchGraph(int dev, uintptr_t exec,
                   const std::vector<VarId>& after = {},
                   const std::vector<VarId>& mutate = {});

  // ---- profiler (reference src/profiler: per-op aggregate stats) ----
  void SetProfiling(bool on);
  // name -> (calls, gpu_ms or cpu_ms); drains and resets the buffer
  std::vector<std::tuple<std::string, long, double>> ProfilerSummary();

  void StopWorkers();
  // pthread_atfork hooks: drain before fork, rebuild workers in child
  void AtForkPrepare();
  void AtForkParent();
  void AtForkChild();  // tests / atfork

 private:
  Engine();
  struct Impl;
  Impl* impl_;
};

}  // namespace mxcore
