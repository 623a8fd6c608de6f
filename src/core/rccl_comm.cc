// RCCL collectives integrated with the native engine.
//
// Reference parity: src/kvstore/kvstore_nccl.h:62 — there, grouped
// ncclReduce/ncclBcast pairs are sequenced by a CPU engine op plus a
// separate stream-sync op (the 3-op pattern, :267-443).  MI355X redesign:
// one process per GPU over xGMI; each collective is ONE engine op on the
// device's dedicated comm stream (FnProperty::kGPUPrioritized).  The
// engine's generic event plumbing gives the reference's guarantees for
// free: the comm worker hipStreamWaitEvents the producing compute kernels
// before launching rcclAllReduce, records a completion event afterwards,
// and any consumer (optimizer update on the compute stream) waits that
// event — comm/compute overlap with no host synchronisation.
//
// Bootstrap: rank 0 creates the rcclUniqueId and serves it over a TCP
// socket on MASTER_ADDR:MASTER_PORT+1 (single-node xGMI is the target;
// the same exchange works multi-node).
#include <arpa/inet.h>
#include <netinet/in.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <mutex>
#include <thread>

#include <dlfcn.h>

#include <rccl/rccl.h>

#include "engine.h"
#include "ndarray.h"

namespace mxcore {

// librccl is resolved with dlopen at init time instead of link-time:
// torch-ROCm already maps its own librccl into the process, and loading a
// SECOND copy (ld-linked /opt/rocm one) corrupts shared HSA state — the
// soname lookup below returns the already-loaded copy when there is one.
namespace {

struct RcclFns {
  ncclResult_t (*GetUniqueId)(ncclUniqueId*);
  ncclResult_t (*CommInitRank)(ncclComm_t*, int, ncclUniqueId, int);
  ncclResult_t (*AllReduce)(const void*, void*, size_t, ncclDataType_t,
                            ncclRedOp_t, ncclComm_t, hipStream_t);
  ncclResult_t (*Broadcast)(const void*, void*, size_t, ncclDataType_t,
                            int, ncclComm_t, hipStream_t);
  const char* (*GetErrorString)(ncclResult_t);
};

RcclFns* GetRccl() {
  static RcclFns* fns = [] {
    void* h = dlopen("librccl.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("librccl.so", RTLD_NOW | RTLD_GLOBAL);
    MX_CHECK(h, "librccl not found: " << dlerror());
    auto* f = new RcclFns();
    f->GetUniqueId =
        (decltype(f->GetUniqueId))dlsym(h, "ncclGetUniqueId");
    f->CommInitRank =
        (decltype(f->CommInitRank))dlsym(h, "ncclCommInitRank");
    f->AllReduce = (decltype(f->AllReduce))dlsym(h, "ncclAllReduce");
    f->Broadcast = (decltype(f->Broadcast))dlsym(h, "ncclBroadcast");
    f->GetErrorString =
        (decltype(f->GetErrorString))dlsym(h, "ncclGetErrorString");
    MX_CHECK(f->GetUniqueId && f->CommInitRank && f->AllReduce &&
                 f->Broadcast && f->GetErrorString,
             "librccl symbols missing");
    return f;
  }();
  return fns;
}

}  // namespace

#define MX_RCCL_CALL(expr)                                               \
  do {                                                                   \
    ncclResult_t r_ = (expr);                                            \
    MX_CHECK(r_ == ncclSuccess, "RCCL error: "                           \
                                    << GetRccl()->GetErrorString(r_)     \
                                    << " at " #expr);                    \
  } while (0)

namespace {

struct RcclState {
  ncclComm_t comm = nullptr;
  int world = 1, rank = 0, dev = 0;
};
RcclState g_rccl;
std::mutex g_mu;

// exchange the unique id through a short-lived TCP socket
void ExchangeId(ncclUniqueId* id, int world, int rank) {
  const char* addr = getenv("MASTER_ADDR");
  const char* port_s = getenv("MASTER_PORT");
  int port = (port_s ? atoi(port_s) : 29500) + 1;
  std::string host = addr ? addr : "127.0.0.1";
  if (rank == 0) {
    int srv = socket(AF_INET, SOCK_STREAM, 0);
    MX_CHECK(srv >= 0, "rccl bootstrap: socket failed");
    int one = 1;
    setsockopt(srv, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in sa{};
    sa.sin_family = AF_INET;
    sa.sin_addr.s_addr = INADDR_ANY;
    sa.sin_port = htons(port);
    MX_CHECK(bind(srv, (sockaddr*)&sa, sizeof(sa)) == 0,
             "rccl bootstrap: bind failed on port " << port);
    listen(srv, world);
    for (int i = 1; i < world; ++i) {
      int cl = accept(srv, nullptr, nullptr);
      MX_CHECK(cl >= 0, "rccl bootstrap: accept failed");
      size_t off = 0;
      while (off < sizeof(*id)) {
        ssize_t w = write(cl, (char*)id + off, sizeof(*id) - off);
        MX_CHECK(w > 0, "rccl bootstrap: write failed");
        off += w;
      }
      close(cl);
    }
    close(srv);
  } else {
    int cl = -1;
    for (int attempt = 0; attempt < 600; ++attempt) {
      cl = socket(AF_INET, SOCK_STREAM, 0);
      sockaddr_in sa{};
      sa.sin_family = AF_INET;
      sa.sin_port = htons(port);
      inet_pton(AF_INET, host.c_str(), &sa.sin_addr);
      if (connect(cl, (sockaddr*)&sa, sizeof(sa)) == 0) break;
      close(cl);
      cl = -1;
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
    MX_CHECK(cl >= 0, "rccl bootstrap: connect to " << host << ":" << port
                                                    << " failed");
    size_t off = 0;
    while (off < sizeof(*id)) {
      ssize_t r = read(cl, (char*)id + off, sizeof(*id) - off);
      MX_CHECK(r > 0, "rccl bootstrap: read failed");
      off += r;
    }
    close(cl);
  }
}

ncclDataType_t RcclType(int flag) {
  switch (flag) {
    case kFloat32: return ncclFloat32;
    case kFloat64: return ncclFloat64;
    case kFloat16: return ncclFloat16;
    case kBFloat16: return ncclBfloat16;
    case kInt32: return ncclInt32;
    case kInt64: return ncclInt64;
    case kUint8: return ncclUint8;
    case kInt8: return ncclInt8;
    default:
      MX_CHECK(false, "rccl: unsupported dtype " << dtype_name(flag));
      return ncclFloat32;
  }
}

}  // namespace

void RcclInit(int world, int rank, int dev) {
  std::lock_guard<std::mutex> g(g_mu);
  if (g_rccl.comm) return;
  MX_HIP_CALL(hipSetDevice(dev));
  ncclUniqueId id;
  if (rank == 0) MX_RCCL_CALL(GetRccl()->GetUniqueId(&id));
  if (world > 1) ExchangeId(&id, world, rank);
  MX_RCCL_CALL(GetRccl()->CommInitRank(&g_rccl.comm, world, id, rank));
  g_rccl.world = world;
  g_rccl.rank = rank;
  g_rccl.dev = dev;
  // create the comm stream/worker up front (lazy creation inside the
  // first collective would race the bootstrap)
  Engine::Get()->CommStream(dev);
}

int RcclWorld() { return g_rccl.comm ? g_rccl.world : 1; }
int RcclRank() { return g_rccl.rank; }

// in-place all-reduce, engine-sequenced on the comm stream; average=true
// divides by world inside the same pass
void RcclAllReduce(const NDArray& a, bool average) {
  MX_CHECK(g_rccl.comm, "RcclAllReduce before RcclInit");
  if (g_rccl.world == 1 && !average) return;
  auto chunk = a.chunk_;
  void* p = a.dptr();
  int64_t n = a.size();
  int dtype = a.dtype();
  ncclComm_t comm = g_rccl.comm;
  int world = g_rccl.world;
  Engine::Get()->PushAsync(
      [chunk, p, n, dtype, comm, world, average](const RunContext& rc) {
        if (world > 1)
          MX_RCCL_CALL(GetRccl()->AllReduce(p, p, n, RcclType(dtype),
                                            ncclSum, comm, rc.stream));
        // average in the same stream order (ring result / world)
        if (average && world > 1) {
          // scale via a small elementwise launch on the comm stream
          extern void ScaleInPlace(void* p, int64_t n, int dtype,
                                   float alpha, hipStream_t s);
          ScaleInPlace(p, n, dtype, 1.f / world, rc.stream);
        }
      },
      a.ctx(), {}, {a.var()}, FnProperty::kGPUPrioritized, "RcclAllReduce");
}

void RcclBroadcast(const NDArray& a, int root) {
  MX_CHECK(g_rccl.comm, "RcclBroadcast before RcclInit");
  if (g_rccl.world == 1) return;
  auto chunk = a.chunk_;
  void* p = a.dptr();
  int64_t n = a.size();
  int dtype = a.dtype();
  ncclComm_t comm = g_rccl.comm;
  Engine::Get()->PushAsync(
      [chunk, p, n, dtype, comm, root](const RunContext& rc) {
        MX_RCCL_CALL(GetRccl()->Broadcast(p, p, n, RcclType(dtype), root,
                                          comm, rc.stream));
      },
      a.ctx(), {}, {a.var()}, FnProperty::kGPUPrioritized, "RcclBroadcast");
}

}  // namespace mxcore
