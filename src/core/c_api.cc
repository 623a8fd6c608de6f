// C ABI implementation over the native runtime (see
// include/mxnet_amd/c_api.h; reference src/c_api/c_api.cc).
#include "../../include/mxnet_amd/c_api.h"

#include <dlfcn.h>

#include <cstring>
#include <fstream>
#include <string>
#include <vector>

#include "ndarray.h"
#include "op.h"

using namespace mxcore;

namespace {

thread_local std::string g_last_error;
// per-thread scratch that outlives the call so returned pointers stay valid
thread_local std::vector<int64_t> g_shape_buf;
thread_local std::vector<NDArrayHandle> g_out_buf;
thread_local std::vector<std::string> g_name_store;
thread_local std::vector<const char*> g_name_buf;

#define API_BEGIN() try {
#define API_END()                          \
  }                                        \
  catch (const std::exception& e) {        \
    g_last_error = e.what();               \
    return -1;                             \
  }                                        \
  return 0;

NDArray* ND(NDArrayHandle h) { return static_cast<NDArray*>(h); }

}  // namespace

const char* MXGetLastError() { return g_last_error.c_str(); }

int MXNDArrayCreate(const int64_t* shape, int ndim, int dev_type,
                    int dev_id, int dtype, NDArrayHandle* out) {
  API_BEGIN()
  TShape s(shape, shape + ndim);
  *out = new NDArray(s, Context{dev_type, dev_id}, dtype);
  API_END()
}

int MXNDArrayFree(NDArrayHandle h) {
  API_BEGIN()
  delete ND(h);
  API_END()
}

int MXNDArrayGetShape(NDArrayHandle h, int* ndim, const int64_t** shape) {
  API_BEGIN()
  g_shape_buf = ND(h)->shape();
  *ndim = (int)g_shape_buf.size();
  *shape = g_shape_buf.data();
  API_END()
}

int MXNDArrayGetDType(NDArrayHandle h, int* dtype) {
  API_BEGIN()
  *dtype = ND(h)->dtype();
  API_END()
}

int MXNDArrayGetContext(NDArrayHandle h, int* dev_type, int* dev_id) {
  API_BEGIN()
  *dev_type = ND(h)->ctx().dev_type;
  *dev_id = ND(h)->ctx().dev_id;
  API_END()
}

int MXNDArraySyncCopyFromCPU(NDArrayHandle h, const void* data,
                             size_t nbytes) {
  API_BEGIN()
  NDArray* a = ND(h);
  MX_CHECK(nbytes == (size_t)a->size() * dtype_size(a->dtype()),
           "size mismatch");
  if (a->ctx().is_gpu()) {
    NDArray host(a->shape(), Context::CPU(), a->dtype());
    std::memcpy(host.dptr(), data, nbytes);
    CopyFromTo(host, *a);
    Engine::Get()->WaitForVar(a->var());
  } else {
    Engine::Get()->WaitForVar(a->var());
    std::memcpy(a->dptr(), data, nbytes);
  }
  API_END()
}

int MXNDArraySyncCopyToCPU(NDArrayHandle h, void* data, size_t nbytes) {
  API_BEGIN()
  NDArray* a = ND(h);
  MX_CHECK(nbytes == (size_t)a->size() * dtype_size(a->dtype()),
           "size mismatch");
  if (a->ctx().is_gpu()) {
    NDArray host(a->shape(), Context::CPU(), a->dtype());
    CopyFromTo(*a, host);
    host.WaitToRead();
    Engine::Get()->Throw(host.var());
    std::memcpy(data, host.dptr(), nbytes);
  } else {
    a->WaitToRead();
    Engine::Get()->Throw(a->var());
    std::memcpy(data, a->dptr(), nbytes);
  }
  API_END()
}

int MXNDArrayWaitToRead(NDArrayHandle h) {
  API_BEGIN()
  ND(h)->WaitToRead();
  Engine::Get()->Throw(ND(h)->var());
  API_END()
}

int MXNDArrayWaitAll() {
  API_BEGIN()
  Engine::Get()->WaitForAll();
  API_END()
}

int MXImperativeInvoke(const char* op_name, int num_inputs,
                       NDArrayHandle* inputs, int* num_outputs,
                       NDArrayHandle** outputs, int num_attrs,
                       const char** attr_keys, const char** attr_vals) {
  API_BEGIN()
  OpEntry* op = OpRegistry::Get()->Find(op_name);
  MX_CHECK(op, "op not registered: " << op_name);
  NodeAttrs attrs;
  for (int i = 0; i < num_attrs; ++i) attrs.d[attr_keys[i]] = attr_vals[i];
  std::vector<NDArray> in;
  for (int i = 0; i < num_inputs; ++i) in.push_back(*ND(inputs[i]));
  auto outs = Imperative::Get()->Invoke(op, attrs, in);
  g_out_buf.clear();
  for (auto& o : outs) g_out_buf.push_back(new NDArray(o));
  *num_outputs = (int)outs.size();
  *outputs = g_out_buf.data();
  API_END()
}

int MXListOps(int* count, const char*** names) {
  API_BEGIN()
  auto ops = OpRegistry::Get()->List();
  g_name_store.assign(ops.begin(), ops.end());
  g_name_buf.clear();
  for (auto& s : g_name_store) g_name_buf.push_back(s.c_str());
  *count = (int)g_name_buf.size();
  *names = g_name_buf.data();
  API_END()
}

int MXAutogradSetIsRecording(int recording, int* prev) {
  API_BEGIN()
  bool old = Imperative::Get()->is_recording();
  Imperative::Get()->set_recording(recording != 0);
  if (prev) *prev = old ? 1 : 0;
  API_END()
}

int MXAutogradMarkVariables(int num, NDArrayHandle* vars,
                            NDArrayHandle* grads, const int* reqs) {
  API_BEGIN()
  for (int i = 0; i < num; ++i)
    Imperative::Get()->MarkVariable(*ND(vars[i]), *ND(grads[i]),
                                    reqs ? reqs[i] : 1);
  API_END()
}

int MXAutogradBackward(int num_heads, NDArrayHandle* heads,
                       NDArrayHandle* head_grads, int retain_graph) {
  API_BEGIN()
  std::vector<NDArray> ys, gs;
  for (int i = 0; i < num_heads; ++i) {
    ys.push_back(*ND(heads[i]));
    if (head_grads && head_grads[i]) gs.push_back(*ND(head_grads[i]));
  }
  Imperative::Get()->Backward(ys, gs, retain_graph != 0);
  API_END()
}

// ---------------------------------------------------------------------------
// external operator libraries (reference lib_api.h / MXLoadLib)
// ---------------------------------------------------------------------------
namespace {

std::vector<MXTensorView> ViewsOf(const std::vector<TBlob>& blobs,
                                  std::vector<std::vector<int64_t>>* keep) {
  std::vector<MXTensorView> v(blobs.size());
  keep->resize(blobs.size());
  for (size_t i = 0; i < blobs.size(); ++i) {
    (*keep)[i] = blobs[i].shape;
    v[i].ndim = (int)(*keep)[i].size();
    v[i].shape = (*keep)[i].data();
    v[i].dtype = blobs[i].dtype;
    v[i].data = blobs[i].dptr;
  }
  return v;
}

void RegisterExternalOp(void*, const MXCustomOpDef* def) {
  // copy the def — the pointer is only valid during lib init
  MXCustomOpDef d = *def;
  std::string name = def->name;
  OpEntry& e = OpRegistry::Get()->Register(name);
  e.n_in = d.n_in;
  e.n_out = 1;
  MXCustomInferFn infer = d.infer;
  e.infer = [infer, name](const NodeAttrs&, const std::vector<TShape>& is,
                          const std::vector<int>& it,
                          std::vector<TShape>* os, std::vector<int>* ot) {
    std::vector<std::vector<int64_t>> keep(is.size());
    std::vector<MXTensorView> ins(is.size());
    for (size_t i = 0; i < is.size(); ++i) {
      keep[i] = is[i];
      ins[i].ndim = (int)keep[i].size();
      ins[i].shape = keep[i].data();
      ins[i].dtype = it[i];
      ins[i].data = nullptr;  // shapes only at infer time
    }
    int64_t oshape[8];
    int ondim = 0, odtype = it.empty() ? kFloat32 : it[0];
    MX_CHECK(infer((int)ins.size(), ins.data(), oshape, &ondim,
                   &odtype) == 0,
             "external op '" << name << "': infer failed");
    MX_CHECK(ondim >= 0 && ondim <= 8, "external op: bad out ndim");
    os->assign(1, TShape(oshape, oshape + ondim));
    ot->assign(1, odtype);
  };
  if (d.fcompute_cpu) {
    MXCustomComputeFn f = d.fcompute_cpu;
    e.fcompute_cpu = [f, name](const NodeAttrs&, const OpCtx&,
                               const std::vector<TBlob>& in,
                               const std::vector<TBlob>& out) {
      std::vector<std::vector<int64_t>> ki, ko;
      auto iv = ViewsOf(in, &ki);
      auto ov = ViewsOf(out, &ko);
      MX_CHECK(f((int)iv.size(), iv.data(), ov.data(), nullptr) == 0,
               "external op '" << name << "' failed (cpu)");
    };
  }
  if (d.fcompute_gpu) {
    MXCustomComputeFn f = d.fcompute_gpu;
    e.fcompute_gpu = [f, name](const NodeAttrs&, const OpCtx& o,
                               const std::vector<TBlob>& in,
                               const std::vector<TBlob>& out) {
      std::vector<std::vector<int64_t>> ki, ko;
      auto iv = ViewsOf(in, &ki);
      auto ov = ViewsOf(out, &ko);
      MX_CHECK(f((int)iv.size(), iv.data(), ov.data(),
                 (void*)o.rc.stream) == 0,
               "external op '" << name << "' failed (gpu)");
    };
  }
}

}  // namespace

int MXLoadLib(const char* path) {
  API_BEGIN()
  void* h = dlopen(path, RTLD_NOW | RTLD_LOCAL);
  MX_CHECK(h, "MXLoadLib: dlopen failed: " << dlerror());
  using InitFn = int (*)(MXRegisterOpFn, void*);
  auto init = (InitFn)dlsym(h, "mxnet_amd_lib_init");
  MX_CHECK(init, "MXLoadLib: library has no mxnet_amd_lib_init");
  MX_CHECK(init(&RegisterExternalOp, nullptr) == 0,
           "MXLoadLib: library init returned nonzero");
  API_END()
}

// ---------------------------------------------------------------------------
// .params serialization (reference ndarray.cc:1729)
// ---------------------------------------------------------------------------
namespace {
constexpr uint64_t kListMagic = 0x112;
constexpr uint32_t kV2Magic = 0xF993fac9;
constexpr uint32_t kV1Magic = 0xF993fac8;

template <typename T>
void W(std::ostream& os, T v) {
  os.write((const char*)&v, sizeof(v));
}
template <typename T>
T R(std::istream& is) {
  T v{};
  is.read((char*)&v, sizeof(v));
  return v;
}
}  // namespace

int MXNDArraySave(const char* fname, int num, NDArrayHandle* arrays,
                  const char** names) {
  API_BEGIN()
  std::ofstream f(fname, std::ios::binary);
  MX_CHECK(f.good(), "cannot open " << fname);
  W<uint64_t>(f, kListMagic);
  W<uint64_t>(f, 0);
  W<uint64_t>(f, (uint64_t)num);
  for (int i = 0; i < num; ++i) {
    NDArray* a = ND(arrays[i]);
    // pull bytes to host
    std::vector<char> buf((size_t)a->size() * dtype_size(a->dtype()));
    int rc = MXNDArraySyncCopyToCPU(arrays[i], buf.data(), buf.size());
    MX_CHECK(rc == 0, g_last_error);
    W<uint32_t>(f, kV2Magic);
    W<int32_t>(f, 0);  // dense
    W<int32_t>(f, (int32_t)a->shape().size());
    for (auto d : a->shape()) W<int64_t>(f, d);
    W<int32_t>(f, 1);  // ctx cpu
    W<int32_t>(f, 0);
    W<int32_t>(f, a->dtype());
    f.write(buf.data(), buf.size());
  }
  W<uint64_t>(f, names ? (uint64_t)num : 0);
  if (names)
    for (int i = 0; i < num; ++i) {
      uint64_t len = strlen(names[i]);
      W<uint64_t>(f, len);
      f.write(names[i], len);
    }
  API_END()
}

int MXNDArrayLoad(const char* fname, int* out_count,
                  NDArrayHandle** out_arrays, const char*** out_names) {
  API_BEGIN()
  std::ifstream f(fname, std::ios::binary);
  MX_CHECK(f.good(), "cannot open " << fname);
  MX_CHECK(R<uint64_t>(f) == kListMagic, "not an NDArray list file");
  R<uint64_t>(f);
  uint64_t count = R<uint64_t>(f);
  g_out_buf.clear();
  for (uint64_t i = 0; i < count; ++i) {
    uint32_t magic = R<uint32_t>(f);
    int32_t ndim;
    if (magic == kV2Magic || magic == kV2Magic + 1) {
      int32_t stype = R<int32_t>(f);
      MX_CHECK(stype == 0, "sparse load: not supported in the C ABI");
      ndim = R<int32_t>(f);
    } else if (magic == kV1Magic) {
      ndim = R<int32_t>(f);
    } else {
      MX_CHECK(false, "unsupported NDArray record magic");
      ndim = 0;
    }
    if (ndim <= 0) {  // none array
      g_out_buf.push_back(new NDArray(TShape{0}, Context::CPU(), 0));
      continue;
    }
    TShape shape(ndim);
    for (int d = 0; d < ndim; ++d) shape[d] = R<int64_t>(f);
    R<int32_t>(f);  // dev_type
    R<int32_t>(f);  // dev_id
    int32_t dtype = R<int32_t>(f);
    auto* a = new NDArray(shape, Context::CPU(), dtype);
    f.read((char*)a->dptr(),
           (size_t)a->size() * dtype_size(dtype));
    g_out_buf.push_back(a);
  }
  uint64_t ncount = 0;
  if (f.peek() != EOF) ncount = R<uint64_t>(f);
  g_name_store.clear();
  g_name_buf.clear();
  for (uint64_t i = 0; i < ncount; ++i) {
    uint64_t len = R<uint64_t>(f);
    std::string s(len, '\0');
    f.read(&s[0], len);
    g_name_store.push_back(std::move(s));
  }
  for (auto& s : g_name_store) g_name_buf.push_back(s.c_str());
  *out_count = (int)g_out_buf.size();
  *out_arrays = g_out_buf.data();
  *out_names = g_name_buf.data();
  API_END()
}
