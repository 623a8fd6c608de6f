// mxnet_amd._core — Python bindings for the native runtime:
// Storage (pooled HIP allocator), ThreadedEngine (HIP streams/events),
// NDArray (chunk/view), op registry + imperative invoke + autograd tape.
#include <pthread.h>
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>

#include "engine.h"
#include "ndarray.h"
#include "op.h"
#include "storage.h"

namespace py = pybind11;
using namespace mxcore;

namespace {

Context MakeCtx(int dev_type, int dev_id) { return Context{dev_type, dev_id}; }

int NumpyToFlag(const py::dtype& dt) {
  switch (dt.char_()) {
    case 'f': return dt.itemsize() == 4 ? kFloat32 : kFloat64;
    case 'e': return kFloat16;
    case 'd': return kFloat64;
    case 'i': return dt.itemsize() == 4 ? kInt32 : kInt64;
    case 'l': return kInt64;
    case 'q': return kInt64;
    case 'B': return kUint8;
    case 'b': return kInt8;
    case '?': return kBool;
    default: break;
  }
  throw std::runtime_error("unsupported numpy dtype");
}

py::dtype FlagToNumpy(int flag) {
  switch (flag) {
    case kFloat32: return py::dtype("float32");
    case kFloat64: return py::dtype("float64");
    case kFloat16: return py::dtype("float16");
    case kBFloat16: return py::dtype("uint16");  // bit-pattern view
    case kInt32: return py::dtype("int32");
    case kInt64: return py::dtype("int64");
    case kUint8: return py::dtype("uint8");
    case kInt8: return py::dtype("int8");
    case kBool: return py::dtype("bool");
    default: throw std::runtime_error("unsupported dtype flag");
  }
}

NodeAttrs DictToAttrs(const py::dict& d) {
  NodeAttrs a;
  for (auto kv : d)
    a.d[py::str(kv.first).cast<std::string>()] =
        py::str(kv.second).cast<std::string>();
  return a;
}

NDArray FromNumpy(py::array arr, int dev_type, int dev_id) {
  py::array c = py::array::ensure(arr, py::array::c_style);
  TShape shape(c.ndim());
  for (int i = 0; i < c.ndim(); ++i) shape[i] = c.shape(i);
  int flag = NumpyToFlag(c.dtype());
  Context cpu = Context::CPU();
  NDArray host(shape, cpu, flag);
  std::memcpy(host.dptr(), c.data(), (size_t)host.size() * dtype_size(flag));
  if (dev_type == Context::kCPU) return host;
  NDArray dst(shape, MakeCtx(dev_type, dev_id), flag);
  CopyFromTo(host, dst);
  return dst;
}

py::array ToNumpy(const NDArray& a) {
  std::vector<ssize_t> shape(a.shape().begin(), a.shape().end());
  py::array out(FlagToNumpy(a.dtype()), shape);
  size_t nbytes = (size_t)a.size() * dtype_size(a.dtype());
  if (a.ctx().is_gpu()) {
    NDArray host(a.shape(), Context::CPU(), a.dtype());
    CopyFromTo(a, host);
    host.WaitToRead();
    Engine::Get()->Throw(host.var());
    std::memcpy(out.mutable_data(), host.dptr(), nbytes);
  } else {
    {
      py::gil_scoped_release rel;
      a.WaitToRead();
    }
    std::memcpy(out.mutable_data(), a.dptr(), nbytes);
  }
  return out;
}

std::vector<NDArray> InvokePy(const std::string& name,
                              const std::vector<NDArray>& inputs,
                              const py::dict& attrs) {
  OpEntry* op = OpRegistry::Get()->Find(name);
  if (!op) throw std::runtime_error("op not registered: " + name);
  NodeAttrs a = DictToAttrs(attrs);
  py::gil_scoped_release rel;
  return Imperative::Get()->Invoke(op, a, inputs);
}

void InvokeIntoPy(const std::string& name, const std::vector<NDArray>& inputs,
                  const std::vector<NDArray>& outputs, const py::dict& attrs) {
  OpEntry* op = OpRegistry::Get()->Find(name);
  if (!op) throw std::runtime_error("op not registered: " + name);
  NodeAttrs a = DictToAttrs(attrs);
  py::gil_scoped_release rel;
  Imperative::Get()->InvokeInto(op, a, inputs, outputs);
}

}  // namespace

void init_raw(py::module_& m);

namespace mxcore {
void RcclInit(int world, int rank, int dev);
int RcclWorld();
int RcclRank();
void RcclAllReduce(const NDArray& a, bool average);
void RcclBroadcast(const NDArray& a, int root);
}  // namespace mxcore

PYBIND11_MODULE(_core, m) {
  init_raw(m);
  m.doc() = "mxnet_amd native runtime (storage + engine + ndarray + ops)";

  // stop the engine's worker threads BEFORE the interpreter/HIP runtime
  // tear down — workers still polling streams during libc/HSA teardown
  // corrupt the heap at exit (reference: engine shutdown in
  // LibraryInitializer, src/initialize.cc)
  {
    auto atexit = py::module_::import("atexit");
    atexit.attr("register")(py::cpp_function([]() {
      {
        py::gil_scoped_release rel;
        try {
          Engine::Get()->WaitForAll();
        } catch (...) {
        }
        Engine::Get()->StopWorkers();
      }
    }));
  }

  // fork safety (reference LibraryInitializer pthread_atfork): drain
  // the engine before a fork; the child detaches the dead worker
  // handles, drops GPU state and restarts CPU workers — so python
  // multiprocessing (fork start method) keeps working after native ops
  // ran in the parent
  pthread_atfork([] { Engine::Get()->AtForkPrepare(); },
                 [] { Engine::Get()->AtForkParent(); },
                 [] { Engine::Get()->AtForkChild(); });

  py::class_<NDArray>(m, "NDArray")
      .def(py::init([](const std::vector<int64_t>& shape, int dev_type,
                       int dev_id, int dtype) {
             return NDArray(TShape(shape.begin(), shape.end()),
                            MakeCtx(dev_type, dev_id), dtype);
           }),
           py::arg("shape"), py::arg("dev_type") = 1, py::arg("dev_id") = 0,
           py::arg("dtype") = 0)
      .def_property_readonly("shape",
                             [](const NDArray& a) {
                               return std::vector<int64_t>(a.shape());
                             })
      .def_property_readonly("dtype", &NDArray::dtype)
      .def_property_readonly("size", &NDArray::size)
      .def_property_readonly(
          "ctx",
          [](const NDArray& a) {
            return py::make_tuple(a.ctx().dev_type, a.ctx().dev_id);
          })
      .def_property_readonly("var", &NDArray::var)
      .def_property_readonly("handle_id",
                             [](const NDArray& a) {
                               return (uintptr_t)a.chunk_.get();
                             })
      .def_property_readonly("data_ptr",
                             [](const NDArray& a) {
                               return (uintptr_t)a.dptr();
                             })
      .def("reshape",
           [](const NDArray& a, const std::vector<int64_t>& s) {
             return a.Reshape(TShape(s.begin(), s.end()));
           })
      .def("slice", &NDArray::Slice)
      .def("astype_view", &NDArray::AsType)
      .def("wait_to_read",
           [](const NDArray& a) {
             {
               py::gil_scoped_release rel;
               a.WaitToRead();
             }
             Engine::Get()->Throw(a.var());
           })
      .def("asnumpy", &ToNumpy)
      .def("copyto",
           [](const NDArray& src, const NDArray& dst) {
             py::gil_scoped_release rel;
             CopyFromTo(src, dst);
           })
      // pickle via numpy round-trip (multiprocessing DataLoader workers
      // ship samples/batches through ForkingPickler; GPU arrays
      // re-materialize on the SAME device id in the consumer)
      .def(py::pickle(
          [](const NDArray& a) {
            return py::make_tuple(ToNumpy(a), a.ctx().dev_type,
                                  a.ctx().dev_id, a.dtype());
          },
          [](py::tuple t) {
            NDArray host = FromNumpy(t[0].cast<py::array>(), 1, 0);
            int dev_type = t[1].cast<int>();
            int dev_id = t[2].cast<int>();
            int dtype = t[3].cast<int>();
            if (host.dtype() != dtype) host = host.AsType(dtype);
            if (dev_type == 1) return host;
            NDArray dst(host.shape(), MakeCtx(dev_type, dev_id), dtype);
            CopyFromTo(host, dst);
            return dst;
          }));

  m.def("from_numpy", &FromNumpy, py::arg("array"), py::arg("dev_type") = 1,
        py::arg("dev_id") = 0);
  m.def("invoke", &InvokePy);
  m.def("invoke_into", &InvokeIntoPy);
  m.def("list_ops", [] { return OpRegistry::Get()->List(); });
  m.def("has_op", [](const std::string& n) {
    return OpRegistry::Get()->Find(n) != nullptr;
  });

  // engine
  m.def("wait_all", [] {
    py::gil_scoped_release rel;
    Engine::Get()->WaitForAll();
  });
  m.def("new_variable", [] { return Engine::Get()->NewVariable(); });
  m.def("var_version", [](int64_t v) { return Engine::Get()->Version(v); });
  m.def("begin_capture", [](int dev) {
    py::gil_scoped_release rel;
    Engine::Get()->BeginCapture(dev);
  });
  m.def("end_capture", [](int dev) {
    py::gil_scoped_release rel;
    return Engine::Get()->EndCapture(dev);
  });
  m.def("launch_graph",
        [](int dev, uintptr_t exec, const std::vector<NDArray>& after,
           const std::vector<NDArray>& mutate) {
          std::vector<VarId> rvars, wvars;
          for (auto& a : after) rvars.push_back(a.var());
          for (auto& a : mutate) wvars.push_back(a.var());
          py::gil_scoped_release rel;
          Engine::Get()->LaunchGraph(dev, exec, rvars, wvars);
        },
        py::arg("dev"), py::arg("exec"),
        py::arg("after") = std::vector<NDArray>(),
        py::arg("mutate") = std::vector<NDArray>());
  m.def("release_graph", [](uintptr_t exec) {
    py::gil_scoped_release rel;
    Engine::Get()->ReleaseGraph(exec);
  });

  // autograd tape
  m.def("set_recording", [](bool r) {
    bool old = Imperative::Get()->is_recording();
    Imperative::Get()->set_recording(r);
    return old;
  });
  m.def("is_recording", [] { return Imperative::Get()->is_recording(); });
  m.def("set_training", [](bool t) {
    bool old = Imperative::Get()->is_training();
    Imperative::Get()->set_training(t);
    return old;
  });
  m.def("is_training", [] { return Imperative::Get()->is_training(); });
  m.def("mark_variable",
        [](const NDArray& x, const NDArray& g, int req) {
          Imperative::Get()->MarkVariable(x, g, req);
        });
  m.def("drop_variable",
        [](const NDArray& x) { Imperative::Get()->DropVariable(x); });
  m.def("backward", [](const std::vector<NDArray>& ys,
                       const std::vector<NDArray>& ygrads, bool retain) {
    py::gil_scoped_release rel;
    Imperative::Get()->Backward(ys, ygrads, retain);
  });
  m.def("clear_tape", [] { Imperative::Get()->ClearTape(); });
  m.def("tape_size", [] { return Imperative::Get()->TapeSize(); });

  // profiler (aggregate per-op stats, reference AggregateStats)
  m.def("profiler_set_state", [](bool on) {
    Engine::Get()->SetProfiling(on);
  });
  m.def("profiler_summary", [] {
    py::gil_scoped_release rel;
    return Engine::Get()->ProfilerSummary();
  });

  // storage telemetry
  m.def("pool_size", [](int dev_type, int dev_id) {
    return Storage::Get()->PoolSize(MakeCtx(dev_type, dev_id));
  });
  m.def("used_size", [](int dev_type, int dev_id) {
    return Storage::Get()->UsedSize(MakeCtx(dev_type, dev_id));
  });
  m.def("release_all", [](int dev_type, int dev_id) {
    Storage::Get()->ReleaseAll(MakeCtx(dev_type, dev_id));
  });

  // RCCL collectives (engine-sequenced on the comm stream)
  m.def("rccl_init", [](int world, int rank, int dev) {
    py::gil_scoped_release rel;
    RcclInit(world, rank, dev);
  });
  m.def("rccl_world", &RcclWorld);
  m.def("rccl_rank", &RcclRank);
  m.def("rccl_allreduce", [](const NDArray& a, bool average) {
    py::gil_scoped_release rel;
    RcclAllReduce(a, average);
  });
  m.def("rccl_broadcast", [](const NDArray& a, int root) {
    py::gil_scoped_release rel;
    RcclBroadcast(a, root);
  });

  m.def("device_count", [] {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) {
      (void)hipGetLastError();
      return 0;
    }
    return n;
  });
}
