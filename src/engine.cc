// ThreadedEngine — host-side async dependency scheduler (C++).
//
// Reference parity: include/mxnet/engine.h:117 (Engine interface),
// src/engine/threaded_engine.h:120-229 (ThreadedVar read/write queues),
// threaded_engine.cc:318 PushAsync / :379 WaitForVar / :416 WaitForAll /
// :441 OnComplete; naive_engine.cc (the sync fallback lives in python).
//
// MI355X-native split: GPU kernels are already asynchronous on HIP
// streams (stream order = dependency order), so THIS engine schedules
// the host-side async work the reference ran on its CPU worker threads:
// data-pipeline stages, checkpoint IO, CPU reduces.  Ops declare
// read (const_vars) / write (mutable_vars) sets over versioned Vars; a
// worker pool runs ops whose dependencies are satisfied; exceptions
// propagate through vars to WaitForVar/WaitForAll like the reference's
// ExceptionRef plumbing.
#include <pybind11/pybind11.h>

#include <condition_variable>
#include <cstdlib>
#include <deque>
#include <exception>
#include <memory>
#include <mutex>
#include <queue>
#include <thread>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace {

struct Opr;

struct Block {
  Opr* opr;
  bool write;
  bool granted = false;  // this access has its turn (counted in opr->wait)
};

// versioned variable with a FIFO of pending accesses (ThreadedVar)
struct Var {
  uint64_t version = 0;
  std::deque<Block> queue;  // pending accesses in program order
  int running_reads = 0;    // granted reads not yet completed
  std::exception_ptr exc;
};

struct Opr {
  py::object fn;
  std::vector<Var*> const_vars;
  std::vector<Var*> mutable_vars;
  int wait = 0;  // ungranted dependencies
};

class ThreadedEngine {
 public:
  explicit ThreadedEngine(int num_workers) { Start(num_workers); }

  void Start(int num_workers) {
    std::lock_guard<std::mutex> g(mu_);
    if (!workers_.empty()) return;
    shutdown_ = false;
    for (int i = 0; i < num_workers; ++i)
      workers_.emplace_back([this] { WorkerLoop(); });
  }

  void Stop() {
    {
      std::lock_guard<std::mutex> g(mu_);
      shutdown_ = true;
    }
    ready_cv_.notify_all();
    done_cv_.notify_all();
    for (auto& t : workers_)
      if (t.joinable()) t.join();
    workers_.clear();
  }

  int64_t NewVariable() {
    std::lock_guard<std::mutex> g(mu_);
    int64_t id = next_var_id_++;
    vars_.emplace(id, std::make_unique<Var>());
    return id;
  }

  void DeleteVariable(int64_t id) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = vars_.find(id);
    if (it != vars_.end() && it->second->queue.empty() &&
        it->second->running_reads == 0)
      vars_.erase(it);
  }

  uint64_t Version(int64_t id) {
    std::lock_guard<std::mutex> g(mu_);
    return GetVar(id)->version;
  }

  void Push(py::object fn, const std::vector<int64_t>& const_ids,
            const std::vector<int64_t>& mutable_ids) {
    auto* opr = new Opr();
    opr->fn = std::move(fn);
    bool ready;
    {
      std::lock_guard<std::mutex> g(mu_);
      ++inflight_;
      for (int64_t id : const_ids) opr->const_vars.push_back(GetVar(id));
      for (int64_t id : mutable_ids) opr->mutable_vars.push_back(GetVar(id));
      opr->wait = (int)(opr->const_vars.size() + opr->mutable_vars.size());
      // AppendReadDependency: a read is granted unless a write is queued
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
      // AppendWriteDependency: a write is granted only on an idle var
      for (Var* v : opr->mutable_vars) {
        bool idle = v->queue.empty() && v->running_reads == 0;
        v->queue.push_back({opr, true, idle});
        if (idle) --opr->wait;
      }
      ready = opr->wait == 0;
      if (ready) ready_q_.push(opr);
    }
    if (ready) ready_cv_.notify_one();
  }

  void WaitForVar(int64_t id) {
    std::exception_ptr exc;
    {
      std::unique_lock<std::mutex> lk(mu_);
      Var* v = GetVar(id);
      done_cv_.wait(lk, [&] {
        return (v->queue.empty() && v->running_reads == 0) || shutdown_;
      });
      exc = v->exc;
      v->exc = nullptr;
    }
    if (exc) std::rethrow_exception(exc);
  }

  void WaitForAll() {
    std::exception_ptr exc;
    {
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [&] { return inflight_ == 0 || shutdown_; });
      exc = global_exc_;
      global_exc_ = nullptr;
    }
    if (exc) std::rethrow_exception(exc);
  }

 private:
  Var* GetVar(int64_t id) {
    auto it = vars_.find(id);
    if (it == vars_.end())
      it = vars_.emplace(id, std::make_unique<Var>()).first;
    return it->second.get();
  }

  void WorkerLoop() {
    for (;;) {
      Opr* opr;
      {
        std::unique_lock<std::mutex> lk(mu_);
        ready_cv_.wait(lk, [&] { return shutdown_ || !ready_q_.empty(); });
        if (shutdown_ && ready_q_.empty()) return;
        opr = ready_q_.front();
        ready_q_.pop();
      }
      std::exception_ptr exc;
      {
        py::gil_scoped_acquire gil;
        try {
          opr->fn();
        } catch (py::error_already_set& e) {
          // capture as a plain C++ exception: a py exception object can
          // only be restored once, but this may surface at several
          // wait sites (reference flattens to dmlc::Error text too)
          exc = std::make_exception_ptr(
              std::runtime_error(std::string("engine op failed: ") +
                                 e.what()));
        } catch (const std::exception& e) {
          exc = std::make_exception_ptr(std::runtime_error(e.what()));
        } catch (...) {
          exc = std::make_exception_ptr(
              std::runtime_error("engine op failed (unknown exception)"));
        }
        opr->fn = py::object();  // drop the callable under the GIL
      }
      OnComplete(opr, exc);
    }
  }

  // OnComplete (threaded_engine.cc:441): retire this op's accesses and
  // grant turns to newly unblocked heads
  void OnComplete(Opr* opr, std::exception_ptr exc) {
    std::vector<Opr*> now_ready;
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
        GrantHead(v, &now_ready);
      }
      for (Var* v : opr->mutable_vars) {
        ++v->version;
        for (auto it = v->queue.begin(); it != v->queue.end(); ++it)
          if (it->opr == opr && it->write) {
            v->queue.erase(it);
            break;
          }
        GrantHead(v, &now_ready);
      }
      --inflight_;
      delete opr;
      for (Opr* o : now_ready) ready_q_.push(o);
    }
    done_cv_.notify_all();
    for (size_t i = 0; i < now_ready.size(); ++i) ready_cv_.notify_one();
  }

  // grant consecutive head reads, or the head write once readers drain
  void GrantHead(Var* v, std::vector<Opr*>* now_ready) {
    for (auto it = v->queue.begin(); it != v->queue.end() && !it->write; ++it) {
      if (!it->granted) {
        it->granted = true;
        ++v->running_reads;
        if (--it->opr->wait == 0) now_ready->push_back(it->opr);
      }
    }
    if (!v->queue.empty() && v->queue.front().write &&
        v->running_reads == 0 && !v->queue.front().granted) {
      v->queue.front().granted = true;
      if (--v->queue.front().opr->wait == 0)
        now_ready->push_back(v->queue.front().opr);
    }
  }

  std::mutex mu_;
  std::condition_variable ready_cv_, done_cv_;
  std::queue<Opr*> ready_q_;
  std::unordered_map<int64_t, std::unique_ptr<Var>> vars_;
  std::vector<std::thread> workers_;
  int64_t next_var_id_ = 1;
  int inflight_ = 0;
  bool shutdown_ = false;
  std::exception_ptr global_exc_;
};

ThreadedEngine* GetEngine() {
  static ThreadedEngine* engine = [] {
    const char* env = std::getenv("MXNET_CPU_WORKER_NTHREADS");
    int n = env ? std::atoi(env) : 4;
    return new ThreadedEngine(n > 0 ? n : 4);
  }();
  return engine;
}

}  // namespace

PYBIND11_MODULE(_engine, m) {
  m.doc() = "mxnet_amd host-side ThreadedEngine (C++ dependency scheduler)";
  py::class_<ThreadedEngine>(m, "Engine")
      .def("new_variable", &ThreadedEngine::NewVariable)
      .def("delete_variable", &ThreadedEngine::DeleteVariable)
      .def("version", &ThreadedEngine::Version)
      .def(
          "push",
          [](ThreadedEngine& e, py::object fn, py::object const_vars,
             py::object mutable_vars) {
            std::vector<int64_t> cv, mv;
            for (auto h : const_vars) cv.push_back(h.cast<int64_t>());
            for (auto h : mutable_vars) mv.push_back(h.cast<int64_t>());
            {
              py::gil_scoped_release rel;
              e.Push(std::move(fn), cv, mv);
            }
          },
          py::arg("fn"), py::arg("const_vars") = py::tuple(),
          py::arg("mutable_vars") = py::tuple())
      .def("wait_for_var", &ThreadedEngine::WaitForVar,
           py::call_guard<py::gil_scoped_release>())
      .def("wait_for_all", &ThreadedEngine::WaitForAll,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &ThreadedEngine::Stop,
           py::call_guard<py::gil_scoped_release>())
      .def("start", &ThreadedEngine::Start, py::arg("num_workers") = 4);
  m.def("get", &GetEngine, py::return_value_policy::reference);
}
