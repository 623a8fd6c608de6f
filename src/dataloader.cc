// Native threaded data pipeline (reference src/io/dataloader.cc
// ThreadedDataLoader + batchify.cc Stack: C++ worker threads assemble
// batches so the python loop never touches per-sample work).
//
// MI355X design: the hot case is array-backed datasets (synthetic /
// pre-decoded tensors).  Workers gather sample rows into batch buffers
// with memcpy while the GIL is RELEASED; a bounded queue hands finished
// batches (as numpy arrays, zero-copy into torch) to the iterator.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <map>
#include <mutex>
#include <thread>
#include <vector>

namespace py = pybind11;

namespace {

struct Batch {
  std::vector<py::array> arrays;  // created with GIL, filled without
  size_t seq = 0;
};

class ThreadedBatcher {
 public:
  // data: list of C-contiguous numpy arrays sharing dim 0 (e.g. X, Y);
  // indices: flat sample order; batch_size rows per batch.
  ThreadedBatcher(std::vector<py::array> data, std::vector<long> indices,
                  long batch_size, int num_workers, bool drop_last)
      : data_(std::move(data)),
        indices_(std::move(indices)),
        bs_(batch_size),
        drop_last_(drop_last) {
    for (auto& a : data_) {
      auto info = a.request();
      row_bytes_.push_back(info.strides[0]);
      base_.push_back(static_cast<const char*>(info.ptr));
    }
    n_batches_ = drop_last_ ? indices_.size() / bs_
                            : (indices_.size() + bs_ - 1) / bs_;
    next_build_ = 0;
    stop_ = false;
    int nw = std::max(1, num_workers);
    for (int i = 0; i < nw; ++i)
      workers_.emplace_back([this] { Worker(); });
  }

  ~ThreadedBatcher() { Shutdown(); }

  void Shutdown() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
    }
    cv_room_.notify_all();
    cv_ready_.notify_all();
    {
      // workers may be acquiring the GIL for allocation: release it
      // while joining or the join deadlocks
      py::gil_scoped_release nogil;
      for (auto& t : workers_)
        if (t.joinable()) t.join();
    }
    workers_.clear();
    ready_.clear();  // GIL held here (py::array dtors)
  }

  size_t num_batches() const { return n_batches_; }

  // next batch in order; throws StopIteration at the end
  py::list Next() {
    if (served_ == n_batches_) throw py::stop_iteration();
    Batch b;
    {
      // workers take the GIL to allocate outputs: drop it while waiting
      py::gil_scoped_release nogil;
      std::unique_lock<std::mutex> lk(mu_);
      // completion order is arbitrary: wait for OUR sequence number
      cv_ready_.wait(lk, [&] { return ready_.count(served_) != 0; });
      auto it = ready_.find(served_);
      b = std::move(it->second);
      ready_.erase(it);
      ++served_;
      lk.unlock();
      cv_room_.notify_all();
    }
    py::list out;
    for (auto& a : b.arrays) out.append(std::move(a));
    return out;
  }

 private:
  void Worker() {
    for (;;) {
      size_t seq;
      {
        std::lock_guard<std::mutex> lk(mu_);
        if (stop_ || next_build_ >= n_batches_) return;
        seq = next_build_++;
      }
      const size_t i0 = seq * bs_;
      const size_t i1 = std::min(indices_.size(), i0 + (size_t)bs_);
      const long rows = (long)(i1 - i0);
      Batch b;
      b.seq = seq;
      std::vector<char*> dst;
      {
        // output allocation needs the GIL; the copy below does not
        py::gil_scoped_acquire gil;
        for (size_t d = 0; d < data_.size(); ++d) {
          auto info = data_[d].request();
          std::vector<py::ssize_t> shape(info.shape.begin(),
                                         info.shape.end());
          shape[0] = rows;
          py::array out(py::dtype(data_[d].dtype()), shape);
          dst.push_back(static_cast<char*>(out.mutable_data()));
          b.arrays.push_back(std::move(out));
        }
      }
      for (size_t d = 0; d < data_.size(); ++d) {
        const size_t rb = row_bytes_[d];
        for (long r = 0; r < rows; ++r)
          std::memcpy(dst[d] + (size_t)r * rb,
                      base_[d] + (size_t)indices_[i0 + r] * rb, rb);
      }
      {
        std::unique_lock<std::mutex> lk(mu_);
        // always admit the batch the consumer is blocked on, or an
        // out-of-order worker set could fill the queue and deadlock
        cv_room_.wait(lk, [&] {
          return stop_ || ready_.size() < kMaxQueued || seq == served_;
        });
        if (stop_) {
          lk.unlock();  // never hold mu_ while waiting for the GIL
          py::gil_scoped_acquire gil;
          b.arrays.clear();
          return;
        }
        ready_.emplace(seq, std::move(b));
      }
      cv_ready_.notify_all();
    }
  }

  static constexpr size_t kMaxQueued = 8;
  std::vector<py::array> data_;
  std::vector<long> indices_;
  std::vector<size_t> row_bytes_;
  std::vector<const char*> base_;
  long bs_;
  bool drop_last_;
  size_t n_batches_ = 0;
  size_t next_build_ = 0;
  size_t served_ = 0;
  bool stop_;
  std::map<size_t, Batch> ready_;
  std::mutex mu_;
  std::condition_variable cv_ready_, cv_room_;
  std::vector<std::thread> workers_;
};

}  // namespace

PYBIND11_MODULE(_dataloader, m) {
  m.doc() = "native threaded batch assembly (reference io/dataloader.cc)";
  py::class_<ThreadedBatcher>(m, "ThreadedBatcher")
      .def(py::init<std::vector<py::array>, std::vector<long>, long, int,
                    bool>(),
           py::arg("data"), py::arg("indices"), py::arg("batch_size"),
           py::arg("num_workers") = 2, py::arg("drop_last") = false)
      .def("next", &ThreadedBatcher::Next)
      .def("shutdown", &ThreadedBatcher::Shutdown)
      .def_property_readonly("num_batches", &ThreadedBatcher::num_batches);
}
