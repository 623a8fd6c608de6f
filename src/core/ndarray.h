// NDArray — ref-counted chunk (Storage handle + engine Var) with views.
//
// Reference parity: include/mxnet/ndarray.h:82 (NDArray), :851-1122 (Chunk,
// byte_offset_ views, delayed alloc), src/ndarray/ndarray.cc CopyFromTo.
// Dense only: the sparse formats live in the Python layer for now.
#pragma once

#include <memory>

#include "engine.h"
#include "storage.h"

namespace mxcore {

class NDArray {
 public:
  NDArray() = default;
  NDArray(const TShape& shape, Context ctx, int dtype, bool delay_alloc = false)
      : shape_(shape), dtype_(dtype),
        chunk_(std::make_shared<Chunk>(
            (size_t)shape_size(shape) * dtype_size(dtype), ctx, delay_alloc)) {}

  bool is_none() const { return chunk_ == nullptr; }
  const TShape& shape() const { return shape_; }
  int dtype() const { return dtype_; }
  Context ctx() const { return chunk_ ? chunk_->ctx : Context::CPU(); }
  VarId var() const { return chunk_ ? chunk_->var : 0; }
  int64_t size() const { return shape_size(shape_); }
  size_t byte_offset() const { return byte_offset_; }

  void* dptr() const {
    chunk_->EnsureAlloc();
    return (char*)chunk_->shandle.dptr + byte_offset_;
  }

  TBlob data() const { return TBlob{dptr(), shape_, dtype_}; }

  // zero-copy view with a new shape (same chunk, same var)
  NDArray Reshape(const TShape& s) const {
    MX_CHECK(shape_size(s) == size(), "reshape size mismatch");
    NDArray r = *this;
    r.shape_ = s;
    return r;
  }

  // zero-copy slice along axis 0: rows [begin, end)
  NDArray Slice(int64_t begin, int64_t end) const {
    MX_CHECK(!shape_.empty() && begin >= 0 && end <= shape_[0] && begin < end,
             "bad slice [" << begin << "," << end << ")");
    NDArray r = *this;
    r.shape_[0] = end - begin;
    int64_t inner = size() / shape_[0];
    r.byte_offset_ = byte_offset_ + (size_t)begin * inner * dtype_size(dtype_);
    return r;
  }

  // reinterpret dtype (sizes must match elementwise)
  NDArray AsType(int dtype) const {
    MX_CHECK(dtype_size(dtype) == dtype_size(dtype_), "astype view mismatch");
    NDArray r = *this;
    r.dtype_ = dtype;
    return r;
  }

  void WaitToRead() const {
    if (chunk_) {
      Engine::Get()->WaitForVar(chunk_->var);
    }
  }

  struct Chunk {
    Storage::Handle shandle;
    VarId var = 0;
    Context ctx;
    size_t nbytes = 0;
    bool allocated = false;
    std::mutex alloc_mu;

    Chunk(size_t nbytes_, Context c, bool delay) : ctx(c), nbytes(nbytes_) {
      var = Engine::Get()->NewVariable();
      if (!delay) EnsureAlloc();
    }
    void EnsureAlloc() {
      if (allocated) return;
      std::lock_guard<std::mutex> g(alloc_mu);
      if (allocated) return;
      shandle = Storage::Get()->Alloc(nbytes ? nbytes : 1, ctx);
      allocated = true;
    }
    ~Chunk() {
      Storage::Handle h = shandle;
      bool was_alloc = allocated;
      Engine::Get()->PushDeleteVariable(var, [h, was_alloc](const RunContext&) {
        if (was_alloc) Storage::Get()->Free(h);
      });
    }
  };

  std::shared_ptr<Chunk> chunk_;

 private:
  TShape shape_;
  int dtype_ = kFloat32;
  size_t byte_offset_ = 0;
};

// async engine-sequenced copy (reference ndarray.cc CopyFromTo):
// routes H2D/D2H to the copy workers, D2D to compute
void CopyFromTo(const NDArray& src, const NDArray& dst, int priority = 0);

}  // namespace mxcore
