// mxnet_amd native core — basic types shared by storage/engine/ndarray.
//
// Reference parity: include/mxnet/base.h (Context), mshadow dtype enum
// (include/mxnet/tensor_blob.h); re-designed MI355X-first: HIP is the only
// device runtime, contexts are {cpu, gpu(i), cpu_pinned}.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstring>
#include <sstream>
#include <stdexcept>
#include <string>
#include <vector>

namespace mxcore {

// ---------------------------------------------------------------------------
// errors
// ---------------------------------------------------------------------------
#define MX_CHECK(cond, msg)                                          \
  do {                                                               \
    if (!(cond)) {                                                   \
      std::ostringstream os_;                                        \
      os_ << "Check failed: " #cond ": " << msg << " (" << __FILE__  \
          << ":" << __LINE__ << ")";                                 \
      throw std::runtime_error(os_.str());                           \
    }                                                                \
  } while (0)

#define MX_HIP_CALL(expr)                                             \
  do {                                                                \
    hipError_t e_ = (expr);                                           \
    if (e_ != hipSuccess) {                                           \
      std::ostringstream os_;                                         \
      os_ << "HIP error: " << hipGetErrorString(e_) << " at " #expr   \
          << " (" << __FILE__ << ":" << __LINE__ << ")";              \
      throw std::runtime_error(os_.str());                            \
    }                                                                 \
  } while (0)

// ---------------------------------------------------------------------------
// Context (reference include/mxnet/base.h:147: dev_type i32, dev_id i32)
// ---------------------------------------------------------------------------
struct Context {
  // serialized values match the reference .params Context block
  enum DevType : int { kCPU = 1, kGPU = 2, kCPUPinned = 3 };
  int dev_type = kCPU;
  int dev_id = 0;

  bool is_gpu() const { return dev_type == kGPU; }
  bool operator==(const Context& o) const {
    return dev_type == o.dev_type && dev_id == o.dev_id;
  }
  std::string str() const {
    const char* names[] = {"", "cpu", "gpu", "cpu_pinned"};
    return std::string(names[dev_type]) + "(" + std::to_string(dev_id) + ")";
  }
  static Context CPU() { return {kCPU, 0}; }
  static Context GPU(int id) { return {kGPU, id}; }
  static Context Pinned() { return {kCPUPinned, 0}; }
};

// ---------------------------------------------------------------------------
// dtype — mshadow type_flag parity (the .params on-disk enum)
// ---------------------------------------------------------------------------
enum DTypeFlag : int {
  kFloat32 = 0,
  kFloat64 = 1,
  kFloat16 = 2,
  kUint8 = 3,
  kInt32 = 4,
  kInt8 = 5,
  kInt64 = 6,
  kBool = 7,
  kInt16 = 8,
  kUint16 = 9,
  kUint32 = 10,
  kUint64 = 11,
  kBFloat16 = 12,  // matches python base.py TYPE_FLAG (reference bfloat16)
};

inline size_t dtype_size(int flag) {
  switch (flag) {
    case kFloat64:
    case kInt64:
    case kUint64:
      return 8;
    case kFloat32:
    case kInt32:
    case kUint32:
      return 4;
    case kFloat16:
    case kBFloat16:
    case kInt16:
    case kUint16:
      return 2;
    default:
      return 1;
  }
}

inline const char* dtype_name(int flag) {
  switch (flag) {
    case kFloat32: return "float32";
    case kFloat64: return "float64";
    case kFloat16: return "float16";
    case kBFloat16: return "bfloat16";
    case kUint8: return "uint8";
    case kInt8: return "int8";
    case kInt16: return "int16";
    case kInt32: return "int32";
    case kInt64: return "int64";
    case kBool: return "bool";
    default: return "unknown";
  }
}

// ---------------------------------------------------------------------------
// shape
// ---------------------------------------------------------------------------
using TShape = std::vector<int64_t>;

inline int64_t shape_size(const TShape& s) {
  int64_t n = 1;
  for (int64_t d : s) n *= d;
  return n;
}

// dense blob view handed to kernels: raw pointer + shape + dtype
struct TBlob {
  void* dptr = nullptr;
  TShape shape;
  int dtype = kFloat32;
  int64_t size() const { return shape_size(shape); }
  int ndim() const { return (int)shape.size(); }
};

inline int env_int(const char* name, int dflt) {
  const char* e = getenv(name);
  return e ? atoi(e) : dflt;
}

}  // namespace mxcore
