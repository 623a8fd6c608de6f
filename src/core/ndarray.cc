#include "ndarray.h"

#include <cstring>

namespace mxcore {

void CopyFromTo(const NDArray& src, const NDArray& dst, int /*priority*/) {
  MX_CHECK(src.size() == dst.size(), "CopyFromTo size mismatch");
  MX_CHECK(dtype_size(src.dtype()) == dtype_size(dst.dtype()),
           "CopyFromTo dtype size mismatch (cast with the cast op)");
  size_t nbytes = (size_t)src.size() * dtype_size(src.dtype());
  Context sctx = src.ctx(), dctx = dst.ctx();
  Engine* eng = Engine::Get();

  // capture chunks so the buffers outlive python-side frees
  auto schunk = src.chunk_;
  auto dchunk = dst.chunk_;
  void* sp = src.dptr();
  void* dp = dst.dptr();

  if (!sctx.is_gpu() && !dctx.is_gpu()) {
    eng->PushAsync(
        [=](const RunContext&) {
          (void)schunk; (void)dchunk;
          std::memcpy(dp, sp, nbytes);
        },
        Context::CPU(), {src.var()}, {dst.var()}, FnProperty::kNormal,
        "CopyCPU");
    return;
  }

  FnProperty prop;
  Context run_ctx;
  hipMemcpyKind kind;
  if (sctx.is_gpu() && dctx.is_gpu()) {
    MX_CHECK(sctx.dev_id == dctx.dev_id, "cross-GPU copy via comm path only");
    prop = FnProperty::kNormal;  // same-stream with compute: ordered for free
    run_ctx = dctx;
    kind = hipMemcpyDeviceToDevice;
  } else if (sctx.is_gpu()) {
    prop = FnProperty::kCopyFromGPU;
    run_ctx = sctx;
    kind = hipMemcpyDeviceToHost;
  } else {
    prop = FnProperty::kCopyToGPU;
    run_ctx = dctx;
    kind = hipMemcpyHostToDevice;
  }
  eng->PushAsync(
      [=](const RunContext& rc) {
        (void)schunk; (void)dchunk;
        MX_HIP_CALL(hipMemcpyAsync(dp, sp, nbytes, kind, rc.stream));
      },
      run_ctx, {src.var()}, {dst.var()}, prop, "Copy");
}

}  // namespace mxcore
