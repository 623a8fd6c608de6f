// Example external operator library (reference example/extensions/
// lib_custom_op): compiles to a standalone .so that MXLoadLib dlopens.
//   g++ -shared -fPIC -I <repo>/include my_relu.cc -o libmy_relu.so
//   mx.library.load('libmy_relu.so'); mx.nd.ops._ninv('my_relu', [x])
#include <cstdint>
#include <cstring>

#include "mxnet_amd/c_api.h"

static int relu_infer(int n_in, const MXTensorView* ins, int64_t* out_shape,
                      int* out_ndim, int* out_dtype) {
  if (n_in != 1) return 1;
  *out_ndim = ins[0].ndim;
  for (int i = 0; i < ins[0].ndim; ++i) out_shape[i] = ins[0].shape[i];
  *out_dtype = ins[0].dtype;
  return 0;
}

static int relu_cpu(int n_in, const MXTensorView* ins, MXTensorView* out,
                    void*) {
  if (ins[0].dtype != 0) return 1;  // float32 only
  int64_t n = 1;
  for (int i = 0; i < ins[0].ndim; ++i) n *= ins[0].shape[i];
  const float* x = (const float*)ins[0].data;
  float* y = (float*)out->data;
  for (int64_t i = 0; i < n; ++i) y[i] = x[i] > 0 ? x[i] : 0;
  return 0;
}

extern "C" int mxnet_amd_lib_init(MXRegisterOpFn reg, void* ctx) {
  MXCustomOpDef def{};
  def.name = "my_relu";
  def.n_in = 1;
  def.infer = relu_infer;
  def.fcompute_cpu = relu_cpu;
  def.fcompute_gpu = nullptr;
  reg(ctx, &def);
  return 0;
}
