/*
 * C ABI for the mxnet_amd native runtime.
 *
 * Reference parity: include/mxnet/c_api.h (238 MXNET_DLL entry points,
 * handle-based; MXNDArrayCreate/Save/Load, MXImperativeInvoke,
 * MXAutogradBackwardEx, per-thread MXGetLastError).  This is the
 * MI355X-native equivalent over src/core/: the same handle discipline,
 * scoped to the surface a C/FFI client needs — array lifecycle, imperative
 * invoke through the op registry, autograd, serialization, engine syncs.
 * Exported from the in-tree mxnet_amd/_core*.so.
 */
#ifndef MXNET_AMD_C_API_H_
#define MXNET_AMD_C_API_H_

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef void* NDArrayHandle;

/* every call returns 0 on success, -1 on failure; the message is
 * per-thread (reference c_api_error.h) */
const char* MXGetLastError();

int MXNDArrayCreate(const int64_t* shape, int ndim, int dev_type,
                    int dev_id, int dtype, NDArrayHandle* out);
int MXNDArrayFree(NDArrayHandle h);
int MXNDArrayGetShape(NDArrayHandle h, int* ndim, const int64_t** shape);
int MXNDArrayGetDType(NDArrayHandle h, int* dtype);
int MXNDArrayGetContext(NDArrayHandle h, int* dev_type, int* dev_id);
int MXNDArraySyncCopyFromCPU(NDArrayHandle h, const void* data,
                             size_t nbytes);
int MXNDArraySyncCopyToCPU(NDArrayHandle h, void* data, size_t nbytes);
int MXNDArrayWaitToRead(NDArrayHandle h);
int MXNDArrayWaitAll();

/* imperative invoke through the native op registry: attrs as parallel
 * key/value string arrays; outputs are allocated by the runtime */
int MXImperativeInvoke(const char* op_name, int num_inputs,
                       NDArrayHandle* inputs, int* num_outputs,
                       NDArrayHandle** outputs, int num_attrs,
                       const char** attr_keys, const char** attr_vals);

int MXListOps(int* count, const char*** names);

/* autograd over the native tape */
int MXAutogradSetIsRecording(int recording, int* prev);
int MXAutogradMarkVariables(int num, NDArrayHandle* vars,
                            NDArrayHandle* grads, const int* reqs);
int MXAutogradBackward(int num_heads, NDArrayHandle* heads,
                       NDArrayHandle* head_grads, int retain_graph);

/* External operator libraries (reference include/mxnet/lib_api.h,
 * MXLoadLib): a user .so exports
 *     int mxnet_amd_lib_init(MXRegisterOpFn reg, void* reg_ctx);
 * and calls `reg(reg_ctx, &op)` once per MXCustomOpDef.  Compute
 * callbacks receive raw buffers + shapes and, on GPU, the hipStream_t
 * of the engine's compute stream (NULL on CPU). */
typedef struct {
  int ndim;
  const int64_t* shape;
  int dtype;          /* DTypeFlag: 0=f32 2=f16 6=i64 ... */
  void* data;
} MXTensorView;

typedef int (*MXCustomInferFn)(int n_in, const MXTensorView* ins,
                               int64_t* out_shape, int* out_ndim,
                               int* out_dtype);
typedef int (*MXCustomComputeFn)(int n_in, const MXTensorView* ins,
                                 MXTensorView* out, void* stream);

typedef struct {
  const char* name;
  int n_in;
  MXCustomInferFn infer;          /* single-output shape/dtype */
  MXCustomComputeFn fcompute_cpu; /* either may be NULL */
  MXCustomComputeFn fcompute_gpu;
} MXCustomOpDef;

typedef void (*MXRegisterOpFn)(void* reg_ctx, const MXCustomOpDef* def);

/* dlopen `path` and register every op it defines; ops become invokable
 * through MXImperativeInvoke / the python frontend like built-ins */
int MXLoadLib(const char* path);

/* .params list serialization (reference MXNDArraySave/Load,
 * byte format of SURVEY.md Appendix A) */
int MXNDArraySave(const char* fname, int num, NDArrayHandle* arrays,
                  const char** names);
int MXNDArrayLoad(const char* fname, int* out_count,
                  NDArrayHandle** out_arrays, const char*** out_names);

#ifdef __cplusplus
}
#endif

#endif  /* MXNET_AMD_C_API_H_ */
