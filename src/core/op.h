// Op registry + imperative invoke + autograd tape.
//
// Reference parity: the nnvm registration model (SURVEY §2.2 — FCompute /
// FInferShape / FGradient attributes per op; src/imperative/imperative.cc:
// Invoke :98, RecordOp :204, Backward :387).  MI355X redesign: FCompute
// receives raw TBlobs + the engine RunContext (device compute stream);
// gradients execute EAGERLY in reverse tape order (no symbolic grad graph:
// the engine's dependency tracking already overlaps/orders everything, and
// CachedOp-style whole-step hipGraph capture recovers static-graph launch
// cost).
#pragma once

#include <functional>
#include <string>
#include <unordered_map>
#include <vector>

#include "ndarray.h"

namespace mxcore {

// ---------------------------------------------------------------------------
// attrs: string dict like the reference's NodeAttrs, with typed accessors
// ---------------------------------------------------------------------------
struct NodeAttrs {
  std::unordered_map<std::string, std::string> d;

  bool has(const std::string& k) const { return d.count(k) != 0; }
  int64_t GetInt(const std::string& k, int64_t dflt) const {
    auto it = d.find(k);
    return it == d.end() ? dflt : strtoll(it->second.c_str(), nullptr, 10);
  }
  double GetFloat(const std::string& k, double dflt) const {
    auto it = d.find(k);
    return it == d.end() ? dflt : strtod(it->second.c_str(), nullptr);
  }
  bool GetBool(const std::string& k, bool dflt) const {
    auto it = d.find(k);
    if (it == d.end()) return dflt;
    return it->second == "1" || it->second == "true" || it->second == "True";
  }
  // "(3, 3)" / "3" python-repr tuples (reference param stringification)
  std::vector<int64_t> GetTuple(const std::string& k,
                                std::vector<int64_t> dflt) const {
    auto it = d.find(k);
    if (it == d.end()) return dflt;
    std::vector<int64_t> out;
    const char* p = it->second.c_str();
    while (*p) {
      if (*p == '-' || (*p >= '0' && *p <= '9')) {
        char* e;
        out.push_back(strtoll(p, &e, 10));
        p = e;
      } else {
        ++p;
      }
    }
    return out;
  }
};

// per-call context handed to FCompute: engine stream + scratch space
struct OpCtx {
  RunContext rc;
  // compute-stream-ordered scratch arena (reference ResourceRequest
  // kTempSpace); valid for this op only
  std::function<void*(size_t)> workspace;
};

using FCompute = std::function<void(const NodeAttrs&, const OpCtx&,
                                    const std::vector<TBlob>& inputs,
                                    const std::vector<TBlob>& outputs)>;
// shape/dtype inference: fills out_shapes/out_dtypes from inputs
using FInferShape = std::function<void(
    const NodeAttrs&, const std::vector<TShape>& in_shapes,
    const std::vector<int>& in_dtypes, std::vector<TShape>* out_shapes,
    std::vector<int>* out_dtypes)>;
// eager gradient: given the recorded node, return one grad per input
// (empty NDArray = not differentiable / no grad)
struct TapeNode;
using FBackward = std::function<std::vector<NDArray>(
    const TapeNode&, const std::vector<NDArray>& out_grads)>;

struct OpEntry {
  std::string name;
  int n_in = -1;   // -1 = variadic
  int n_out = 1;
  // input positions the op WRITES (BatchNorm running stats, optimizer
  // states): tracked as engine mutable vars instead of read deps
  std::vector<int> mutate_inputs;
  FInferShape infer;
  FCompute fcompute_gpu;
  FCompute fcompute_cpu;
  FBackward fbackward;
};

class OpRegistry {
 public:
  static OpRegistry* Get();
  OpEntry& Register(const std::string& name);
  OpEntry* Find(const std::string& name);
  std::vector<std::string> List() const;

 private:
  std::unordered_map<std::string, OpEntry*> ops_;
};

// registration helper:  MXCORE_REGISTER_OP(foo).n_in = 2; ...
#define MXCORE_REGISTER_OP(name)                               \
  static ::mxcore::OpEntry& __mxcore_op_##name##__ =           \
      ::mxcore::OpRegistry::Get()->Register(#name)

struct TapeNode {
  const OpEntry* op;
  NodeAttrs attrs;
  std::vector<NDArray> inputs;
  std::vector<NDArray> outputs;
  // filled by Backward before fbackward runs: whether each input needs a
  // gradient (leaf or derived from one) — impls skip dead grads, e.g.
  // the data gradient of the stem convolution (reference
  // needs_input_grad / OpReqType kNullOp)
  std::vector<char> need_igrad;
};

// ---------------------------------------------------------------------------
// imperative runtime: invoke + record + backward
// ---------------------------------------------------------------------------
class Imperative {
 public:
  static Imperative* Get();

  bool is_recording() const { return recording_; }
  void set_recording(bool r) { recording_ = r; }
  bool is_training() const { return training_; }
  void set_training(bool t) { training_ = t; }

  // allocate outputs (ctx/dtype from inputs + infer), push the engine op,
  // record on the tape when recording
  std::vector<NDArray> Invoke(const OpEntry* op, const NodeAttrs& attrs,
                              const std::vector<NDArray>& inputs);
  // write into caller-provided outputs (optimizer updates, accumulation);
  // never recorded
  void InvokeInto(const OpEntry* op, const NodeAttrs& attrs,
                  const std::vector<NDArray>& inputs,
                  const std::vector<NDArray>& outputs);

  // autograd leaves: x's gradient accumulates into grad (req: 1=write 2=add)
  void MarkVariable(const NDArray& x, const NDArray& grad, int req);
  void DropVariable(const NDArray& x);

  // eager reverse-mode sweep over the tape; clears the tape afterwards
  void Backward(const std::vector<NDArray>& ys,
                const std::vector<NDArray>& y_grads, bool retain_graph);
  void ClearTape();
  size_t TapeSize() const { return tape_.size(); }

  // low-level: used by FBackward impls to run ops without re-recording
  static std::vector<NDArray> Run(const OpEntry* op, const NodeAttrs& attrs,
                                  const std::vector<NDArray>& inputs);
  static void RunInto(const OpEntry* op, const NodeAttrs& attrs,
                      const std::vector<NDArray>& inputs,
                      const std::vector<NDArray>& outputs);

 private:
  static void PushOp(const OpEntry* op, const NodeAttrs& attrs,
                     const std::vector<NDArray>& inputs,
                     const std::vector<NDArray>& outputs);

  bool recording_ = false;
  bool training_ = false;
  std::vector<TapeNode> tape_;
  struct LeafInfo {
    NDArray grad;
    int req;
  };
  std::unordered_map<NDArray::Chunk*, LeafInfo> leaves_;
  std::unordered_map<NDArray::Chunk*, NDArray> leaf_keepalive_;
};

// helpers shared by op implementations
NDArray MakeLike(const NDArray& a);
NDArray Make(const TShape& s, Context ctx, int dtype);

// scratch arenas (see op.cc): lane 0 = engine compute worker,
// lane 1 = frontend shim path on the torch stream
void ArenaReset(const Context& c, int lane = 0);
void* ArenaAlloc(const Context& c, size_t n, int lane = 0);

}  // namespace mxcore
