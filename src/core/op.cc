#include "op.h"

#include <algorithm>
#include <map>
#include <mutex>
#include <unordered_set>

namespace mxcore {

// defined in src/ops/elemwise.hip (kernel layer); consulted here so the
// tape only defers leaf-grad writes the multi-copy kernel can service
int multi_copy_mode(int src_dtype, int dst_dtype);

OpRegistry* OpRegistry::Get() {
  static OpRegistry inst;
  return &inst;
}

OpEntry& OpRegistry::Register(const std::string& name) {
  auto* e = new OpEntry();
  e->name = name;
  ops_[name] = e;
  return *e;
}

OpEntry* OpRegistry::Find(const std::string& name) {
  auto it = ops_.find(name);
  return it == ops_.end() ? nullptr : it->second;
}

std::vector<std::string> OpRegistry::List() const {
  std::vector<std::string> out;
  for (auto& kv : ops_) out.push_back(kv.first);
  std::sort(out.begin(), out.end());
  return out;
}

Imperative* Imperative::Get() {
  static Imperative inst;
  return &inst;
}

NDArray Make(const TShape& s, Context ctx, int dtype) {
  return NDArray(s, ctx, dtype);
}
NDArray MakeLike(const NDArray& a) {
  return NDArray(a.shape(), a.ctx(), a.dtype());
}

// Per-device grow-only scratch arena for GPU compute ops (reference
// kTempSpace resource): every user runs on that device's single compute
// worker in stream order, so cursor-reset-per-op reuse is race-free.
// Blocks are never reallocated (pointers handed out stay valid); capacity
// converges to the largest step's scratch footprint.
namespace {
struct ArenaBlock {
  void* ptr;
  size_t cap;
  size_t used;
};
struct Arena {
  std::mutex mu;  // shim path (python thread) vs compute worker
  std::vector<ArenaBlock> blocks;
  size_t total = 0;
  Context ctx;
};
Arena& GetArena(const Context& c, int lane) {
  // lane 0 = engine compute worker, lane 1 = frontend shim (torch stream):
  // separate arenas so the two stream domains never alias scratch
  static Arena arenas[136];
  int idx = c.is_gpu() ? (c.dev_id & 63) : 64 + (c.dev_type & 3);
  idx += lane * 68;
  arenas[idx].ctx = c;
  return arenas[idx];
}
}  // namespace

void ArenaReset(const Context& c, int lane) {
  Arena& a = GetArena(c, lane);
  std::lock_guard<std::mutex> g(a.mu);
  for (auto& b : a.blocks) b.used = 0;
}

void* ArenaAlloc(const Context& c, size_t n, int lane) {
  n = (n + 255) & ~(size_t)255;  // 256-B aligned carve-outs
  Arena& a = GetArena(c, lane);
  std::lock_guard<std::mutex> g(a.mu);
  for (auto& b : a.blocks) {
    if (b.cap - b.used >= n) {
      void* p = (char*)b.ptr + b.used;
      b.used += n;
      return p;
    }
  }
  size_t want = std::max(n, a.total);  // at least double the arena
  auto h = Storage::Get()->Alloc(want, c);
  a.blocks.push_back({h.dptr, want, n});
  a.total += want;
  return h.dptr;
}

void Imperative::PushOp(const OpEntry* op, const NodeAttrs& attrs,
                        const std::vector<NDArray>& inputs,
                        const std::vector<NDArray>& outputs) {
  Context ctx = outputs.empty() ? (inputs.empty() ? Context::CPU()
                                                  : inputs[0].ctx())
                                : outputs[0].ctx();
  const FCompute& fc = ctx.is_gpu() ? op->fcompute_gpu : op->fcompute_cpu;
  MX_CHECK(fc, "op '" << op->name << "' has no " << (ctx.is_gpu() ? "GPU" : "CPU")
                      << " implementation (native extension required)");

  std::vector<TBlob> in_blobs, out_blobs;
  in_blobs.reserve(inputs.size());
  out_blobs.reserve(outputs.size());
  std::vector<VarId> cvars, mvars;
  for (auto& o : outputs) {
    out_blobs.push_back(o.data());
    mvars.push_back(o.var());
  }
  for (size_t i = 0; i < inputs.size(); ++i) {
    const NDArray& a = inputs[i];
    in_blobs.push_back(a.data());
    bool mut = std::find(op->mutate_inputs.begin(), op->mutate_inputs.end(),
                         (int)i) != op->mutate_inputs.end();
    if (mut) {
      auto cit = std::find(cvars.begin(), cvars.end(), a.var());
      if (cit != cvars.end()) cvars.erase(cit);  // write dep supersedes
      if (std::find(mvars.begin(), mvars.end(), a.var()) == mvars.end())
        mvars.push_back(a.var());
      continue;
    }
    // in-place (input aliases an output): keep only the write dep
    if (std::find(mvars.begin(), mvars.end(), a.var()) == mvars.end() &&
        std::find(cvars.begin(), cvars.end(), a.var()) == cvars.end())
      cvars.push_back(a.var());
  }
  // keep chunks alive until the op ran
  std::vector<std::shared_ptr<NDArray::Chunk>> hold;
  for (auto& a : inputs) hold.push_back(a.chunk_);
  for (auto& o : outputs) hold.push_back(o.chunk_);

  NodeAttrs at = attrs;  // by value: lambda outlives the caller
  Engine::Get()->PushAsync(
      [op, at, fc, in_blobs, out_blobs, ctx, hold](const RunContext& rc) {
        OpCtx octx;
        octx.rc = rc;
        std::vector<std::unique_ptr<char[]>> cpu_scratch;
        if (ctx.is_gpu()) {
          // single compute worker per device: cursor reset is race-free
          ArenaReset(ctx);
          octx.workspace = [&](size_t n) { return ArenaAlloc(ctx, n); };
        } else {
          // CPU pool has several workers: per-op heap scratch
          octx.workspace = [&](size_t n) -> void* {
            cpu_scratch.emplace_back(new char[n]);
            return cpu_scratch.back().get();
          };
        }
        fc(at, octx, in_blobs, out_blobs);
      },
      ctx, cvars, mvars, FnProperty::kNormal, op->name.c_str());
}

std::vector<NDArray> Imperative::Run(const OpEntry* op, const NodeAttrs& attrs,
                                     const std::vector<NDArray>& inputs) {
  MX_CHECK(op->infer, "op '" << op->name << "' has no shape inference");
  std::vector<TShape> in_shapes;
  std::vector<int> in_dtypes;
  for (auto& a : inputs) {
    in_shapes.push_back(a.shape());
    in_dtypes.push_back(a.dtype());
  }
  std::vector<TShape> out_shapes;
  std::vector<int> out_dtypes;
  op->infer(attrs, in_shapes, in_dtypes, &out_shapes, &out_dtypes);
  // guard against corrupted shape attrs (e.g. a stringified array
  // leaking into a shape tuple): negative dims or numel overflow would
  // otherwise travel into the allocator/kernels
  for (auto& sh : out_shapes) {
    __int128 numel = 1;
    for (auto d : sh) {
      MX_CHECK(d >= 0 && d <= (int64_t(1) << 40),
               "op '" << op->name << "': inferred dim " << d
                      << " out of range (bad shape attr?)");
      numel *= d;
    }
    MX_CHECK(numel <= (__int128(1) << 44),
             "op '" << op->name << "': inferred tensor too large");
  }
  Context ctx = inputs.empty() ? Context::CPU() : inputs[0].ctx();
  if (attrs.has("__ctx_gpu__")) {  // source ops (zeros/random) carry ctx
    int id = (int)attrs.GetInt("__ctx_gpu__", -1);
    ctx = id >= 0 ? Context::GPU(id) : Context::CPU();
  }
  std::vector<NDArray> outputs;
  for (size_t i = 0; i < out_shapes.size(); ++i)
    outputs.emplace_back(out_shapes[i], ctx, out_dtypes[i]);
  PushOp(op, attrs, inputs, outputs);
  return outputs;
}

void Imperative::RunInto(const OpEntry* op, const NodeAttrs& attrs,
                         const std::vector<NDArray>& inputs,
                         const std::vector<NDArray>& outputs) {
  PushOp(op, attrs, inputs, outputs);
}

std::vector<NDArray> Imperative::Invoke(const OpEntry* op,
                                        const NodeAttrs& attrs,
                                        const std::vector<NDArray>& inputs) {
  auto outputs = Run(op, attrs, inputs);
  if (recording_) {
    // record every op: if a gradient later flows into one that has no
    // fbackward, Backward throws instead of silently detaching
    tape_.push_back(TapeNode{op, attrs, inputs, outputs});
  }
  return outputs;
}

void Imperative::InvokeInto(const OpEntry* op, const NodeAttrs& attrs,
                            const std::vector<NDArray>& inputs,
                            const std::vector<NDArray>& outputs) {
  RunInto(op, attrs, inputs, outputs);
}

void Imperative::MarkVariable(const NDArray& x, const NDArray& grad, int req) {
  leaves_[x.chunk_.get()] = LeafInfo{grad, req};
  leaf_keepalive_[x.chunk_.get()] = x;
}

void Imperative::DropVariable(const NDArray& x) {
  leaves_.erase(x.chunk_.get());
  leaf_keepalive_.erase(x.chunk_.get());
}

void Imperative::ClearTape() { tape_.clear(); }

void Imperative::Backward(const std::vector<NDArray>& ys,
                          const std::vector<NDArray>& y_grads,
                          bool retain_graph) {
  OpEntry* add_into = OpRegistry::Get()->Find("_grad_add");
  OpEntry* ones_op = OpRegistry::Get()->Find("ones_like");
  OpEntry* copy_into_op = OpRegistry::Get()->Find("_copy_into");
  OpEntry* multi_copy_op = OpRegistry::Get()->Find("_multi_copy");
  MX_CHECK(add_into && ones_op && copy_into_op && multi_copy_op,
           "core grad ops missing");
  // leaves whose attached grad buffer already received this sweep's value
  std::unordered_set<NDArray::Chunk*> leaf_written;
  // single-contribution leaf writes are DEFERRED and issued as ONE
  // multi-tensor copy at the end of the sweep (a ~200-param model
  // otherwise pays ~200 tiny copy launches per step); a second
  // contribution arriving later flushes that leaf's copy immediately so
  // the following _grad_add lands on a written buffer
  struct Pending { NDArray src, dst; bool flushed = false; };
  std::vector<Pending> pending;
  std::unordered_map<NDArray::Chunk*, size_t> pending_idx;
  auto deferrable = [](const NDArray& src, const NDArray& dst) {
    return src.ctx() == dst.ctx() &&
           multi_copy_mode(src.dtype(), dst.dtype()) >= 0;
  };

  // forward pass over the tape: which chunks require grad at all
  // (leaves plus anything computed from them) — reference OpReqType
  // kNullOp semantics; skips e.g. the input gradient of the first conv
  std::unordered_set<NDArray::Chunk*> need;
  for (auto& kv : leaves_) need.insert(kv.first);
  for (auto& node : tape_) {
    bool any_in = false;
    for (auto& in : node.inputs)
      if (need.count(in.chunk_.get())) {
        any_in = true;
        break;
      }
    if (any_in)
      for (auto& out : node.outputs) need.insert(out.chunk_.get());
  }

  std::unordered_map<NDArray::Chunk*, NDArray> grads;
  for (size_t i = 0; i < ys.size(); ++i) {
    NDArray g;
    if (i < y_grads.size() && !y_grads[i].is_none()) {
      g = y_grads[i];
    } else {
      g = Run(ones_op, {}, {ys[i]})[0];
    }
    grads[ys[i].chunk_.get()] = g;
  }

  for (auto it = tape_.rbegin(); it != tape_.rend(); ++it) {
    TapeNode& node = *it;
    bool any = false;
    std::vector<NDArray> ograds(node.outputs.size());
    for (size_t i = 0; i < node.outputs.size(); ++i) {
      auto git = grads.find(node.outputs[i].chunk_.get());
      if (git != grads.end()) {
        ograds[i] = git->second;
        any = true;
      }
    }
    if (!any) continue;
    MX_CHECK(node.op->fbackward,
             "op '" << node.op->name
                    << "' is not differentiable but a gradient flows into it");
    node.need_igrad.assign(node.inputs.size(), 0);
    bool any_need = false;
    for (size_t i = 0; i < node.inputs.size(); ++i) {
      node.need_igrad[i] = need.count(node.inputs[i].chunk_.get()) ? 1 : 0;
      any_need |= node.need_igrad[i] != 0;
    }
    if (!any_need) continue;
    // missing head grads stay NONE (= zero): materializing a zeros
    // array per unused output (BN saved stats/mask, conv stats, pool
    // argmax) cost ~220 fills per ResNet step; fbackward impls only
    // read the grads that exist
    std::vector<NDArray> igrads = node.op->fbackward(node, ograds);
    MX_CHECK(igrads.size() == node.inputs.size(),
             "op '" << node.op->name << "' backward returned "
                    << igrads.size() << " grads for " << node.inputs.size()
                    << " inputs");
    for (size_t i = 0; i < igrads.size(); ++i) {
      if (igrads[i].is_none() || !node.need_igrad[i]) continue;
      NDArray::Chunk* key = node.inputs[i].chunk_.get();
      auto lit = leaves_.find(key);
      if (lit != leaves_.end()) {
        // leaf: stream contributions straight into the attached buffer
        // (write req overwrites on the first one; add req accumulates)
        bool first = leaf_written.insert(key).second;
        if (first && lit->second.req != 2) {
          if (deferrable(igrads[i], lit->second.grad)) {
            pending_idx[key] = pending.size();
            pending.push_back({igrads[i], lit->second.grad});
          } else {
            RunInto(copy_into_op, {}, {igrads[i]}, {lit->second.grad});
          }
        } else {
          auto pit = pending_idx.find(key);
          if (pit != pending_idx.end() && !pending[pit->second].flushed) {
            Pending& pd = pending[pit->second];
            RunInto(copy_into_op, {}, {pd.src}, {pd.dst});
            pd.flushed = true;
          }
          RunInto(add_into, {}, {igrads[i]}, {lit->second.grad});
        }
        continue;
      }
      auto git = grads.find(key);
      if (git == grads.end()) {
        grads[key] = igrads[i];
      } else {
        // out-of-place: a stored grad may be a view of another node's
        // ograd (reshape-style backwards), so never mutate it
        OpEntry* add2 = OpRegistry::Get()->Find("elemwise_add");
        grads[key] = Run(add2, {}, {git->second, igrads[i]})[0];
      }
    }
  }

  // write accumulated grads into attached leaf buffers
  for (auto& kv : leaves_) {
    auto git = grads.find(kv.first);
    if (git == grads.end()) continue;
    if (kv.second.req == 2) {  // add
      RunInto(add_into, {}, {git->second}, {kv.second.grad});
    } else if (deferrable(git->second, kv.second.grad)) {
      pending.push_back({git->second, kv.second.grad});
    } else {
      RunInto(copy_into_op, {}, {git->second}, {kv.second.grad});
    }
  }
  // flush the deferred leaf writes: one multi-tensor copy per device
  std::map<std::pair<int, int>, std::vector<size_t>> by_dev;
  for (size_t i = 0; i < pending.size(); ++i) {
    if (pending[i].flushed) continue;
    Context c = pending[i].dst.ctx();
    by_dev[{c.dev_type, c.dev_id}].push_back(i);
  }
  for (auto& kv : by_dev) {
    std::vector<NDArray> srcs, dsts;
    for (size_t i : kv.second) {
      srcs.push_back(pending[i].src);
      dsts.push_back(pending[i].dst);
    }
    RunInto(multi_copy_op, {}, srcs, dsts);
  }
  if (!retain_graph) tape_.clear();
}

}  // namespace mxcore
