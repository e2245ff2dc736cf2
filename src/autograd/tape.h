// Record-replay reverse-mode autodiff.
// Forward op calls push backward closures onto the Tape; Tape::backward()
// replays them in reverse. Parameter gradients land in Param::grad (a view
// into a flat optimizer/DDP bucket once an optimizer is attached) and fire
// a grad-ready hook that feeds the DDP bucketer.
#pragma once

#include <functional>
#include <memory>
#include <vector>

#include "src/core/ops.h"

namespace nz {

struct Param {
  std::string name;
  Tensor data;    // bf16 on HIP, f32 on CPU
  Tensor grad;    // same dtype as data; may be a bucket view
  bool grad_init = false;  // false => next grad write uses beta=0
  std::function<void(Param&)> ready_hook;  // set by Ddp

  int64_t numel() const { return data.numel(); }
  // Returns beta for the next in-place grad write and flips the flag.
  float grad_beta() {
    float b = grad_init ? 1.0f : 0.0f;
    grad_init = true;
    return b;
  }
  void mark_ready() {
    if (ready_hook) ready_hook(*this);
  }
};

// Guard for residual_add's grad aliasing (modules.cc): when one grad
// buffer is aliased into both inputs of a residual, any accumulation into
// the trunk input mutates the shared buffer — the branch input's grad must
// be fully consumed BEFORE that happens. The guard turns a violation of
// that replay-order invariant into a hard error instead of silent grad
// corruption (a third model wired differently would otherwise train wrong).
struct GradAliasGuard {
  bool mutated = false;  // some alias holder accumulated into the buffer
};

struct Var {
  Tensor value;
  Tensor grad;            // lazily allocated, same dtype as value
  bool grad_init = false;
  bool needs_grad = true; // false for the input leaf
  // dgrad-side GELU fusion (modules.cc Linear): the producer of this var's
  // value is a fused-GELU Linear; dgelu_pre_t holds its saved pre-activation.
  // A downstream consumer that can fuse gelu' into its dgrad GEMM epilogue
  // sets dgelu_applied; the producer's backward then skips the separate
  // multiply. grad_writes counts grad_beta() calls so the producer can
  // verify the fused grad was the ONLY contribution.
  Tensor dgelu_pre_t;
  bool dgelu_applied = false;
  int grad_writes = 0;
  std::shared_ptr<GradAliasGuard> alias_guard;  // set by residual_add
  bool alias_trunk = false;  // this var is the residual trunk input

  Tensor& ensure_grad() {
    if (!grad.defined())
      grad = Tensor(value.ctx(), value.dtype(), value.shape());
    return grad;
  }
  float grad_beta() {
    ensure_grad();
    float b = grad_init ? 1.0f : 0.0f;
    grad_init = true;
    ++grad_writes;
    if (b == 1.0f) note_grad_mutation();
    return b;
  }
  // Every in-place accumulation into an existing grad must pass through
  // grad_beta() or call this explicitly.
  void note_grad_mutation() {
    if (alias_guard) alias_guard->mutated = true;
  }
  // Upstream-grad read used by backward closures: fails if this var's grad
  // aliases a residual buffer that has already been mutated by the trunk.
  Tensor& grad_for_read() {
    NZ_CHECK(grad.defined(), "backward without upstream grad");
    NZ_CHECK(!alias_guard || alias_trunk || !alias_guard->mutated,
             "residual grad alias read after trunk mutation (replay-order "
             "invariant violated — model wiring incompatible with aliasing)");
    return grad;
  }
};
using VarPtr = std::shared_ptr<Var>;

inline VarPtr make_var(Tensor t, bool needs_grad = true) {
  auto v = std::make_shared<Var>();
  v->value = std::move(t);
  v->needs_grad = needs_grad;
  return v;
}

class Tape {
 public:
  void push(std::function<void()> fn) { nodes_.push_back(std::move(fn)); }
  // Replays nodes newest-first, then clears the tape.
  void backward() {
    for (auto it = nodes_.rbegin(); it != nodes_.rend(); ++it) (*it)();
    nodes_.clear();
  }
  void clear() { nodes_.clear(); }
  size_t size() const { return nodes_.size(); }

 private:
  std::vector<std::function<void()>> nodes_;
};

}  // namespace nz
