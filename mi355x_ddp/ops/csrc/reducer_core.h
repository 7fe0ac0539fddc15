// ReducerCore: the C++ hot half of the gradient-bucket Reducer.
//
// The Python Reducer (mi355x_ddp/parallel/reducer.py) owns everything that
// runs ONCE — bucket assignment, flat-buffer construction, parameter
// rebinding, wrap-time broadcast. This class owns everything that runs
// EVERY STEP on the GPU path: the autograd post-hooks that fire per
// parameter during backward, the bucket ready-counting, and the in-order
// launch of bucket all-reduces on the comm stream — all without touching
// the GIL (the reference gets this from torch's C++ reducer,
// torch/csrc/distributed/c10d/reducer.cpp).
//
// Hooks are installed directly on each parameter's AccumulateGrad node
// (torch::autograd::impl::grad_accumulator), exactly where stock DDP
// attaches its autograd_hook — so they run inside the autograd engine
// worker thread, after the grad lands, with no Python frame. The Python
// hook path remains for CPU/gloo (tests) and copy-mode transport.
#pragma once
#include <torch/extension.h>

#include <memory>
#include <mutex>
#include <vector>

#include "rccl_comm.h"

namespace torch::autograd {
struct Node;
}

namespace mi355x {

class ReducerCore : public std::enable_shared_from_this<ReducerCore> {
 public:
  // One entry per bucket: the bucket's parameters, the matching grad views
  // into flat_grad (views transport: autograd accumulates straight into
  // the bucket), and the flat gradient buffer the collective reduces.
  // comm may be null (world size 1: hooks only maintain the view
  // rebinding; finalize is a reset).
  ReducerCore(std::vector<std::vector<at::Tensor>> bucket_params,
              std::vector<std::vector<at::Tensor>> bucket_views,
              std::vector<at::Tensor> bucket_flat_grads, RcclComm* comm);
  ~ReducerCore();

  ReducerCore(const ReducerCore&) = delete;
  ReducerCore& operator=(const ReducerCore&) = delete;

  // Install the AccumulateGrad post-hooks. Separate from the constructor
  // because the hooks hold weak_ptrs to this (shared_from_this).
  void attach_hooks();
  // Remove every installed hook (idempotent; also run by the destructor).
  void detach_hooks();

  // Called once per parameter per backward by the hook trampoline.
  void mark_ready(size_t bucket, size_t index);

  // Post-backward: launch straggler buckets (unused params contribute the
  // zeros already in flat_grad), fence compute on the comm stream, reset
  // per-step counters. Mirrors reducer.py Reducer.finalize.
  void finalize();

  void set_skip_comm(bool v) { skip_comm_ = v; }
  bool skip_comm() const { return skip_comm_; }
  int64_t steps() const { return steps_; }
  // True while bucket collectives have been launched on the comm stream
  // but finalize() has not yet fenced the compute stream on them —
  // reading flat_grad in that window is the reducer's one real race
  // (SURVEY §5.2). Consumers (FusedSGD) check this under
  // MI355X_DEBUG_SYNC=1.
  bool unfenced() const { return unfenced_; }

 private:
  struct Bucket {
    std::vector<at::Tensor> params;
    std::vector<at::Tensor> views;
    at::Tensor flat_grad;
    int pending;
    bool ready;
  };

  void launch_ready_locked();

  std::vector<Bucket> buckets_;
  RcclComm* comm_;  // non-owning; the Python adapter keeps it alive
  bool skip_comm_ = false;
  bool unfenced_ = false;
  int next_launch_ = 0;
  int64_t steps_ = 0;
  std::mutex mu_;
  // keep the AccumulateGrad nodes alive (they are weakly held by the
  // variable) and remember the hook keys for detach.
  std::vector<std::pair<std::shared_ptr<torch::autograd::Node>, uintptr_t>>
      hooks_;
};

}  // namespace mi355x
