// See reducer_core.h. Host-side C++ (no device code): autograd hook
#include <cstring>
#include <cstdlib>
// trampoline + bucket launch logic for the GPU reducer path.
#include "reducer_core.h"

#include <torch/csrc/autograd/function.h>
#include <torch/csrc/autograd/function_hook.h>
#include <torch/csrc/autograd/variable.h>

namespace mi355x {

namespace {

// Fires after a parameter's AccumulateGrad node writes the gradient.
// Runs on the autograd engine's device worker thread WITHOUT the GIL —
// ready-counting and the bucket all-reduce launch never enter Python.
struct CoreHook : torch::autograd::FunctionPostHook {
  std::weak_ptr<ReducerCore> core;
  size_t bucket, index;

  CoreHook(std::weak_ptr<ReducerCore> c, size_t b, size_t i)
      : core(std::move(c)), bucket(b), index(i) {}

  torch::autograd::variable_list operator()(
      const torch::autograd::variable_list& outputs,
      const torch::autograd::variable_list& /*inputs*/) override {
    // MI355X_CORE_DEBUG: bisection aid (noop skips the whole body) — used
    // to isolate the capture-time segfault investigation; not a prod knob.
    static const char* dbg = getenv("MI355X_CORE_DEBUG");
    if (dbg && strcmp(dbg, "noop") == 0) return outputs;
    if (auto c = core.lock()) {  // expired core: reducer was dropped — no-op
      c->mark_ready(bucket, index);
    }
    return outputs;
  }
};

}  // namespace

ReducerCore::ReducerCore(std::vector<std::vector<at::Tensor>> bucket_params,
                         std::vector<std::vector<at::Tensor>> bucket_views,
                         std::vector<at::Tensor> bucket_flat_grads,
                         RcclComm* comm)
    : comm_(comm) {
  TORCH_CHECK(bucket_params.size() == bucket_views.size() &&
                  bucket_params.size() == bucket_flat_grads.size(),
              "ReducerCore: bucket list sizes disagree");
  buckets_.reserve(bucket_params.size());
  for (size_t b = 0; b < bucket_params.size(); ++b) {
    TORCH_CHECK(bucket_params[b].size() == bucket_views[b].size(),
                "ReducerCore: bucket ", b, " params/views disagree");
    TORCH_CHECK(bucket_flat_grads[b].is_cuda(),
                "ReducerCore is the GPU path; bucket ", b, " is not on GPU");
    for (size_t i = 0; i < bucket_params[b].size(); ++i) {
      TORCH_CHECK(bucket_params[b][i].requires_grad(),
                  "ReducerCore: bucket ", b, " param ", i,
                  " does not require grad");
      TORCH_CHECK(bucket_views[b][i].numel() == bucket_params[b][i].numel(),
                  "ReducerCore: view/param numel mismatch at ", b, "/", i);
    }
    Bucket bk;
    bk.params = std::move(bucket_params[b]);
    bk.views = std::move(bucket_views[b]);
    bk.flat_grad = std::move(bucket_flat_grads[b]);
    bk.pending = (int)bk.params.size();
    bk.ready = false;
    buckets_.push_back(std::move(bk));
  }
}

ReducerCore::~ReducerCore() { detach_hooks(); }

void ReducerCore::attach_hooks() {
  TORCH_CHECK(hooks_.empty(), "ReducerCore hooks already attached");
  auto self = weak_from_this();
  TORCH_CHECK(!self.expired(),
              "ReducerCore must be held by shared_ptr before attach_hooks");
  for (size_t b = 0; b < buckets_.size(); ++b) {
    for (size_t i = 0; i < buckets_[b].params.size(); ++i) {
      // grad_accumulator() creates the AccumulateGrad node if the param
      // doesn't have one yet; the node is weakly held by the variable, so
      // WE keep the shared_ptr (as stock DDP's reducer does).
      auto node = torch::autograd::impl::grad_accumulator(buckets_[b].params[i]);
      TORCH_CHECK(node, "no grad accumulator for bucket ", b, " param ", i);
      uintptr_t key =
          node->add_post_hook(std::make_unique<CoreHook>(self, b, i));
      hooks_.emplace_back(std::move(node), key);
    }
  }
}

void ReducerCore::detach_hooks() {
  for (auto& [node, key] : hooks_) {
    node->del_post_hook(key);
  }
  hooks_.clear();
}

void ReducerCore::mark_ready(size_t bucket, size_t index) {
  static const char* dbg = getenv("MI355X_CORE_DEBUG");
  Bucket& bk = buckets_[bucket];
  const at::Tensor& p = bk.params[index];
  const at::Tensor& view = bk.views[index];
  if (dbg && strcmp(dbg, "norebind") == 0) {
    std::lock_guard<std::mutex> lk2(mu_);
    bk.pending -= 1;
    if (bk.pending == 0) {
      bk.ready = true;
      launch_ready_locked();
    }
    return;
  }
  // Views transport: autograd normally accumulates IN PLACE into our view
  // (p.grad was bound to it at construction). If the engine replaced the
  // grad tensor (first iteration after p.grad=None, or an out-of-place
  // accumulation), fold the fresh grad into the bucket and re-bind so the
  // next backward accumulates in place again. Mirrors reducer.py hook().
  auto& g = p.mutable_grad();
  if (g.defined() && g.data_ptr() != view.data_ptr()) {
    view.add_(g);
    g = view;
  }
  std::lock_guard<std::mutex> lk(mu_);
  bk.pending -= 1;
  if (bk.pending == 0) {
    bk.ready = true;
    launch_ready_locked();
  }
}

void ReducerCore::launch_ready_locked() {
  // Collectives must be issued in the same bucket order on every rank:
  // launch strictly in index order, skipping ahead only over ready ones.
  while (next_launch_ < (int)buckets_.size() &&
         buckets_[next_launch_].ready) {
    if (comm_ != nullptr && !skip_comm_) {
      comm_->all_reduce_avg(buckets_[next_launch_].flat_grad);
      unfenced_ = true;
    }
    ++next_launch_;
  }
}

void ReducerCore::finalize() {
  std::lock_guard<std::mutex> lk(mu_);
  for (; next_launch_ < (int)buckets_.size(); ++next_launch_) {
    // stragglers: params that never produced a grad left their segments
    // zeroed (the fused SGD zeroes flat_grad every step)
    if (comm_ != nullptr && !skip_comm_) {
      comm_->all_reduce_avg(buckets_[next_launch_].flat_grad);
    }
  }
  if (comm_ != nullptr && !skip_comm_) {
    comm_->join_compute();
  }
  unfenced_ = false;
  for (auto& bk : buckets_) {
    bk.pending = (int)bk.params.size();
    bk.ready = false;
  }
  next_launch_ = 0;
  ++steps_;
}

}  // namespace mi355x
