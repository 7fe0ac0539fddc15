// Host-side launcher declarations for the MI355X (gfx950) kernel pack.
// Implementations in kernels.hip; bound to Python in bindings.cpp.
#pragma once
#include <torch/extension.h>
#include <cstdint>

#include "p2p_mesh.h"

namespace mi355x {

// y = x @ w^T + b          (reference call site: model(source), single_gpu.py:23)
torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor w,
                         c10::optional<torch::Tensor> bias);

// dw = dy^T @ x, db = sum_i dy  (reference: loss.backward(), single_gpu.py:25)
// dw/db may be preallocated views into a gradient bucket; accumulate=true
// adds into them instead of overwriting.
void linear_bwd_weight(torch::Tensor x, torch::Tensor dy,
                       torch::Tensor dw, torch::Tensor db, bool accumulate);

// dx = dy @ w
torch::Tensor linear_bwd_input(torch::Tensor dy, torch::Tensor w);

// Standalone bf16 MFMA GEMM (v_mfma_f32_16x16x32_bf16, f32 accumulate):
// y = x @ w^T + bias for larger bf16 shapes.
torch::Tensor gemm_bf16(torch::Tensor x, torch::Tensor w,
                        c10::optional<torch::Tensor> bias);

// Cross-entropy with class-probability targets (C may be 1: the reference's
// degenerate Linear(20,1) case, single_gpu.py:24 + utils.py:7).
// Returns (loss[scalar], probs[B,C], tsum[B]).
std::vector<torch::Tensor> ce_fwd(torch::Tensor y, torch::Tensor t);
// gout: optional device f32 scalar (autograd's incoming grad) folded
// into the kernel — saves the separate dy*gout launch per step.
torch::Tensor ce_bwd(torch::Tensor probs, torch::Tensor t,
                     torch::Tensor tsum, double grad_scale,
                     c10::optional<torch::Tensor> gout = c10::nullopt);

// MSE (reference multinode_torchrun.py:46). Returns loss[scalar].
torch::Tensor mse_fwd(torch::Tensor y, torch::Tensor t);
torch::Tensor mse_bwd(torch::Tensor y, torch::Tensor t, double grad_scale,
                      c10::optional<torch::Tensor> gout = c10::nullopt);

// Fused SGD over a flat bucket: p -= lr*g; optionally g = 0 in the same
// kernel (folds reference optimizer.step() + zero_grad, single_gpu.py:22,26).
void sgd_flat(torch::Tensor param_flat, torch::Tensor grad_flat,
              double lr, bool zero_grad);

// Bucket gather/scatter ("flatten/unflatten"): one launch
// moves every listed tensor <-> its segment of the flat bucket.
// plan: int64 CPU tensor [n, 3] rows (src_ptr, bucket_elem_offset, numel)
// prebuilt by the reducer; copied to device once and reused every step.
torch::Tensor build_copy_plan(const std::vector<torch::Tensor>& tensors,
                              const std::vector<int64_t>& offsets,
                              torch::Device device);
void flatten_into(torch::Tensor bucket, torch::Tensor plan,
                  int64_t total_blocks, bool zero_src);
void unflatten_from(torch::Tensor bucket, torch::Tensor plan, int64_t total_blocks);

// Fused toy training step: given X[B,K], T[B,1], flat params (w|b) and flat
// grads (dw|db) for Linear(K,1), computes fwd+MSE-or-CE loss grad+bwd and
// writes gradients into the bucket in ONE kernel launch (the latency-bound
// toy path, SURVEY §7 hard-part 2).
// lr > 0: apply the SGD update in-kernel (world-size-1 single-launch step);
// lr <= 0: write gradients into grad_flat for the all-reduce path.
void toy_fused_fwd_bwd(torch::Tensor x, torch::Tensor t,
                       torch::Tensor param_flat, torch::Tensor grad_flat,
                       torch::Tensor loss_out, bool use_mse,
                       int64_t w_off, int64_t b_off, double lr);

// Epoch shard gather with an in-kernel seeded permutation (see
// kernels.hip): returns (xs [n/world, K], ts [n/world, 1]) for this rank.
std::vector<torch::Tensor> epoch_shard(torch::Tensor X, torch::Tensor Tg,
                                       int64_t seed, int64_t rank,
                                       int64_t world);

// Multi-epoch form: gathers `epochs` consecutive shards (seeds seed0,
// seed0+1, …) in ONE launch; block e is bitwise epoch_shard(seed0+e).
std::vector<torch::Tensor> epoch_shard_multi(torch::Tensor X, torch::Tensor Tg,
                                             int64_t seed0, int64_t epochs,
                                             int64_t rank, int64_t world);

// Multi-step persistent toy trainer (world-1): x is [S*batch, K] of S
// consecutive batches; runs S full fwd+loss+bwd+SGD steps in ONE launch
// with the weights resident in LDS (bitwise-identical per-step arithmetic
// to toy_fused_fwd_bwd with in-kernel SGD). loss_out, when non-empty,
// receives the LAST step's loss.
void toy_multistep(torch::Tensor x, torch::Tensor t, torch::Tensor param_flat,
                   torch::Tensor loss_out, bool use_mse,
                   int64_t w_off, int64_t b_off, double lr, int64_t batch);

// Multi-step trainer with an IN-KERNEL xGMI mesh all-reduce per step
// (world > 1): the deferred-launch engine at any world size, one
// collective exchange per step without leaving the kernel.
void toy_multistep_mesh(torch::Tensor x, torch::Tensor t,
                        torch::Tensor param_flat, torch::Tensor loss_out,
                        bool use_mse, int64_t w_off, int64_t b_off, double lr,
                        int64_t batch, P2pMesh& mesh);

}  // namespace mi355x
