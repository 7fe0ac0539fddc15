// RCCL-over-xGMI communicator for the mi355x_ddp reducer.
//
// The reference relies on c10d's ProcessGroupNCCL for every collective
// (init sites multigpu.py:20 etc.). Here the framework
// drives RCCL directly: this class owns the ncclComm_t, a dedicated comm
// HIP stream per rank, and the events that order bucket all-reduces against
// the compute stream (SURVEY §3.5 — the DDP backward hot path).
#pragma once
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <torch/extension.h>
#include <string>

namespace mi355x {

class RcclComm {
 public:
  // unique_id: the 128-byte ncclUniqueId serialized by rank 0 and
  // distributed out-of-band (the framework exchanges it over the torchrun
  // env-rendezvous TCP store).
  RcclComm(const std::string& unique_id, int rank, int world, int device);
  ~RcclComm();

  static std::string make_unique_id();

  int rank() const { return rank_; }
  int world() const { return world_; }

  // Bucket all-reduce (sum/world) launched on the dedicated comm stream,
  // ordered after the CURRENT torch stream via event: the collective
  // overlaps with the remaining backward compute.
  void all_reduce_avg(torch::Tensor t);
  // Same, but synchronous with the current stream (used under graph
  // capture fallbacks and for debugging).
  void all_reduce_avg_inline(torch::Tensor t);
  // Rank-root broadcast on the current stream (DDP wrap-time module-state
  // sync; reference call site DDP(...) multigpu.py:36).
  void broadcast(torch::Tensor t, int root);
  // Make the current torch stream wait for every collective issued so far.
  void join_compute();
  // Host-blocking barrier (1-element all-reduce + stream sync).
  void barrier();

 private:
  ncclComm_t comm_{};
  hipStream_t comm_stream_{};
  hipEvent_t ready_ev_{};
  hipEvent_t done_ev_{};
  int rank_, world_, device_;
  bool any_inflight_ = false;
};

}  // namespace mi355x
