// Device-side mesh all-reduce for tiny gradient payloads over xGMI.
//
// The toy DDP step's collective is 84 B — pure latency. A
// host-initiated ring/tree collective pays host launch + protocol setup
// per step. This mesh does ONE kernel: every rank stores its contribution
// directly into a slot of every peer's mailbox (peer-mapped via HIP IPC,
// point-to-point over xGMI), publishes a monotonically increasing sequence
// number with a system-scope release store, spin-waits (bounded) for all
// peers' sequence numbers, and reduces locally in a fixed rank order — so
// every rank computes the bitwise-identical average.
//
// Safety: the poll has a wall-clock timeout; on expiry the kernel sets a
// host-visible error flag and exits, and the host layer raises so callers
// fall back to RCCL (never a hang). Correctness against a second
// transport is checked at setup time by the Python adapter
// (mi355x_ddp/parallel/comm.py P2pMeshComm.validate).
#pragma once
#include <torch/extension.h>

#include <hip/hip_runtime.h>

#include <string>
#include <vector>

namespace mi355x {

struct MeshSlot {
  float data[64];
  unsigned long long seq;
  unsigned long long pad[7];  // 576 B/slot, keeps seq on its own 64B line
};

class P2pMesh {
 public:
  P2pMesh(int rank, int world, int device);
  ~P2pMesh();
  P2pMesh(const P2pMesh&) = delete;

  // 64-byte opaque hipIpcMemHandle_t for THIS rank's mailbox.
  std::string handle_bytes() const;
  // peers' handles, rank-ordered (entry [rank_] is ignored).
  void connect(const std::vector<std::string>& handles);

  // In-place average over the mesh on the current stream. t: contiguous
  // f32/bf16 cuda tensor, numel <= 64. Deterministic: fixed rank-order sum.
  void all_reduce_avg_inline(torch::Tensor t);

  // Throws if any prior mesh kernel timed out (checked host-side; call
  // after a stream sync).
  void check() const;

  int rank() const { return rank_; }
  int world() const { return world_; }

  // device-view accessors for kernels that embed the mesh exchange
  // (toy_multistep_mesh in kernels.hip)
  MeshSlot* my_mb() const { return my_mb_; }
  MeshSlot* const* peer_slots() const { return peer_slot_dev_; }
  unsigned int* err_flag() const { return err_host_; }
  // reserve n sequence numbers; returns the first. Every rank must make
  // identical reservation sequences (they do: same engine code path).
  unsigned long long alloc_seq(unsigned long long n) {
    unsigned long long first = seq_ + 1;
    seq_ += n;
    return first;
  }

 private:
  int rank_, world_, device_;
  MeshSlot* my_mb_ = nullptr;          // device, fine-grained, world slots
  MeshSlot** peer_slot_dev_ = nullptr; // device array: &mb_p[rank_] per p
  std::vector<void*> mapped_;          // opened IPC ptrs (to close)
  unsigned int* err_host_ = nullptr;   // pinned host flag
  unsigned long long seq_ = 0;
  bool fine_grained_ = false;
};

}  // namespace mi355x
