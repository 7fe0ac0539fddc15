// See p2p_mesh.h. Native HIP for gfx950; no CUDA compatibility paths.
#include "p2p_mesh.h"

#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>

#include <cstring>

#define HIP_OK(expr)                                                          \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));      \
  } while (0)

namespace mi355x {

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ~5 s at the 100 MHz constant s_memrealtime clock: generous enough that
// rank skew (first-launch overheads, allocator warmup) can never trip a
// false abort, still far below anything that looks like a hang to a
// watchdog. Steady-state exchanges complete in microseconds.
static constexpr unsigned long long kTimeout = 500ull * 1000 * 1000;

template <typename T>
__device__ __forceinline__ float mesh_ldf(const T* p);
template <>
__device__ __forceinline__ float mesh_ldf<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float mesh_ldf<__hip_bfloat16>(
    const __hip_bfloat16* p) {
  return __bfloat162float(*p);
}
template <typename T>
__device__ __forceinline__ void mesh_stf(T* p, float v);
template <>
__device__ __forceinline__ void mesh_stf<float>(float* p, float v) { *p = v; }
template <>
__device__ __forceinline__ void mesh_stf<__hip_bfloat16>(__hip_bfloat16* p,
                                                         float v) {
  *p = __float2bfloat16(v);
}

template <typename T>
__global__ void __launch_bounds__(64, 1)
k_mesh_allreduce(T* __restrict__ grad, int n, unsigned long long seq,
                 MeshSlot* const* __restrict__ peer_slots,
                 MeshSlot* __restrict__ my_mb, int world, int inv_scale_world,
                 unsigned int* __restrict__ err_host) {
  const int t = threadIdx.x;
  const float g = (t < n) ? mesh_ldf(&grad[t]) : 0.f;

  // scatter my contribution into slot[rank] of every rank's mailbox
  for (int p = 0; p < world; ++p) {
    if (t < n) peer_slots[p]->data[t] = g;
  }
  __threadfence_system();
  if (t == 0) {
    for (int p = 0; p < world; ++p)
      __hip_atomic_store(&peer_slots[p]->seq, seq, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_SYSTEM);
  }

  // bounded wait for every rank's contribution in MY mailbox
  bool timed_out = false;
  if (t < world) {
    const unsigned long long start = __builtin_amdgcn_s_memrealtime();
    while (__hip_atomic_load(&my_mb[t].seq, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_SYSTEM) < seq) {
      if (__builtin_amdgcn_s_memrealtime() - start > kTimeout) {
        timed_out = true;
        break;
      }
      __builtin_amdgcn_s_sleep(8);
    }
  }
  if (__any(timed_out)) {
    if (t == 0) *err_host = 1u;
    return;  // grad left as-is; host check() raises before results are used
  }
  __threadfence_system();  // remote mailbox writes visible to all lanes

  if (t < n) {
    float sum = 0.f;
    for (int p = 0; p < world; ++p) sum += my_mb[p].data[t];
    mesh_stf(&grad[t], sum * __int_as_float(inv_scale_world));
  }
}

P2pMesh::P2pMesh(int rank, int world, int device)
    : rank_(rank), world_(world), device_(device) {
  TORCH_CHECK(world >= 2 && world <= 64, "mesh world must be in [2, 64]");
  HIP_OK(hipSetDevice(device_));
  const size_t bytes = sizeof(MeshSlot) * world_;
  // fine-grained device memory: system-scope atomics over xGMI are only
  // defined on fine-grained allocations
  hipError_t e = hipExtMallocWithFlags(reinterpret_cast<void**>(&my_mb_),
                                       bytes, hipDeviceMallocFinegrained);
  fine_grained_ = (e == hipSuccess);
  if (!fine_grained_) {
    HIP_OK(hipMalloc(reinterpret_cast<void**>(&my_mb_), bytes));
  }
  HIP_OK(hipMemset(my_mb_, 0, bytes));
  HIP_OK(hipMalloc(reinterpret_cast<void**>(&peer_slot_dev_),
                   sizeof(MeshSlot*) * world_));
  HIP_OK(hipHostMalloc(reinterpret_cast<void**>(&err_host_),
                       sizeof(unsigned int)));
  *err_host_ = 0u;
}

P2pMesh::~P2pMesh() {
  for (void* p : mapped_) (void)hipIpcCloseMemHandle(p);
  if (my_mb_) (void)hipFree(my_mb_);
  if (peer_slot_dev_) (void)hipFree(peer_slot_dev_);
  if (err_host_) (void)hipHostFree(err_host_);
}

std::string P2pMesh::handle_bytes() const {
  hipIpcMemHandle_t h;
  HIP_OK(hipIpcGetMemHandle(&h, my_mb_));
  return std::string(reinterpret_cast<const char*>(&h), sizeof(h));
}

void P2pMesh::connect(const std::vector<std::string>& handles) {
  TORCH_CHECK((int)handles.size() == world_, "need one handle per rank");
  std::vector<MeshSlot*> slots(world_);
  for (int p = 0; p < world_; ++p) {
    MeshSlot* base;
    if (p == rank_) {
      base = my_mb_;
    } else {
      TORCH_CHECK(handles[p].size() == sizeof(hipIpcMemHandle_t),
                  "bad ipc handle size");
      hipIpcMemHandle_t h;
      std::memcpy(&h, handles[p].data(), sizeof(h));
      void* ptr = nullptr;
      HIP_OK(hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess));
      mapped_.push_back(ptr);
      base = reinterpret_cast<MeshSlot*>(ptr);
    }
    slots[p] = base + rank_;  // my slot in rank p's mailbox
  }
  HIP_OK(hipMemcpy(peer_slot_dev_, slots.data(),
                   sizeof(MeshSlot*) * world_, hipMemcpyHostToDevice));
}

void P2pMesh::all_reduce_avg_inline(torch::Tensor t) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous(), "mesh: contiguous cuda only");
  TORCH_CHECK(t.numel() <= 64, "mesh all-reduce is the tiny-payload path");
  const int n = (int)t.numel();
  ++seq_;
  const float inv = 1.f / (float)world_;
  int inv_bits;
  std::memcpy(&inv_bits, &inv, sizeof(inv));
  if (t.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(k_mesh_allreduce<float>, dim3(1), dim3(64), 0,
                       cur_stream(), t.data_ptr<float>(), n, seq_,
                       (MeshSlot* const*)peer_slot_dev_, my_mb_, world_,
                       inv_bits, err_host_);
  } else if (t.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(k_mesh_allreduce<__hip_bfloat16>, dim3(1), dim3(64), 0,
                       cur_stream(),
                       reinterpret_cast<__hip_bfloat16*>(t.data_ptr()), n,
                       seq_, (MeshSlot* const*)peer_slot_dev_, my_mb_, world_,
                       inv_bits, err_host_);
  } else {
    TORCH_CHECK(false, "mesh: dtype must be f32 or bf16");
  }
  HIP_OK(hipGetLastError());
}

void P2pMesh::check() const {
  TORCH_CHECK(*err_host_ == 0u,
              "P2pMesh: a mesh all-reduce timed out waiting for a peer — "
              "results since then are invalid; fall back to RCCL");
}

}  // namespace mi355x
