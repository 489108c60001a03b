// One-shot symmetric-memory all-reduce over xGMI peer mappings (K16 in
// SURVEY.md §2.3; reference analog: inference/communication/torch_symm_triton
// one-shot allreduce with multimem PTX).
//
// Use case: TP decode steps move tiny activations ([b, h] with b<=64) where
// RCCL's ring latency (~20-30us) dominates; with every peer's buffer
// directly mapped (hipIpc, dmabuf mode), one kernel does
// flag-barrier + sum-from-all-peers + write in a few microseconds, riding
// the fully-connected 7-link xGMI fabric (each load lands on a direct link).
//
// Protocol (vLLM custom-allreduce style, 2 flag phases to make the buffer
// reusable immediately):
//   symmetric region per rank = [ payload | flags[world] ] (flags 64B apart)
//   1. every rank copies its contribution into its own payload
//   2. kernel: block0/thread r writes seq to PEER r's flags[my_rank]
//      (system fence before), then all blocks spin until local flags[0..w)
//      reach seq — every peer's payload is now globally visible
//   3. grid-strided sum across all world payload pointers -> out
// The sequence number increments per call so no flag reset is needed.
//
// Multi-GPU validation requires an 8-GPU node (the driver's scaling tier);
// the single-GPU degenerate path (world=1: plain copy) is exercised by the
// gpu test tier.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

#define MAX_WORLD 8
#define FLAG_STRIDE 16  // uint32 slots 64B apart to avoid same-cacheline spin

struct PeerPtrs {
  void* payload[MAX_WORLD];
  unsigned int* flags[MAX_WORLD];
};

__global__ void symm_barrier_kernel(PeerPtrs p, int world, int rank,
                                    unsigned int seq) {
  // announce: my payload is ready (to every peer's flag row)
  if (blockIdx.x == 0 && threadIdx.x < world) {
    __threadfence_system();
    __atomic_store_n(p.flags[threadIdx.x] + rank * FLAG_STRIDE, seq,
                     __ATOMIC_RELEASE);
  }
  // wait: everyone announced to me.  The spin is bounded (~seconds) so a
  // protocol bug degrades to a wrong answer the tests catch instead of a
  // wedged GPU; production stalls this long are real failures anyway.
  if (threadIdx.x < world) {
    long guard = 0;
    while (__atomic_load_n(p.flags[rank] + threadIdx.x * FLAG_STRIDE,
                           __ATOMIC_ACQUIRE) < seq) {
      if (++guard > (1L << 33)) break;
    }
  }
  __syncthreads();
}

template <typename T>
__global__ void symm_sum_kernel(PeerPtrs p, T* __restrict__ out, long n,
                                int world) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float acc = 0.f;
#pragma unroll
    for (int r = 0; r < MAX_WORLD; ++r) {
      if (r < world) acc += (float)((const T*)p.payload[r])[i];
    }
    out[i] = (T)acc;
  }
}

// ---------------------------------------------------------------------------
// host side
// ---------------------------------------------------------------------------

std::vector<int64_t> symm_ipc_handle(torch::Tensor buf) {
  // returns the hipIpcMemHandle_t bytes of buf's allocation as int64 list
  hipIpcMemHandle_t h;
  auto err = hipIpcGetMemHandle(&h, buf.data_ptr());
  TORCH_CHECK(err == hipSuccess, "hipIpcGetMemHandle: ", hipGetErrorString(err));
  std::vector<int64_t> out(sizeof(h));
  auto* b = reinterpret_cast<const unsigned char*>(&h);
  for (size_t i = 0; i < sizeof(h); ++i) out[i] = b[i];
  return out;
}

int64_t symm_open_handle(std::vector<int64_t> bytes) {
  hipIpcMemHandle_t h;
  TORCH_CHECK(bytes.size() == sizeof(h), "bad handle size");
  auto* b = reinterpret_cast<unsigned char*>(&h);
  for (size_t i = 0; i < sizeof(h); ++i) b[i] = (unsigned char)bytes[i];
  void* ptr = nullptr;
  auto err = hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(err == hipSuccess, "hipIpcOpenMemHandle: ", hipGetErrorString(err));
  return reinterpret_cast<int64_t>(ptr);
}

void symm_close_handle(int64_t ptr) {
  (void)hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr));
}

// peer_ptrs: device pointers to each rank's symmetric region (self included,
// ordered by rank).  Region layout: payload_bytes then MAX_WORLD flag slots.
void symm_allreduce(std::vector<int64_t> peer_ptrs, int64_t payload_bytes,
                    torch::Tensor local_in, torch::Tensor out, int64_t rank,
                    int64_t seq) {
  int world = (int)peer_ptrs.size();
  TORCH_CHECK(world >= 1 && world <= MAX_WORLD, "world out of range");
  TORCH_CHECK(local_in.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(local_in.numel() == out.numel());
  TORCH_CHECK(local_in.numel() * local_in.element_size() <= payload_bytes,
              "payload larger than symmetric region");
  auto stream = at::cuda::getCurrentHIPStream();

  PeerPtrs p{};
  for (int r = 0; r < world; ++r) {
    auto* base = reinterpret_cast<unsigned char*>(peer_ptrs[r]);
    p.payload[r] = base;
    p.flags[r] = reinterpret_cast<unsigned int*>(base + payload_bytes);
  }
  // 1. stage my contribution into my payload
  auto err = hipMemcpyAsync(p.payload[rank], local_in.data_ptr(),
                            local_in.numel() * local_in.element_size(),
                            hipMemcpyDeviceToDevice, stream);
  TORCH_CHECK(err == hipSuccess, hipGetErrorString(err));
  if (world == 1) {
    err = hipMemcpyAsync(out.data_ptr(), p.payload[rank],
                         out.numel() * out.element_size(),
                         hipMemcpyDeviceToDevice, stream);
    TORCH_CHECK(err == hipSuccess, hipGetErrorString(err));
    return;
  }
  // 2. flag barrier   3. fan-in sum
  hipLaunchKernelGGL(symm_barrier_kernel, dim3(1), dim3(64), 0, stream, p,
                     world, (int)rank, (unsigned int)seq);
  long n = local_in.numel();
  int blocks = (int)std::min<long>((n + 255) / 256, 1024);
  if (local_in.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(symm_sum_kernel<__hip_bfloat16>, dim3(blocks), dim3(256),
                       0, stream, p, (__hip_bfloat16*)out.data_ptr(), n, world);
  } else if (local_in.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(symm_sum_kernel<float>, dim3(blocks), dim3(256), 0,
                       stream, p, (float*)out.data_ptr(), n, world);
  } else {
    TORCH_CHECK(false, "symm_allreduce: bf16/fp32 only");
  }
}
