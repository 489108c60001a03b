// Fused vocab-parallel cross-entropy (K8 in SURVEY.md §2.3).
//
// Reference behavior: megatron/core/fusions/fused_cross_entropy.py:13-65
// (3 jit_fuser stages + 2 TP all-reduces) over tensor_parallel/
// cross_entropy.py:121.  MI355X-first design: ONE online pass over the
// bf16 logits in forward (flash-style running (max, sumexp) merge — no
// fp32 [T, V] materialization, no separate max pass) and ONE pass in
// backward that writes the bf16 grad IN-PLACE over the logits buffer.
// The two tiny [T] all-reduces (row max, corrected sumexp + target pick)
// stay in Python between kernels; the local sumexp is rebased from the
// local max to the global max with one [T] elementwise op, so TP>1 costs
// no extra pass over [T, V].
//
// HBM cost per microbatch: fwd = 1 read of [T,V] bf16; bwd = 1 read +
// 1 write.  (Previous torch-composed path: ~10 passes incl. fp32 temps.)
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

// merge two online-softmax states (m, s): result covers both.
__device__ __forceinline__ void lse_merge(float& m, float& s, float m2, float s2) {
  float nm = fmaxf(m, m2);
  // exp(-inf - -inf) -> NaN guard: if both -inf, s stays 0
  float a = (m == nm) ? s : s * __expf(m - nm);
  float b = (m2 == nm) ? s2 : s2 * __expf(m2 - nm);
  s = a + b;
  m = nm;
}

// forward: one workgroup per row (grid-strided).  Each thread keeps an
// online (max, sumexp) over its strided short8 chunks plus the raw target
// logit if it owns it; wave shfl + LDS merge produce per-row results.
template <int BLOCK>
__global__ void ce_fwd_kernel(
    const short8* __restrict__ logits, const long* __restrict__ target,
    float* __restrict__ row_max, float* __restrict__ row_sumexp,
    float* __restrict__ target_logit,
    int rows, int v8, long vocab_start, long vocab_part) {
  __shared__ float lds_m[BLOCK / WAVE];
  __shared__ float lds_s[BLOCK / WAVE];
  __shared__ float lds_t[BLOCK / WAVE];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  constexpr int NW = BLOCK / WAVE;

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short8* xr = logits + (long)row * v8;
    long mt = target[row] - vocab_start;           // local index or out of range
    bool in_shard = (mt >= 0) && (mt < vocab_part);
    float m = -INFINITY, s = 0.f, tval = 0.f;
    for (int i = threadIdx.x; i < v8; i += BLOCK) {
      short8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = sbf2f(v[j]);
        // branchless online update (2 exp/elem, HBM latency dominates)
        float nm = fmaxf(m, f);
        s = s * __expf(m - nm) + __expf(f - nm);
        m = nm;
        if (in_shard && ((long)i * 8 + j == mt)) tval = f;
      }
    }
    // wave merge
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float m2 = __shfl_xor(m, off, 64);
      float s2 = __shfl_xor(s, off, 64);
      tval += __shfl_xor(tval, off, 64);
      lse_merge(m, s, m2, s2);
    }
    if (lane == 0) { lds_m[wid] = m; lds_s[wid] = s; lds_t[wid] = tval; }
    __syncthreads();
    if (threadIdx.x == 0) {
      float fm = lds_m[0], fs = lds_s[0], ft = lds_t[0];
#pragma unroll
      for (int w = 1; w < NW; ++w) {
        lse_merge(fm, fs, lds_m[w], lds_s[w]);
        ft += lds_t[w];
      }
      row_max[row] = fm;
      row_sumexp[row] = fs;
      target_logit[row] = ft;
    }
    __syncthreads();
  }
}

// backward: grad = (exp(x - m)/s - onehot(target)) * grad_out, written
// in-place over the logits buffer as bf16.
template <int BLOCK>
__global__ void ce_bwd_kernel(
    short8* __restrict__ logits, const long* __restrict__ target,
    const float* __restrict__ row_max, const float* __restrict__ row_inv_sumexp,
    const float* __restrict__ grad_out,
    int rows, int v8, long vocab_start, long vocab_part) {
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    short8* xr = logits + (long)row * v8;
    long mt = target[row] - vocab_start;
    bool in_shard = (mt >= 0) && (mt < vocab_part);
    const float m = row_max[row];
    const float inv_s = row_inv_sumexp[row];
    const float go = grad_out[row];
    for (int i = threadIdx.x; i < v8; i += BLOCK) {
      short8 v = xr[i];
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = sbf2f(v[j]);
        float p = __expf(f - m) * inv_s;
        if (in_shard && ((long)i * 8 + j == mt)) p -= 1.f;
        o[j] = f2sbf(p * go);
      }
      xr[i] = o;
    }
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor target, long vocab_start) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kBFloat16 && logits.dim() == 2,
              "ce_fwd: logits must be 2-D bf16 CUDA");
  TORCH_CHECK(logits.size(1) % 8 == 0, "ce_fwd: vocab shard must be a multiple of 8");
  TORCH_CHECK(target.dtype() == torch::kInt64);
  auto lc = logits.contiguous();
  auto tc = target.contiguous();
  long rows = lc.size(0), vp = lc.size(1);
  auto opts = lc.options().dtype(torch::kFloat32);
  auto row_max = torch::empty({rows}, opts);
  auto row_sumexp = torch::empty({rows}, opts);
  auto target_logit = torch::empty({rows}, opts);
  constexpr int BLOCK = 256;
  int grid = (int)std::min<long>(rows, 4096);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((ce_fwd_kernel<BLOCK>), dim3(grid), dim3(BLOCK), 0, stream,
                     (const short8*)lc.data_ptr(), tc.data_ptr<long>(),
                     row_max.data_ptr<float>(), row_sumexp.data_ptr<float>(),
                     target_logit.data_ptr<float>(),
                     (int)rows, (int)(vp / 8), vocab_start, vp);
  return {row_max, row_sumexp, target_logit};
}

void ce_bwd(torch::Tensor logits, torch::Tensor target, torch::Tensor row_max,
            torch::Tensor row_inv_sumexp, torch::Tensor grad_out, long vocab_start) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kBFloat16 && logits.dim() == 2);
  TORCH_CHECK(logits.is_contiguous(), "ce_bwd writes in-place; logits must be contiguous");
  long rows = logits.size(0), vp = logits.size(1);
  auto tc = target.contiguous();
  auto go = grad_out.contiguous();
  constexpr int BLOCK = 256;
  int grid = (int)std::min<long>(rows, 4096);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((ce_bwd_kernel<BLOCK>), dim3(grid), dim3(BLOCK), 0, stream,
                     (short8*)logits.data_ptr(), tc.data_ptr<long>(),
                     row_max.contiguous().data_ptr<float>(),
                     row_inv_sumexp.contiguous().data_ptr<float>(),
                     go.data_ptr<float>(), (int)rows, (int)(vp / 8), vocab_start, vp);
}
