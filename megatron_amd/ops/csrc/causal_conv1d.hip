// Fused depthwise causal conv1d + SiLU for Mamba (K14 in SURVEY.md §2.3).
//
// Reference consumes the external causal_conv1d CUDA package
// (ssm/mamba_mixer.py:63-89).  MI355X-native: channel-LAST layout
// [b, l, C] (the mixer's natural activation layout — no transposes), so a
// thread owning 8 consecutive channels at one (b, l) does K coalesced
// short8 loads; weights/bias for the thread's channels are tiny L2-hot
// scalar loads.  Forward saves the pre-activation for an exact SiLU
// backward.  Backward: one kernel computes dx (K taps of dy*silu'(pre))
// and block-reduces dw/db in LDS before one global atomic per block.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

__device__ __forceinline__ float silu_f(float x) {
  return x / (1.f + __expf(-x));
}
__device__ __forceinline__ float silu_grad_f(float x) {
  float s = 1.f / (1.f + __expf(-x));
  return s * (1.f + x * (1.f - s));
}

template <int K>
__global__ void causal_conv1d_fwd_kernel(
    const short8* __restrict__ x, const bf16* __restrict__ w, const bf16* __restrict__ bias,
    short8* __restrict__ out, short8* __restrict__ pre_save,
    int B, int L, int C8) {
  const long total = (long)B * L * C8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c8 = (int)(i % C8);
    long bl = i / C8;
    int l = (int)(bl % L);
    long b = bl / L;
    float acc[8];
    float wv[K][8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int c = c8 * 8 + j;
      acc[j] = bf2f(bias[c]);
#pragma unroll
      for (int k = 0; k < K; ++k) wv[k][j] = bf2f(w[(long)c * K + k]);
    }
#pragma unroll
    for (int k = 0; k < K; ++k) {
      int src = l - (K - 1) + k;
      if (src < 0) continue;
      short8 v = x[(b * L + src) * C8 + c8];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] = __builtin_fmaf(sbf2f(v[j]), wv[k][j], acc[j]);
    }
    short8 o, p;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      p[j] = f2sbf(acc[j]);
      o[j] = f2sbf(silu_f(acc[j]));
    }
    out[i] = o;
    pre_save[i] = p;
  }
}

// dx + dw/db.  Grid: (bl-chunks, C8); all threads of a block share channels
// c8*8..+7 so dw/db reduce in LDS, one atomicAdd per block per element.
template <int K, int BLOCK>
__global__ void causal_conv1d_bwd_kernel(
    const short8* __restrict__ dy, const short8* __restrict__ x,
    const short8* __restrict__ pre, const bf16* __restrict__ w,
    short8* __restrict__ dx, float* __restrict__ dw, float* __restrict__ db,
    int B, int L) {
  const int c8 = blockIdx.y;
  const int C8 = gridDim.y;
  __shared__ float s_dw[K][8];
  __shared__ float s_db[8];
  if (threadIdx.x < K * 8) (&s_dw[0][0])[threadIdx.x] = 0.f;
  if (threadIdx.x < 8) s_db[threadIdx.x] = 0.f;
  __syncthreads();

  float wv[K][8];
#pragma unroll
  for (int j = 0; j < 8; ++j)
#pragma unroll
    for (int k = 0; k < K; ++k) wv[k][j] = bf2f(w[(long)(c8 * 8 + j) * K + k]);

  float l_dw[K][8];
  float l_db[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    l_db[j] = 0.f;
#pragma unroll
    for (int k = 0; k < K; ++k) l_dw[k][j] = 0.f;
  }

  const long total_bl = (long)B * L;
  for (long bl = (long)blockIdx.x * BLOCK + threadIdx.x; bl < total_bl;
       bl += (long)gridDim.x * BLOCK) {
    int l = (int)(bl % L);
    long b = bl / L;
    const long base = (b * L) * (long)C8 + c8;

    // dx[l] = sum_k w[k] * dpre[l + (K-1) - k]
    float accdx[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) accdx[j] = 0.f;
#pragma unroll
    for (int k = 0; k < K; ++k) {
      int dst = l + (K - 1) - k;
      if (dst >= L) continue;
      short8 g = dy[base + (long)dst * C8];
      short8 pr = pre[base + (long)dst * C8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float dp = sbf2f(g[j]) * silu_grad_f(sbf2f(pr[j]));
        accdx[j] = __builtin_fmaf(dp, wv[k][j], accdx[j]);
      }
    }
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2sbf(accdx[j]);
    dx[base + (long)l * C8] = o;

    // this position's dpre -> dw / db partials
    short8 g = dy[base + (long)l * C8];
    short8 pr = pre[base + (long)l * C8];
    float dp[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      dp[j] = sbf2f(g[j]) * silu_grad_f(sbf2f(pr[j]));
      l_db[j] += dp[j];
    }
#pragma unroll
    for (int k = 0; k < K; ++k) {
      int src = l - (K - 1) + k;
      if (src < 0) continue;
      short8 v = x[base + (long)src * C8];
#pragma unroll
      for (int j = 0; j < 8; ++j) l_dw[k][j] = __builtin_fmaf(dp[j], sbf2f(v[j]), l_dw[k][j]);
    }
  }

#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&s_db[j], l_db[j]);
#pragma unroll
    for (int k = 0; k < K; ++k) atomicAdd(&s_dw[k][j], l_dw[k][j]);
  }
  __syncthreads();
  if (threadIdx.x < 8) {
    int c = c8 * 8 + threadIdx.x;
    atomicAdd(&db[c], s_db[threadIdx.x]);
#pragma unroll
    for (int k = 0; k < K; ++k) atomicAdd(&dw[(long)c * K + k], s_dw[k][threadIdx.x]);
  }
}

std::vector<torch::Tensor> causal_conv1d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 3, "x must be [b,l,C] bf16");
  TORCH_CHECK(x.size(2) % 8 == 0, "conv channels must be a multiple of 8");
  int B = x.size(0), L = x.size(1), C = x.size(2);
  int K = w.size(-1);
  TORCH_CHECK(K == 4, "kernel compiled for d_conv=4");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  auto bc = bias.contiguous();
  auto out = torch::empty_like(xc);
  auto pre = torch::empty_like(xc);
  long total = (long)B * L * (C / 8);
  int grid = grid_for(total, 256);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((causal_conv1d_fwd_kernel<4>), dim3(grid), dim3(256), 0, stream,
                     (const short8*)xc.data_ptr(), (const bf16*)wc.data_ptr(),
                     (const bf16*)bc.data_ptr(), (short8*)out.data_ptr(),
                     (short8*)pre.data_ptr(), B, L, C / 8);
  return {out, pre};
}

std::vector<torch::Tensor> causal_conv1d_bwd(torch::Tensor dy, torch::Tensor x,
                                             torch::Tensor pre, torch::Tensor w) {
  int B = x.size(0), L = x.size(1), C = x.size(2);
  int K = w.size(-1);
  TORCH_CHECK(K == 4, "kernel compiled for d_conv=4");
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  auto dx = torch::empty_like(xc);
  auto dw = torch::zeros({C, K}, x.options().dtype(torch::kFloat32));
  auto db = torch::zeros({C}, x.options().dtype(torch::kFloat32));
  long total_bl = (long)B * L;
  int gx = (int)std::min<long>((total_bl + 255) / 256, 512);
  dim3 grid(gx, C / 8);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((causal_conv1d_bwd_kernel<4, 256>), grid, dim3(256), 0, stream,
                     (const short8*)dyc.data_ptr(), (const short8*)xc.data_ptr(),
                     (const short8*)pre.contiguous().data_ptr(), (const bf16*)wc.data_ptr(),
                     (short8*)dx.data_ptr(), dw.data_ptr<float>(), db.data_ptr<float>(), B, L);
  return {dx, dw, db};
}
