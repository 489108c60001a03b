// RMSNorm forward + backward (K3 in SURVEY.md §2.3).
// bf16 activations, bf16 weight, fp32 accumulation. One workgroup per row
// (fwd); grid-strided rows with LDS-accumulated dw (bwd).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

// ---------------------------------------------------------------------------
// forward: out[r, :] = x[r, :] * rsqrt(mean(x^2) + eps) * w ; saves rstd[r]
// ---------------------------------------------------------------------------

template <int BLOCK>
__global__ void rmsnorm_fwd_kernel(
    const short8* __restrict__ x, const short8* __restrict__ w,
    short8* __restrict__ out, float* __restrict__ rstd,
    int rows, int h8 /* hidden/8 */, float eps) {
  __shared__ float lds[BLOCK / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short8* xr = x + (long)row * h8;
    short8* or_ = out + (long)row * h8;
    float sumsq = 0.f;
    for (int i = threadIdx.x; i < h8; i += BLOCK) {
      short8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = sbf2f(v[j]);
        sumsq += f * f;
      }
    }
    float total = block_reduce_sum<BLOCK>(sumsq, lds);
    float rs = rsqrtf(total / (float)(h8 * 8) + eps);
    if (threadIdx.x == 0) rstd[row] = rs;
    for (int i = threadIdx.x; i < h8; i += BLOCK) {
      short8 v = xr[i];
      short8 wv = w[i];
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = f2sbf(sbf2f(v[j]) * rs * sbf2f(wv[j]));
      or_[i] = o;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// backward:
//   dx = rstd * dy*w - x * rstd^3 / H * sum(dy*w*x)
//   dw = sum_rows dy * (x * rstd)
// Each block owns a row range; dw accumulated in LDS fp32 then one atomic
// pass (grid is capped so contention is ~n_blocks per element).
// ---------------------------------------------------------------------------

template <int BLOCK>
__global__ void rmsnorm_bwd_kernel(
    const short8* __restrict__ dy, const short8* __restrict__ x,
    const short8* __restrict__ w, const float* __restrict__ rstd,
    short8* __restrict__ dx, float* __restrict__ dw,
    int rows, int h8) {
  extern __shared__ float smem[];           // [h8*8] fp32 dw accum
  __shared__ float lds[BLOCK / WAVE];
  float* dw_acc = smem;
  const int H = h8 * 8;
  for (int i = threadIdx.x; i < H; i += BLOCK) dw_acc[i] = 0.f;
  __syncthreads();

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short8* dyr = dy + (long)row * h8;
    const short8* xr = x + (long)row * h8;
    short8* dxr = dx + (long)row * h8;
    float rs = rstd[row];
    // pass 1: c = sum(dy*w*x)
    float c = 0.f;
    for (int i = threadIdx.x; i < h8; i += BLOCK) {
      short8 dyv = dyr[i], xv = xr[i], wv = w[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) c += sbf2f(dyv[j]) * sbf2f(wv[j]) * sbf2f(xv[j]);
    }
    c = block_reduce_sum<BLOCK>(c, lds);
    float k = rs * rs * rs * c / (float)H;
    for (int i = threadIdx.x; i < h8; i += BLOCK) {
      short8 dyv = dyr[i], xv = xr[i], wv = w[i];
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float dyf = sbf2f(dyv[j]), xf = sbf2f(xv[j]), wf = sbf2f(wv[j]);
        o[j] = f2sbf(rs * dyf * wf - xf * k);
        dw_acc[i * 8 + j] += dyf * xf * rs;
      }
      dxr[i] = o;
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    if (dw_acc[i] != 0.f) atomicAdd(&dw[i], dw_acc[i]);
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2);
  TORCH_CHECK(w.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.size(1) % 8 == 0, "hidden must be a multiple of 8");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  auto out = torch::empty_like(xc);
  auto rstd = torch::empty({x.size(0)}, x.options().dtype(torch::kFloat32));
  int rows = x.size(0), h8 = x.size(1) / 8;
  constexpr int BLOCK = 256;
  int grid = std::min(rows, 2048);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((rmsnorm_fwd_kernel<BLOCK>), dim3(grid), dim3(BLOCK), 0, stream,
                     (const short8*)xc.data_ptr(), (const short8*)wc.data_ptr(),
                     (short8*)out.data_ptr(), rstd.data_ptr<float>(), rows, h8, (float)eps);
  return {out, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w, torch::Tensor rstd) {
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  int rows = x.size(0), h8 = x.size(1) / 8;
  auto dx = torch::empty_like(xc);
  auto dw32 = torch::zeros({x.size(1)}, x.options().dtype(torch::kFloat32));
  constexpr int BLOCK = 256;
  int grid = std::min(rows, 512);
  size_t smem = (size_t)h8 * 8 * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((rmsnorm_bwd_kernel<BLOCK>), dim3(grid), dim3(BLOCK), smem, stream,
                     (const short8*)dyc.data_ptr(), (const short8*)xc.data_ptr(),
                     (const short8*)wc.data_ptr(), rstd.data_ptr<float>(),
                     (short8*)dx.data_ptr(), dw32.data_ptr<float>(), rows, h8);
  return {dx, dw32.to(torch::kBFloat16)};
}
