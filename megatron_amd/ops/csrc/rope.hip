// RoPE apply forward/backward (K7): rotate-half convention,
// x [s, b, h, d], cos/sin [s, d_rot/2] fp32 host-precomputed tables
// (on-device trig would turn this memory-bound op VALU-bound — guide App-B).
// One wave handles one (s,b,h) row; lane i rotates pair (i, i+d_rot/2)
// for i < d_rot/2 in steps of 64 lanes; passthrough tail copied.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

template <bool BWD>
__global__ void rope_kernel(const bf16* __restrict__ x, const float* __restrict__ cost,
                            const float* __restrict__ sint, bf16* __restrict__ out,
                            long rows, int bh /* b*h */, int d, int half /* d_rot/2 */) {
  // rows = s*b*h; row -> s index = row / bh
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int waves_per_block = blockDim.x / WAVE;
  for (long row = (long)blockIdx.x * waves_per_block + wid; row < rows;
       row += (long)gridDim.x * waves_per_block) {
    long s = row / bh;
    const bf16* xr = x + row * d;
    bf16* orow = out + row * d;
    const float* c = cost + s * half;
    const float* sn = sint + s * half;
    for (int i = lane; i < half; i += WAVE) {
      float x1 = bf2f(xr[i]);
      float x2 = bf2f(xr[i + half]);
      float cv = c[i], sv = BWD ? -sn[i] : sn[i];
      orow[i] = f2bf(x1 * cv - x2 * sv);
      orow[i + half] = f2bf(x2 * cv + x1 * sv);
    }
    for (int i = 2 * half + lane; i < d; i += WAVE) orow[i] = xr[i];
  }
}

static torch::Tensor rope_run(torch::Tensor x, torch::Tensor cost, torch::Tensor sint, bool bwd) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 4);
  auto xc = x.contiguous();
  auto out = torch::empty_like(xc);
  long s = x.size(0), b = x.size(1), h = x.size(2);
  int d = x.size(3);
  int half = cost.size(1);
  long rows = s * b * h;
  int block = 256;
  long waves_needed = rows;
  int grid = (int)std::min<long>((waves_needed + 3) / 4, 2048);
  auto stream = at::cuda::getCurrentHIPStream();
  if (bwd)
    hipLaunchKernelGGL((rope_kernel<true>), dim3(grid), dim3(block), 0, stream,
                       (const bf16*)xc.data_ptr(), cost.data_ptr<float>(), sint.data_ptr<float>(),
                       (bf16*)out.data_ptr(), rows, (int)(b * h), d, half);
  else
    hipLaunchKernelGGL((rope_kernel<false>), dim3(grid), dim3(block), 0, stream,
                       (const bf16*)xc.data_ptr(), cost.data_ptr<float>(), sint.data_ptr<float>(),
                       (bf16*)out.data_ptr(), rows, (int)(b * h), d, half);
  return out;
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cost, torch::Tensor sint) {
  return rope_run(x, cost, sint, false);
}
torch::Tensor rope_bwd(torch::Tensor dy, torch::Tensor cost, torch::Tensor sint) {
  return rope_run(dy, cost, sint, true);
}
