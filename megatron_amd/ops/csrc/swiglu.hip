// Fused SwiGLU forward/backward (K5): out = silu(x1) * x2, x = [N, 2F].
// Memory-bound elementwise; bf16x8 vector loads (guide G13), grid-stride.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

__device__ __forceinline__ float silu(float x) { return x / (1.f + __expf(-x)); }

__global__ void swiglu_fwd_kernel(const short8* __restrict__ x, short8* __restrict__ out,
                                  long rows, int f8 /* F/8 */) {
  long total = rows * f8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    long row = idx / f8;
    int col = idx % f8;
    short8 a = x[row * 2 * f8 + col];
    short8 b = x[row * 2 * f8 + f8 + col];
    short8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2sbf(silu(sbf2f(a[j])) * sbf2f(b[j]));
    out[idx] = o;
  }
}

__global__ void swiglu_bwd_kernel(const short8* __restrict__ dy, const short8* __restrict__ x,
                                  short8* __restrict__ dx, long rows, int f8) {
  long total = rows * f8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    long row = idx / f8;
    int col = idx % f8;
    short8 a = x[row * 2 * f8 + col];
    short8 b = x[row * 2 * f8 + f8 + col];
    short8 g = dy[idx];
    short8 d1, d2;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x1 = sbf2f(a[j]), x2 = sbf2f(b[j]), dyf = sbf2f(g[j]);
      float sig = 1.f / (1.f + __expf(-x1));
      float si = x1 * sig;
      d1[j] = f2sbf(dyf * x2 * (sig * (1.f + x1 * (1.f - sig))));
      d2[j] = f2sbf(dyf * si);
    }
    dx[row * 2 * f8 + col] = d1;
    dx[row * 2 * f8 + f8 + col] = d2;
  }
}

torch::Tensor swiglu_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2);
  TORCH_CHECK(x.size(1) % 16 == 0);
  auto xc = x.contiguous();
  long rows = x.size(0);
  int F = x.size(1) / 2;
  auto out = torch::empty({rows, F}, x.options());
  int f8 = F / 8;
  int block = 256;
  int grid = grid_for(rows * f8, block);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(block), 0, stream,
                     (const short8*)xc.data_ptr(), (short8*)out.data_ptr(), rows, f8);
  return out;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor x) {
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  long rows = x.size(0);
  int f8 = (int)(x.size(1) / 2) / 8;
  auto dx = torch::empty_like(xc);
  int block = 256;
  int grid = grid_for(rows * f8, block);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(block), 0, stream,
                     (const short8*)dyc.data_ptr(), (const short8*)xc.data_ptr(),
                     (short8*)dx.data_ptr(), rows, f8);
  return dx;
}
