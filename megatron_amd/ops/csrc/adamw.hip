// Multi-tensor AdamW + L2 norm (K10).
// Chunk table built host-side: each chunk = (tensor_idx, offset); kernel
// grid-strides over chunks.  The distributed optimizer feeds a handful of
// large flat fp32 shards, so chunks are big and launch overhead amortizes.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

constexpr int CHUNK = 1 << 16;  // 64Ki elements per chunk

struct TensorPtrs {
  float* p;
  float* g;
  float* m;
  float* v;
  bf16* mp;  // optional bf16 model param mirror (nullptr if absent)
  long n;
};

__global__ void adamw_kernel(const TensorPtrs* __restrict__ tensors,
                             const int2* __restrict__ chunks, int n_chunks,
                             float lr, float beta1, float beta2, float eps, float wd,
                             float bc1, float bc2) {
  for (int ci = blockIdx.x; ci < n_chunks; ci += gridDim.x) {
    int2 c = chunks[ci];
    TensorPtrs t = tensors[c.x];
    long base = (long)c.y * CHUNK;
    long end = min(base + (long)CHUNK, t.n);
    for (long i = base + threadIdx.x; i < end; i += blockDim.x) {
      float g = t.g[i];
      float p = t.p[i] * (1.f - lr * wd);
      float m = t.m[i] = beta1 * t.m[i] + (1.f - beta1) * g;
      float v = t.v[i] = beta2 * t.v[i] + (1.f - beta2) * g * g;
      float denom = sqrtf(v / bc2) + eps;
      p -= (lr / bc1) * m / denom;
      t.p[i] = p;
      if (t.mp) t.mp[i] = f2bf(p);
    }
  }
}

__global__ void l2norm_kernel(const float* const* __restrict__ ptrs, const long* __restrict__ lens,
                              const int2* __restrict__ chunks, int n_chunks,
                              float* __restrict__ out_sq) {
  __shared__ float lds[256 / WAVE];
  float acc = 0.f;
  for (int ci = blockIdx.x; ci < n_chunks; ci += gridDim.x) {
    int2 c = chunks[ci];
    const float* p = ptrs[c.x];
    long base = (long)c.y * CHUNK;
    long end = min(base + (long)CHUNK, lens[c.x]);
    for (long i = base + threadIdx.x; i < end; i += blockDim.x) acc += p[i] * p[i];
  }
  float s = block_reduce_sum<256>(acc, lds);
  if (threadIdx.x == 0) atomicAdd(out_sq, s);
}

// host-side chunk-table builder: returns (meta_tensor_device, chunks_device, n_chunks)
static std::pair<torch::Tensor, int> build_chunks(const std::vector<long>& lens, torch::Device dev) {
  std::vector<int2> chunks;
  for (size_t ti = 0; ti < lens.size(); ++ti) {
    int nc = (int)((lens[ti] + CHUNK - 1) / CHUNK);
    for (int c = 0; c < nc; ++c) chunks.push_back({(int)ti, c});
  }
  auto t = torch::empty({(long)chunks.size() * 2}, torch::TensorOptions().dtype(torch::kInt32));
  memcpy(t.data_ptr(), chunks.data(), chunks.size() * sizeof(int2));
  return {t.to(dev, /*non_blocking=*/true), (int)chunks.size()};
}

void multi_tensor_adamw(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                        std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
                        std::vector<torch::Tensor> model_params,
                        double lr, double beta1, double beta2, double eps, double wd, long step) {
  int n = params.size();
  if (n == 0) return;
  bool has_mp = !model_params.empty();
  std::vector<TensorPtrs> meta(n);
  std::vector<long> lens(n);
  for (int i = 0; i < n; ++i) {
    meta[i] = {params[i].data_ptr<float>(), grads[i].data_ptr<float>(),
               exp_avgs[i].data_ptr<float>(), exp_avg_sqs[i].data_ptr<float>(),
               has_mp ? (bf16*)model_params[i].data_ptr() : nullptr,
               params[i].numel()};
    lens[i] = params[i].numel();
  }
  auto meta_cpu = torch::empty({(long)(n * sizeof(TensorPtrs))},
                               torch::TensorOptions().dtype(torch::kUInt8));
  memcpy(meta_cpu.data_ptr(), meta.data(), n * sizeof(TensorPtrs));
  auto meta_dev = meta_cpu.to(params[0].device(), true);
  auto [chunks_dev, n_chunks] = build_chunks(lens, params[0].device());
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  int grid = std::min(n_chunks, 2048);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(adamw_kernel, dim3(grid), dim3(256), 0, stream,
                     (const TensorPtrs*)meta_dev.data_ptr(), (const int2*)chunks_dev.data_ptr(),
                     n_chunks, (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd, bc1, bc2);
}

torch::Tensor multi_tensor_l2norm(std::vector<torch::Tensor> tensors) {
  int n = tensors.size();
  auto dev = tensors[0].device();
  auto out = torch::zeros({}, torch::TensorOptions().dtype(torch::kFloat32).device(dev));
  std::vector<const float*> ptrs(n);
  std::vector<long> lens(n);
  std::vector<torch::Tensor> keep;
  for (int i = 0; i < n; ++i) {
    auto t = tensors[i].dtype() == torch::kFloat32 ? tensors[i] : tensors[i].to(torch::kFloat32);
    keep.push_back(t);
    ptrs[i] = t.data_ptr<float>();
    lens[i] = t.numel();
  }
  auto ptr_cpu = torch::empty({(long)(n * sizeof(void*))}, torch::TensorOptions().dtype(torch::kUInt8));
  memcpy(ptr_cpu.data_ptr(), ptrs.data(), n * sizeof(void*));
  auto len_cpu = torch::from_blob(lens.data(), {n}, torch::TensorOptions().dtype(torch::kInt64)).clone();
  auto ptr_dev = ptr_cpu.to(dev, true);
  auto len_dev = len_cpu.to(dev, true);
  auto [chunks_dev, n_chunks] = build_chunks(lens, dev);
  int grid = std::min(n_chunks, 2048);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(l2norm_kernel, dim3(grid), dim3(256), 0, stream,
                     (const float* const*)ptr_dev.data_ptr(), len_dev.data_ptr<long>(),
                     (const int2*)chunks_dev.data_ptr(), n_chunks, out.data_ptr<float>());
  return out.sqrt();
}
