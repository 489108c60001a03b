// Weight-grad GEMM with fp32 accumulation directly into main_grad (K9):
//   main_grad(fp32)[N, M] += grad_output(bf16)[T, N]^T @ input(bf16)[T, M]
// One rocBLAS gemm_ex call (Tensile MFMA kernels) with fp32 C/D and beta=1 —
// no intermediate bf16 wgrad tensor, no separate add pass.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <rocblas/rocblas.h>
#include "common.h"

static rocblas_handle get_rocblas_handle() {
  static rocblas_handle handle = nullptr;
  if (!handle) {
    if (rocblas_create_handle(&handle) != rocblas_status_success)
      throw std::runtime_error("rocblas_create_handle failed");
  }
  return handle;
}

void wgrad_gemm_accum(torch::Tensor main_grad, torch::Tensor grad_output, torch::Tensor input) {
  TORCH_CHECK(main_grad.dtype() == torch::kFloat32 && main_grad.is_contiguous());
  TORCH_CHECK(grad_output.dtype() == torch::kBFloat16 && input.dtype() == torch::kBFloat16);
  long T = grad_output.size(0);
  long N = grad_output.size(1);
  long M = input.size(1);
  TORCH_CHECK(input.size(0) == T && main_grad.size(0) == N && main_grad.size(1) == M);

  auto handle = get_rocblas_handle();
  auto stream = at::cuda::getCurrentHIPStream();
  rocblas_set_stream(handle, stream);
  float alpha = 1.f, beta = 1.f;
  // column-major view: D[M, N] = A(M x T, =input^T, opN) * B(T x N, =grad_output^T^T, opT)
  auto st = rocblas_gemm_ex(
      handle, rocblas_operation_none, rocblas_operation_transpose,
      (rocblas_int)M, (rocblas_int)N, (rocblas_int)T, &alpha,
      input.data_ptr(), rocblas_datatype_bf16_r, (rocblas_int)M,
      grad_output.data_ptr(), rocblas_datatype_bf16_r, (rocblas_int)N, &beta,
      main_grad.data_ptr(), rocblas_datatype_f32_r, (rocblas_int)M,
      main_grad.data_ptr(), rocblas_datatype_f32_r, (rocblas_int)M,
      rocblas_datatype_f32_r, rocblas_gemm_algo_standard, 0, 0);
  TORCH_CHECK(st == rocblas_status_success, "rocblas_gemm_ex failed: ", (int)st);
}
