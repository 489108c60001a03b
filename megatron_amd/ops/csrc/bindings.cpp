// Python bindings for the megatron_amd CDNA4 HIP kernels.
#include <torch/extension.h>

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w, torch::Tensor rstd);
torch::Tensor swiglu_fwd(torch::Tensor x);
torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor x);
torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cost, torch::Tensor sint);
torch::Tensor rope_bwd(torch::Tensor dy, torch::Tensor cost, torch::Tensor sint);
void multi_tensor_adamw(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                        std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
                        std::vector<torch::Tensor> model_params,
                        double lr, double beta1, double beta2, double eps, double wd, long step);
torch::Tensor multi_tensor_l2norm(std::vector<torch::Tensor> tensors);
void wgrad_gemm_accum(torch::Tensor main_grad, torch::Tensor grad_output, torch::Tensor input);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    bool causal, double scale, long window);
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor out, torch::Tensor lse,
                                    bool causal, double scale, long window);
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor target, long vocab_start);
void ce_bwd(torch::Tensor logits, torch::Tensor target, torch::Tensor row_max,
            torch::Tensor row_inv_sumexp, torch::Tensor grad_out, long vocab_start);
std::string grouped_gemm_probe();
torch::Tensor grouped_gemm(torch::Tensor a, torch::Tensor b, std::vector<int64_t> sizes, bool trans_b);
void grouped_gemm_wgrad(torch::Tensor dy, torch::Tensor x, std::vector<int64_t> sizes, torch::Tensor dw);
std::vector<torch::Tensor> causal_conv1d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias);
std::vector<torch::Tensor> causal_conv1d_bwd(torch::Tensor dy, torch::Tensor x,
                                             torch::Tensor pre, torch::Tensor w);
std::vector<int64_t> symm_ipc_handle(torch::Tensor buf);
int64_t symm_open_handle(std::vector<int64_t> bytes);
void symm_close_handle(int64_t ptr);
void symm_allreduce(std::vector<int64_t> peer_ptrs, int64_t payload_bytes,
                    torch::Tensor local_in, torch::Tensor out, int64_t rank,
                    int64_t seq);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("rope_bwd", &rope_bwd);
  m.def("multi_tensor_adamw", &multi_tensor_adamw);
  m.def("multi_tensor_l2norm", &multi_tensor_l2norm);
  m.def("wgrad_gemm_accum", &wgrad_gemm_accum);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("grouped_gemm", &grouped_gemm);
  m.def("grouped_gemm_probe", &grouped_gemm_probe);
  m.def("grouped_gemm_wgrad", &grouped_gemm_wgrad);
  m.def("causal_conv1d_fwd", &causal_conv1d_fwd);
  m.def("causal_conv1d_bwd", &causal_conv1d_bwd);
  m.def("symm_ipc_handle", &symm_ipc_handle);
  m.def("symm_open_handle", &symm_open_handle);
  m.def("symm_close_handle", &symm_close_handle);
  m.def("symm_allreduce", &symm_allreduce);
}
