// Grouped GEMM for MoE experts (K11 in SURVEY.md §2.3).
//
// Reference behavior: TEGroupedMLP (megatron/core/transformer/moe/
// experts.py:182) — one grouped GEMM over variable-size expert token
// batches.  MI355X-native: hipBLASLt's grouped-gemm extension
// (hipblaslt_ext::GroupedGemm -> Tensile MFMA kernels, one launch for all
// experts).  The per-expert m sizes are host-side (the dispatcher's single
// chosen sync point per MoE layer, reference token_dispatcher.py:453-460);
// the Tensile algo is cached per (mode, E, n, k) and revalidated when the
// m-vector changes shape class.
//
// Three entry points (all row-major tensors):
//   grouped_gemm(a [M,k], b [E,n,k], sizes, trans_b=true)  -> c [M,n]
//       c_e = a_e @ b_e^T           (expert fc1/fc2 forward)
//   grouped_gemm(a [M,n], b [E,n,k], sizes, trans_b=false) -> c [M,k]
//       c_e = a_e @ b_e             (dgrad)
//   grouped_gemm_wgrad(dy [M,n], x [M,k], sizes, dw fp32 [E,n,k])
//       dw_e += dy_e^T @ x_e        (wgrad, fp32 accumulate, beta=1)
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt-ext.hpp>

#include <map>
#include <memory>
#include <tuple>
#include <vector>

#define HIPBLASLT_CHECK(expr)                                                     \
  do {                                                                            \
    hipblasStatus_t st = (expr);                                                  \
    TORCH_CHECK(st == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)st, " at " \
                __FILE__ ":", __LINE__);                                          \
  } while (0)

namespace {

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t handle = [] {
    hipblasLtHandle_t h;
    TORCH_CHECK(hipblasLtCreate(&h) == HIPBLAS_STATUS_SUCCESS, "hipblasLtCreate failed");
    return h;
  }();
  return handle;
}

#define HIP_CHECK_OK(cmd) TORCH_CHECK((cmd) == hipSuccess, "hip error at " __FILE__)

void* lt_workspace(size_t bytes) {
  static void* ws = nullptr;
  static size_t cap = 0;
  if (bytes > cap) {
    if (ws) (void)hipFree(ws);
    HIP_CHECK_OK(hipMalloc(&ws, bytes));
    cap = bytes;
  }
  return ws;
}

constexpr size_t kMaxWorkspace = 64l * 1024 * 1024;

struct CachedAlgo {
  hipblaslt_ext::GroupedGemm gg;
  bool has_algo = false;
  hipblasLtMatmulAlgo_t algo;
  CachedAlgo(hipblasOperation_t opA, hipblasOperation_t opB, hipDataType tAB,
             hipDataType tCD, hipblasComputeType_t comp)
      : gg(lt_handle(), opA, opB, tAB, tAB, tCD, tCD, comp) {}
};

// run one grouped problem set through hipblaslt (col-major formulation)
void run_grouped(hipblasOperation_t opA, hipblasOperation_t opB,
                 hipDataType tAB, hipDataType tCD, hipblasComputeType_t comp,
                 std::vector<int64_t>& m, std::vector<int64_t>& n, std::vector<int64_t>& k,
                 std::vector<int64_t>& lda, std::vector<int64_t>& ldb,
                 std::vector<int64_t>& ldc,
                 std::vector<void*>& aptr, std::vector<void*>& bptr, std::vector<void*>& cptr,
                 float beta_val, int cache_key_mode) {
  size_t ng = m.size();
  if (ng == 0) return;
  static std::map<std::tuple<int, int64_t, int64_t, int64_t>, std::unique_ptr<CachedAlgo>> cache;
  auto key = std::make_tuple(cache_key_mode, (int64_t)ng, m[0], k[0]);
  auto it = cache.find(key);
  if (it == cache.end())
    it = cache.emplace(key, std::make_unique<CachedAlgo>(opA, opB, tAB, tCD, comp)).first;
  CachedAlgo& ca = *it->second;

  static float one = 1.0f;
  static float zero = 0.0f;
  std::vector<int64_t> batch(ng, 1);
  // valid batch strides even at batch_count 1 (Tensile validates them)
  std::vector<int64_t> strideA(ng), strideB(ng), strideC(ng);
  for (size_t i = 0; i < ng; ++i) {
    strideA[i] = lda[i] * (opA == HIPBLAS_OP_N ? k[i] : m[i]);
    strideB[i] = ldb[i] * (opB == HIPBLAS_OP_N ? n[i] : k[i]);
    strideC[i] = ldc[i] * n[i];
  }
  std::vector<hipblaslt_ext::GemmEpilogue> epi(ng);
  std::vector<hipblaslt_ext::GemmInputs> inputs(ng);
  for (size_t i = 0; i < ng; ++i) {
    inputs[i].setA(aptr[i]);
    inputs[i].setB(bptr[i]);
    inputs[i].setC(cptr[i]);
    inputs[i].setD(cptr[i]);
    inputs[i].setAlpha(&one);
    inputs[i].setBeta(beta_val == 0.f ? (const void*)&zero : (const void*)&one);
  }
  hipblaslt_ext::GemmProblemType ptype(opA, opB, tAB, tAB, tCD, tCD, comp);
  HIPBLASLT_CHECK(ca.gg.setProblem(m, n, k, batch, lda, ldb, ldc, ldc,
                                   strideA, strideB, strideC, strideC, epi, inputs, ptype));
  // algo discovery per the hipBLASLt grouped-gemm sample flow: enumerate
  // all grouped-gemm algos once, pick the first that supports the problem
  auto find_algo = [&]() {
    std::vector<hipblasLtMatmulHeuristicResult_t> all;
    HIPBLASLT_CHECK(hipblaslt_ext::getAllAlgos(
        lt_handle(), hipblaslt_ext::GemmType::HIPBLASLT_GROUPED_GEMM,
        opA, opB, tAB, tAB, tCD, tCD, comp, all));
    for (auto& r : all) {
      size_t ws = kMaxWorkspace;
      if (ca.gg.isAlgoSupported(r.algo, ws) == HIPBLAS_STATUS_SUCCESS && ws <= kMaxWorkspace) {
        ca.algo = r.algo;
        ca.has_algo = true;
        return;
      }
    }
    TORCH_CHECK(false, "grouped_gemm: no hipblaslt grouped algo supports this problem");
  };
  if (!ca.has_algo) {
    find_algo();
  } else {
    size_t ws_bytes = kMaxWorkspace;
    if (ca.gg.isAlgoSupported(ca.algo, ws_bytes) != HIPBLAS_STATUS_SUCCESS)
      find_algo();  // m-vector changed enough that the cached algo no longer applies
  }
  auto stream = at::cuda::getCurrentHIPStream();
  HIPBLASLT_CHECK(ca.gg.initialize(ca.algo, lt_workspace(kMaxWorkspace), false, stream));
  HIPBLASLT_CHECK(ca.gg.run(stream));
}

}  // namespace

torch::Tensor grouped_gemm(torch::Tensor a, torch::Tensor b,
                           std::vector<int64_t> sizes, bool trans_b) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 2 && a.dtype() == torch::kBFloat16);
  TORCH_CHECK(b.is_cuda() && b.dim() == 3 && b.dtype() == torch::kBFloat16);
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  int64_t E = b.size(0), bn = b.size(1), bk = b.size(2);
  int64_t in_dim = trans_b ? bk : bn;
  int64_t out_dim = trans_b ? bn : bk;
  TORCH_CHECK(a.size(1) == in_dim, "grouped_gemm: inner dim mismatch");
  TORCH_CHECK((int64_t)sizes.size() == E, "grouped_gemm: sizes/expert mismatch");
  auto c = torch::empty({a.size(0), out_dim}, a.options());

  std::vector<int64_t> m, n, k, lda, ldb, ldc;
  std::vector<void*> aptr, bptr, cptr;
  char* abase = (char*)ac.data_ptr();
  char* bbase = (char*)bc.data_ptr();
  char* cbase = (char*)c.data_ptr();
  int64_t row = 0;
  for (int64_t e = 0; e < E; ++e) {
    int64_t me = sizes[e];
    if (me == 0) continue;
    // row-major C[me, out] = A[me, in] @ op(B_e); col-major: D(out x me) =
    // op'(B_e_cm) @ A_cm with A_cm = A^T (in x me, ld=in)
    m.push_back(out_dim);
    n.push_back(me);
    k.push_back(in_dim);
    // b_e row-major [bn, bk] viewed col-major is [bk, bn] with ld=bk
    //   trans_b=true : need B_e^T contribution -> op_a = T on (bk x bn) -> (bn x bk) ✓ (m=bn, k=bk)
    //   trans_b=false: need B_e contribution as (bk x bn)^N ✓ (m=bk, k=bn)
    lda.push_back(bk);
    ldb.push_back(in_dim);
    ldc.push_back(out_dim);
    bptr.push_back(abase + row * in_dim * 2);
    aptr.push_back(bbase + e * bn * bk * 2);
    cptr.push_back(cbase + row * out_dim * 2);
    row += me;
  }
  TORCH_CHECK(row == a.size(0), "grouped_gemm: sizes sum != rows");
  run_grouped(trans_b ? HIPBLAS_OP_T : HIPBLAS_OP_N, HIPBLAS_OP_N,
              HIP_R_16BF, HIP_R_16BF, HIPBLAS_COMPUTE_32F,
              m, n, k, lda, ldb, ldc, aptr, bptr, cptr, 0.f,
              trans_b ? 0 : 1);
  return c;
}

std::string grouped_gemm_probe() {
  // diagnostic: how many grouped algos exist, and why isAlgoSupported fails
  std::string out;
  std::vector<hipblasLtMatmulHeuristicResult_t> all;
  hipblasStatus_t st = hipblaslt_ext::getAllAlgos(
      lt_handle(), hipblaslt_ext::GemmType::HIPBLASLT_GROUPED_GEMM,
      HIPBLAS_OP_T, HIPBLAS_OP_N, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF,
      HIPBLAS_COMPUTE_32F, all);
  out += "getAllAlgos status=" + std::to_string((int)st) + " n=" + std::to_string(all.size());
  hipblaslt_ext::GroupedGemm gg(lt_handle(), HIPBLAS_OP_T, HIPBLAS_OP_N,
                                HIP_R_16BF, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF,
                                HIPBLAS_COMPUTE_32F);
  // tiny fixed problem: 2 groups of C[64,128] = A[64,256] @ B[128,256]^T
  auto opt = torch::TensorOptions().dtype(torch::kBFloat16).device(torch::kCUDA);
  auto A = torch::randn({128, 256}, opt);
  auto B = torch::randn({2, 128, 256}, opt);
  auto C = torch::empty({128, 128}, opt);
  std::vector<int64_t> m(2, 128), n(2, 64), k(2, 256), batch(2, 1);
  std::vector<int64_t> lda(2, 256), ldb(2, 256), ldc(2, 128);
  std::vector<int64_t> sA(2, 256 * 128), sB(2, 256 * 64), sC(2, 128 * 64);
  std::vector<hipblaslt_ext::GemmEpilogue> epi(2);
  std::vector<hipblaslt_ext::GemmInputs> inputs(2);
  static float onef = 1.f, zerof = 0.f;
  for (int i = 0; i < 2; ++i) {
    inputs[i].setA((char*)B.data_ptr() + i * 128 * 256 * 2);
    inputs[i].setB((char*)A.data_ptr() + i * 64 * 256 * 2);
    inputs[i].setC((char*)C.data_ptr() + i * 64 * 128 * 2);
    inputs[i].setD((char*)C.data_ptr() + i * 64 * 128 * 2);
    inputs[i].setAlpha(&onef);
    inputs[i].setBeta(&zerof);
  }
  hipblaslt_ext::GemmProblemType ptype(HIPBLAS_OP_T, HIPBLAS_OP_N, HIP_R_16BF, HIP_R_16BF,
                                       HIP_R_16BF, HIP_R_16BF, HIPBLAS_COMPUTE_32F);
  st = gg.setProblem(m, n, k, batch, lda, ldb, ldc, ldc, sA, sB, sC, sC, epi, inputs, ptype);
  out += " setProblem=" + std::to_string((int)st);
  int ok = 0, first_fail = -999;
  for (size_t i = 0; i < all.size() && i < 200; ++i) {
    size_t ws = kMaxWorkspace;
    hipblasStatus_t s2 = gg.isAlgoSupported(all[i].algo, ws);
    if (s2 == HIPBLAS_STATUS_SUCCESS) ok++;
    else if (first_fail == -999) first_fail = (int)s2;
  }
  out += " supported=" + std::to_string(ok) + " first_fail_status=" + std::to_string(first_fail);
  // also try heuristic on the set problem
  hipblaslt_ext::GemmPreference pref;
  pref.setMaxWorkspaceBytes(kMaxWorkspace);
  std::vector<hipblasLtMatmulHeuristicResult_t> res;
  try {
    st = gg.algoGetHeuristic(4, pref, res);
    out += " heuristic_status=" + std::to_string((int)st) + " nres=" + std::to_string(res.size());
  } catch (const std::exception& e) {
    out += std::string(" heuristic_threw=") + e.what();
  }
  return out;
}

void grouped_gemm_wgrad(torch::Tensor dy, torch::Tensor x,
                        std::vector<int64_t> sizes, torch::Tensor dw) {
  // dw[e] += dy_e^T @ x_e ; dy [M, n] bf16, x [M, k] bf16, dw [E, n, k] fp32
  TORCH_CHECK(dy.is_cuda() && x.is_cuda() && dw.is_cuda());
  TORCH_CHECK(dw.dtype() == torch::kFloat32 && dw.dim() == 3 && dw.is_contiguous());
  auto dyc = dy.contiguous();
  auto xc = x.contiguous();
  int64_t E = dw.size(0), wn = dw.size(1), wk = dw.size(2);
  TORCH_CHECK(dy.size(1) == wn && x.size(1) == wk);
  TORCH_CHECK((int64_t)sizes.size() == E);

  std::vector<int64_t> m, n, k, lda, ldb, ldc;
  std::vector<void*> aptr, bptr, cptr;
  char* dybase = (char*)dyc.data_ptr();
  char* xbase = (char*)xc.data_ptr();
  char* dwbase = (char*)dw.data_ptr();
  int64_t row = 0;
  for (int64_t e = 0; e < E; ++e) {
    int64_t me = sizes[e];
    if (me == 0) continue;
    // row-major dW_e[wn, wk] += dy_e^T x_e ; col-major: dW_cm (wk x wn, ld=wk)
    //   = X_cm (wk x me, ld=wk)^N @ dY_cm (wn x me, ld=wn)^T
    m.push_back(wk);
    n.push_back(wn);
    k.push_back(me);
    lda.push_back(wk);   // A = X_cm
    ldb.push_back(wn);   // B = dY_cm (transposed)
    ldc.push_back(wk);
    aptr.push_back(xbase + row * wk * 2);
    bptr.push_back(dybase + row * wn * 2);
    cptr.push_back(dwbase + e * wn * wk * 4);
    row += me;
  }
  TORCH_CHECK(row == dy.size(0), "grouped_gemm_wgrad: sizes sum != rows");
  run_grouped(HIPBLAS_OP_N, HIPBLAS_OP_T, HIP_R_16BF, HIP_R_32F,
              HIPBLAS_COMPUTE_32F,
              m, n, k, lda, ldb, ldc, aptr, bptr, cptr, 1.f, 2);
}
