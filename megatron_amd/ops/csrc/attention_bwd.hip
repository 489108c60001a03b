// Fused attention BACKWARD for CDNA4 (K1).  Atomic-free two-kernel split
// (the FA2-style trade: P/dS recomputed in both kernels, no dq atomics):
//   bwd_dq :  block per (q-tile, b*hq) — dq = dS K
//   bwd_dkv:  block per (kv-tile, b*hkv), q-heads of the GQA group looped
//             inside — dv = P^T dO ; dk = dS^T Q
// plus a rowsum preprocess  Drow = sum_d(dO * O).
//
// Swapped-operand trick throughout (guide §B): computing the TRANSPOSED
// score tile ST = mfma(K_a, Q_b) lets both operands come from row-major
// loads, since the B-fragment of X^T reads the same per-lane elements as
// the A-fragment of X.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define LOG2E 1.44269504088896340736f

// ---------------------------------------------------------------------------
// Drow[b, h, s] = sum_d dO[s,b,h,:] * O[s,b,h,:]
// ---------------------------------------------------------------------------
__global__ void attn_bwd_rowsum_kernel(const short8* __restrict__ dout, const short8* __restrict__ out,
                                       float* __restrict__ drow, long rows, int bh, int sq, int d8) {
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  int wpb = blockDim.x / 64;
  for (long row = (long)blockIdx.x * wpb + wid; row < rows; row += (long)gridDim.x * wpb) {
    const short8* a = dout + row * d8;
    const short8* b = out + row * d8;
    float acc = 0.f;
    for (int i = lane; i < d8; i += 64) {
      short8 x = a[i], y = b[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += sbf2f(x[j]) * sbf2f(y[j]);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) {
      long s = row / bh, rem = row % bh;  // row = s*bh + (b*h... actually b_then_h)
      drow[rem * sq + s] = acc;           // out layout [b*h, s]
    }
  }
}

// ---------------------------------------------------------------------------
// dq kernel
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void attn_bwd_dq_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    const bf16* __restrict__ dout, const float* __restrict__ lse, const float* __restrict__ drow,
    bf16* __restrict__ dq,
    int sq, int skv, int b, int hq, int hkv, float scale, int causal, int window) {
  constexpr int KVBLK = 64;
  constexpr int RPAD = D + 8;      // row-major row stride
  constexpr int TPAD = KVBLK + 8;  // transposed row stride
  constexpr int ND = D / 16, NKD = D / 32;

  __shared__ short K_lds[KVBLK][RPAD];   // row-major (ST A-operand)
  __shared__ short Kt_lds[D][TPAD];      // transposed (dq B-operand)
  __shared__ short V_lds[KVBLK][RPAD];   // row-major (dPT A-operand)
  __shared__ short dS_lds[4][16][TPAD];  // per-wave dS^T->dS round-trip

  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int q0 = blockIdx.x * 64;
  const int batch = blockIdx.y / hq, head = blockIdx.y % hq;
  const int kv_head = head / (hq / hkv);
  const long q_srow = (long)b * hq * D, k_srow = (long)b * hkv * D;
  const bf16* qbase = q + ((long)batch * hq + head) * D;
  const bf16* dobase = dout + ((long)batch * hq + head) * D;
  const bf16* kbase = k + ((long)batch * hkv + kv_head) * D;
  const bf16* vbase = v + ((long)batch * hkv + kv_head) * D;
  bf16* dqbase = dq + ((long)batch * hq + head) * D;
  const float* lse_row = lse + ((long)batch * hq + head) * sq;
  const float* dr_row = drow + ((long)batch * hq + head) * sq;

  const int row0 = q0 + wid * 16, rg = lane >> 4, cl = lane & 15;
  const int off = skv - sq;

  bf16x8 qa[NKD], doa[NKD];
  {
    int qrow = min(row0 + cl, sq - 1);
    const bf16* qr = qbase + (long)qrow * q_srow;
    const bf16* dor = dobase + (long)qrow * q_srow;
#pragma unroll
    for (int kd = 0; kd < NKD; ++kd) {
      short8 t1 = *reinterpret_cast<const short8*>(qr + kd * 32 + rg * 8);
      qa[kd] = *reinterpret_cast<bf16x8*>(&t1);
      short8 t2 = *reinterpret_cast<const short8*>(dor + kd * 32 + rg * 8);
      doa[kd] = *reinterpret_cast<bf16x8*>(&t2);
    }
  }
  float my_lse = lse_row[min(row0 + cl, sq - 1)];
  float my_dr = dr_row[min(row0 + cl, sq - 1)];

  f32x4 dq_acc[ND];
#pragma unroll
  for (int n = 0; n < ND; ++n) dq_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  int t_end = causal ? min((q0 + 63 + off) / KVBLK, (skv - 1) / KVBLK) : (skv - 1) / KVBLK;
  int t_start = (window > 0) ? max(0, (q0 + off - window + 1) / KVBLK) : 0;

  for (int t = t_start; t <= t_end; ++t) {
    const int k0 = t * KVBLK;
    {
      constexpr int G = KVBLK * D / 8;
      for (int idx = threadIdx.x; idx < G; idx += 256) {
        int r = idx / (D / 8), c8 = idx % (D / 8);
        int krow = k0 + r;
        short8 kv8 = short8{0, 0, 0, 0, 0, 0, 0, 0}, vv8 = kv8;
        if (krow < skv) {
          kv8 = *reinterpret_cast<const short8*>(kbase + (long)krow * k_srow + c8 * 8);
          vv8 = *reinterpret_cast<const short8*>(vbase + (long)krow * k_srow + c8 * 8);
        }
        *reinterpret_cast<short8*>(&K_lds[r][c8 * 8]) = kv8;
        *reinterpret_cast<short8*>(&V_lds[r][c8 * 8]) = vv8;
      }
      // transposed K: r-fast mapping avoids the 8-row-stride bank aliasing
      for (int idx = threadIdx.x; idx < G; idx += 256) {
        int r = idx & (KVBLK - 1), c8 = idx / KVBLK;
        int krow = k0 + r;
        short8 kv8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
        if (krow < skv)
          kv8 = *reinterpret_cast<const short8*>(kbase + (long)krow * k_srow + c8 * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) Kt_lds[c8 * 8 + j][r] = kv8[j];
      }
    }
    __syncthreads();

    // per kv-subtile n: ST = K Q^T ; PT ; dPT = V dO^T ; dST -> dS_lds
#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n) {
      f32x4 st = f32x4{0.f, 0.f, 0.f, 0.f};
      f32x4 dpt = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kd = 0; kd < NKD; ++kd) {
        bf16x8 ak = *reinterpret_cast<const bf16x8*>(&K_lds[n * 16 + cl][kd * 32 + rg * 8]);
        bf16x8 av = *reinterpret_cast<const bf16x8*>(&V_lds[n * 16 + cl][kd * 32 + rg * 8]);
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak, qa[kd], st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, doa[kd], dpt, 0, 0, 0);
      }
      // D-layout: row = kv (rg*4+r), col = q (cl); this wave's q cols are
      // row0+cl, matching the my_lse/my_dr loads above.
      int qrow = row0 + cl;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int kvrow = k0 + n * 16 + rg * 4 + r;
        bool ok = (kvrow < skv) && (qrow < sq);
        if (causal) ok &= (kvrow <= qrow + off);
        if (window > 0) ok &= (kvrow > qrow + off - window);
        // lse/drow are per q-row: q index here is cl, but my_lse was loaded
        // with index row0+cl on THIS lane -> matches col ✓
        float pt = ok ? exp2f((st[r] * scale - my_lse) * LOG2E) : 0.f;
        float dst = pt * (dpt[r] - my_dr) * scale;
        dS_lds[wid][cl][n * 16 + rg * 4 + r] = f2sbf(dst);
      }
    }
    __syncthreads();

    // dq += dS K   (A = dS row-major from dS_lds, B = K^T from Kt_lds)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < KVBLK / 32; ++ks) {
      short8 at = *reinterpret_cast<const short8*>(&dS_lds[wid][cl][ks * 32 + rg * 8]);
      bf16x8 a = *reinterpret_cast<bf16x8*>(&at);
#pragma unroll
      for (int n = 0; n < ND; ++n) {
        bf16x8 bK = *reinterpret_cast<const bf16x8*>(&Kt_lds[n * 16 + cl][ks * 32 + rg * 8]);
        dq_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bK, dq_acc[n], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = row0 + rg * 4 + r;
    if (row >= sq) continue;
    bf16* dqr = dqbase + (long)row * q_srow;
#pragma unroll
    for (int n = 0; n < ND; ++n) dqr[n * 16 + cl] = f2bf(dq_acc[n][r]);
  }
}

// ---------------------------------------------------------------------------
// dk/dv kernel
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(512) void attn_bwd_dkv_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    const bf16* __restrict__ dout, const float* __restrict__ lse, const float* __restrict__ drow,
    bf16* __restrict__ dk, bf16* __restrict__ dv,
    int sq, int skv, int b, int hq, int hkv, float scale, int causal, int window) {
  constexpr int QBLK = 64;
  constexpr int RPAD = D + 8;
  constexpr int TPAD = QBLK + 8;
  constexpr int ND = D / 16, NKD = D / 32;

  __shared__ short BufB[QBLK][RPAD];   // row-major Q (phase 1) then dO (phase 3)
  __shared__ short BufA[D][TPAD];      // transposed dO (phase 2) then Q^T (phase 4)
  __shared__ short P_lds[8][16][TPAD]; // per-wave PT / dST round-trips

  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int k0 = blockIdx.x * 128;  // 8 waves x 16 kv rows
  const int batch = blockIdx.y / hkv, kv_head = blockIdx.y % hkv;
  const int rep = hq / hkv;
  const long q_srow = (long)b * hq * D, k_srow = (long)b * hkv * D;
  const bf16* kbase = k + ((long)batch * hkv + kv_head) * D;
  const bf16* vbase = v + ((long)batch * hkv + kv_head) * D;
  bf16* dkbase = dk + ((long)batch * hkv + kv_head) * D;
  bf16* dvbase = dv + ((long)batch * hkv + kv_head) * D;

  const int krow0 = k0 + wid * 16, rg = lane >> 4, cl = lane & 15;
  const int off = skv - sq;

  bf16x8 ka[NKD], va[NKD];
  {
    int krow = min(krow0 + cl, skv - 1);
    const bf16* kr = kbase + (long)krow * k_srow;
    const bf16* vr = vbase + (long)krow * k_srow;
#pragma unroll
    for (int kd = 0; kd < NKD; ++kd) {
      short8 t1 = *reinterpret_cast<const short8*>(kr + kd * 32 + rg * 8);
      ka[kd] = *reinterpret_cast<bf16x8*>(&t1);
      short8 t2 = *reinterpret_cast<const short8*>(vr + kd * 32 + rg * 8);
      va[kd] = *reinterpret_cast<bf16x8*>(&t2);
    }
  }

  f32x4 dk_acc[ND], dv_acc[ND];
#pragma unroll
  for (int n = 0; n < ND; ++n) {
    dk_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  int tq_start = causal ? max(0, (k0 - off) / QBLK) : 0;
  int tq_end = (sq - 1) / QBLK;
  if (window > 0) {
    // kv row k attends from q rows >= k - off ... < k - off + window
    tq_end = min(tq_end, (k0 + 127 - off + window - 1) / QBLK);
  }

  for (int hg = 0; hg < rep; ++hg) {
    const int head = kv_head * rep + hg;
    const bf16* qbase = q + ((long)batch * hq + head) * D;
    const bf16* dobase = dout + ((long)batch * hq + head) * D;
    const float* lse_row = lse + ((long)batch * hq + head) * sq;
    const float* dr_row = drow + ((long)batch * hq + head) * sq;

    for (int t = tq_start; t <= tq_end; ++t) {
      const int qt0 = t * QBLK;
      // ---- phase 1: stage Q row-major + dO transposed ----
      {
        constexpr int G = QBLK * D / 8;
        for (int idx = threadIdx.x; idx < G; idx += 256) {
          int r = idx / (D / 8), c8 = idx % (D / 8);
          int qrow = qt0 + r;
          short8 q8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
          if (qrow < sq)
            q8 = *reinterpret_cast<const short8*>(qbase + (long)qrow * q_srow + c8 * 8);
          *reinterpret_cast<short8*>(&BufB[r][c8 * 8]) = q8;
        }
        for (int idx = threadIdx.x; idx < G; idx += 512) {
          int r = idx & (QBLK - 1), c8 = idx / QBLK;
          int qrow = qt0 + r;
          short8 do8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
          if (qrow < sq)
            do8 = *reinterpret_cast<const short8*>(dobase + (long)qrow * q_srow + c8 * 8);
#pragma unroll
          for (int j = 0; j < 8; ++j) BufA[c8 * 8 + j][r] = do8[j];
        }
      }
      __syncthreads();

      // ---- phase 2: ST = K Q^T ; PT ; dV += PT dO (via BufA=dOt) ----
      f32x4 pt[QBLK / 16];
#pragma unroll
      for (int n = 0; n < QBLK / 16; ++n) {
        f32x4 st = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kd = 0; kd < NKD; ++kd) {
          bf16x8 bq = *reinterpret_cast<const bf16x8*>(&BufB[n * 16 + cl][kd * 32 + rg * 8]);
          st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ka[kd], bq, st, 0, 0, 0);
        }
        // D-layout: row = kv (krow0 + rg*4+r), col = q (qt0 + n*16 + cl)
        // BUT my_lse was loaded at q index qt0+cl — per-n q col is qt0+n*16+cl!
        float lse_n = lse_row[min(qt0 + n * 16 + cl, sq - 1)];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int kvrow = krow0 + rg * 4 + r;
          int qrow = qt0 + n * 16 + cl;
          bool ok = (kvrow < skv) && (qrow < sq);
          if (causal) ok &= (kvrow <= qrow + off);
          if (window > 0) ok &= (kvrow > qrow + off - window);
          pt[n][r] = ok ? exp2f((st[r] * scale - lse_n) * LOG2E) : 0.f;
          P_lds[wid][rg * 4 + r][n * 16 + cl] = f2sbf(pt[n][r]);
        }
      }
      __syncthreads();

      // dV[kv, d] += PT[kv, q] x dO[q, d]  (A from P_lds, B from BufA=dOt)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < QBLK / 32; ++ks) {
        short8 at = *reinterpret_cast<const short8*>(&P_lds[wid][cl][ks * 32 + rg * 8]);
        bf16x8 a = *reinterpret_cast<bf16x8*>(&at);
#pragma unroll
        for (int n = 0; n < ND; ++n) {
          bf16x8 bdo = *reinterpret_cast<const bf16x8*>(&BufA[n * 16 + cl][ks * 32 + rg * 8]);
          dv_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bdo, dv_acc[n], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      __syncthreads();

      // ---- phase 3: restage BufB <- dO row-major ; BufA <- Q^T ----
      {
        constexpr int G = QBLK * D / 8;
        for (int idx = threadIdx.x; idx < G; idx += 256) {
          int r = idx / (D / 8), c8 = idx % (D / 8);
          int qrow = qt0 + r;
          short8 do8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
          if (qrow < sq)
            do8 = *reinterpret_cast<const short8*>(dobase + (long)qrow * q_srow + c8 * 8);
          *reinterpret_cast<short8*>(&BufB[r][c8 * 8]) = do8;
        }
        for (int idx = threadIdx.x; idx < G; idx += 512) {
          int r = idx & (QBLK - 1), c8 = idx / QBLK;
          int qrow = qt0 + r;
          short8 q8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
          if (qrow < sq)
            q8 = *reinterpret_cast<const short8*>(qbase + (long)qrow * q_srow + c8 * 8);
#pragma unroll
          for (int j = 0; j < 8; ++j) BufA[c8 * 8 + j][r] = q8[j];
        }
      }
      __syncthreads();

      // ---- phase 4: dPT = V dO^T (B from BufB=dO row-major); dST; dK += dST Q ----
#pragma unroll
      for (int n = 0; n < QBLK / 16; ++n) {
        f32x4 dpt = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kd = 0; kd < NKD; ++kd) {
          bf16x8 bdo = *reinterpret_cast<const bf16x8*>(&BufB[n * 16 + cl][kd * 32 + rg * 8]);
          dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(va[kd], bdo, dpt, 0, 0, 0);
        }
        float dr_n = dr_row[min(qt0 + n * 16 + cl, sq - 1)];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float dst = pt[n][r] * (dpt[r] - dr_n) * scale;
          P_lds[wid][rg * 4 + r][n * 16 + cl] = f2sbf(dst);
        }
      }
      __syncthreads();

      // dK[kv, d] += dST[kv, q] x Q[q, d]  (A from P_lds, B from BufA=Qt)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < QBLK / 32; ++ks) {
        short8 at = *reinterpret_cast<const short8*>(&P_lds[wid][cl][ks * 32 + rg * 8]);
        bf16x8 a = *reinterpret_cast<bf16x8*>(&at);
#pragma unroll
        for (int n = 0; n < ND; ++n) {
          bf16x8 bq = *reinterpret_cast<const bf16x8*>(&BufA[n * 16 + cl][ks * 32 + rg * 8]);
          dk_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bq, dk_acc[n], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      __syncthreads();
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = krow0 + rg * 4 + r;
    if (row >= skv) continue;
    bf16* dkr = dkbase + (long)row * k_srow;
    bf16* dvr = dvbase + (long)row * k_srow;
#pragma unroll
    for (int n = 0; n < ND; ++n) {
      dkr[n * 16 + cl] = f2bf(dk_acc[n][r]);
      dvr[n * 16 + cl] = f2bf(dv_acc[n][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// host wrapper
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor out, torch::Tensor lse,
                                    bool causal, double scale, long window) {
  int sq = q.size(0), b = q.size(1), hq = q.size(2), d = q.size(3);
  int skv = k.size(0), hkv = k.size(2);
  auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
  auto doc = dout.contiguous(), oc = out.contiguous();
  auto dq = torch::empty_like(qc);
  auto dk = torch::empty_like(kc);
  auto dv = torch::empty_like(vc);
  auto drow = torch::empty({b, hq, sq}, q.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentHIPStream();

  long rows = (long)sq * b * hq;
  int d8 = d / 8;
  hipLaunchKernelGGL(attn_bwd_rowsum_kernel, dim3(2048), dim3(256), 0, stream,
                     (const short8*)doc.data_ptr(), (const short8*)oc.data_ptr(),
                     drow.data_ptr<float>(), rows, b * hq, sq, d8);

  dim3 grid_dq((sq + 63) / 64, b * hq);
  dim3 grid_dkv((skv + 127) / 128, b * hkv);
  if (d == 128) {
    hipLaunchKernelGGL((attn_bwd_dq_kernel<128>), grid_dq, dim3(256), 0, stream,
                       (const bf16*)qc.data_ptr(), (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
                       (const bf16*)doc.data_ptr(), lse.data_ptr<float>(), drow.data_ptr<float>(),
                       (bf16*)dq.data_ptr(), sq, skv, b, hq, hkv, (float)scale, causal ? 1 : 0, (int)window);
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<128>), grid_dkv, dim3(512), 0, stream,
                       (const bf16*)qc.data_ptr(), (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
                       (const bf16*)doc.data_ptr(), lse.data_ptr<float>(), drow.data_ptr<float>(),
                       (bf16*)dk.data_ptr(), (bf16*)dv.data_ptr(),
                       sq, skv, b, hq, hkv, (float)scale, causal ? 1 : 0, (int)window);
  } else {
    hipLaunchKernelGGL((attn_bwd_dq_kernel<64>), grid_dq, dim3(256), 0, stream,
                       (const bf16*)qc.data_ptr(), (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
                       (const bf16*)doc.data_ptr(), lse.data_ptr<float>(), drow.data_ptr<float>(),
                       (bf16*)dq.data_ptr(), sq, skv, b, hq, hkv, (float)scale, causal ? 1 : 0, (int)window);
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<64>), grid_dkv, dim3(512), 0, stream,
                       (const bf16*)qc.data_ptr(), (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
                       (const bf16*)doc.data_ptr(), lse.data_ptr<float>(), drow.data_ptr<float>(),
                       (bf16*)dk.data_ptr(), (bf16*)dv.data_ptr(),
                       sq, skv, b, hq, hkv, (float)scale, causal ? 1 : 0, (int)window);
  }
  return {dq, dk, dv};
}
