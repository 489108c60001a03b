// Fused (flash-style) attention FORWARD for CDNA4 (K1 in SURVEY.md §2.3).
//
// Layout: q [s, b, hq, d], k/v [s, b, hkv, d] bf16, d in {64, 128};
// causal (with kv-offset for decode) or full; GQA (hq multiple of hkv);
// optional sliding window.  Outputs: o [s, b, hq, d] bf16, lse [b, hq, s] f32.
//
// Structure (v1, correctness-first; see guide §B for the technique ladder):
//   block = 4 waves, Q-tile 64 rows (16/wave), KV-tile 64.
//   Q in registers; K staged in LDS row-major (+8 bf16 pad kills the
//   32-way bank conflict); V staged TRANSPOSED (Vt[d][kv]) so the PV
//   B-fragment reads are contiguous 16B ds_reads.
//   mfma_f32_16x16x32_bf16 fragments:
//     A: row = lane&15, k = (lane>>4)*8+j   (8 contiguous bf16)
//     B: col = lane&15, k = (lane>>4)*8+j
//     D: col = lane&15, row = (lane>>4)*4+reg   [guide §3, m89/m91-verified]
//   Online softmax in fp32 with per-row m/l tracked per lane-group.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define LOG2E 1.44269504088896340736f

template <int D>  // head dim: 64 or 128
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    bf16* __restrict__ o, float* __restrict__ lse,
    int sq, int skv, int b, int hq, int hkv, float scale,
    int causal, int window) {
  constexpr int QBLK = 64, KVBLK = 64;  // 4 waves x 16 q rows
  constexpr int KPAD = D + 8;   // K_lds row stride (bf16)
  constexpr int VPAD = KVBLK + 8;
  constexpr int ND = D / 16;    // d-tiles (4 or 8)
  constexpr int NKD = D / 32;   // k-steps over d

  __shared__ short K_lds[KVBLK][KPAD];
  __shared__ short Vt_lds[D][VPAD];
  __shared__ short P_lds[4][16][KVBLK + 8];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int q0 = blockIdx.x * QBLK;
  const int bh = blockIdx.y;            // batch * hq + head
  const int batch = bh / hq;
  const int head = bh % hq;
  const int kv_head = head / (hq / hkv);

  const long q_srow = (long)b * hq * D;   // q stride along s
  const long k_srow = (long)b * hkv * D;
  const bf16* qbase = q + ((long)batch * hq + head) * D;
  const bf16* kbase = k + ((long)batch * hkv + kv_head) * D;
  const bf16* vbase = v + ((long)batch * hkv + kv_head) * D;
  bf16* obase = o + ((long)batch * hq + head) * D;

  const int row0 = q0 + wid * 16;       // this wave's first q row
  const int rg = lane >> 4;             // 16-lane group (0..3)
  const int cl = lane & 15;             // col-in-tile / row-in-A

  // ---- load Q tile into registers (A fragments per 32-wide k step) ----
  bf16x8 qa[NKD];
  {
    int qrow = row0 + cl;  // A-frag row
    const bf16* qr = qbase + (long)min(qrow, sq - 1) * q_srow;
#pragma unroll
    for (int kd = 0; kd < NKD; ++kd) {
      short8 tmp = *reinterpret_cast<const short8*>(qr + kd * 32 + rg * 8);
      qa[kd] = *reinterpret_cast<bf16x8*>(&tmp);
    }
  }

  f32x4 oacc[ND];
#pragma unroll
  for (int n = 0; n < ND; ++n) oacc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }

  // causal: query row i attends to kv <= i + (skv - sq)
  int t_end = causal ? min((q0 + QBLK - 1 + (skv - sq)) / KVBLK, (skv - 1) / KVBLK)
                     : (skv - 1) / KVBLK;
  int t_start = 0;
  if (window > 0) t_start = max(0, (q0 + (skv - sq) - window + 1) / KVBLK);

  for (int t = t_start; t <= t_end; ++t) {
    const int k0 = t * KVBLK;
    // ---- stage K (row-major, coalesced) and V (transposed, r-fast) ----
    {
      constexpr int G = KVBLK * D / 8;
      for (int idx = threadIdx.x; idx < G; idx += 256) {
        // K: coalesced loads, row-major vector stores
        int r = idx / (D / 8), c8 = idx % (D / 8);
        int krow = k0 + r;
        short8 kv8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
        if (krow < skv)
          kv8 = *reinterpret_cast<const short8*>(kbase + (long)krow * k_srow + c8 * 8);
        *reinterpret_cast<short8*>(&K_lds[r][c8 * 8]) = kv8;
      }
      for (int idx = threadIdx.x; idx < G; idx += 256) {
        // V: r-fast mapping -> transposed scalar stores hit ~all banks
        // (row-major mapping would put all 16 same-j lanes in one bank:
        //  8-row x 16B-aligned stride aliases mod 32 banks)
        int r = idx & (KVBLK - 1), c8 = idx / KVBLK;
        int krow = k0 + r;
        short8 vv8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
        if (krow < skv)
          vv8 = *reinterpret_cast<const short8*>(vbase + (long)krow * k_srow + c8 * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) Vt_lds[c8 * 8 + j][r] = vv8[j];
      }
    }
    __syncthreads();

    // ---- S = scale * Q K^T  (4 col-tiles of 16) ----
    f32x4 s[KVBLK / 16];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n) {
      s[n] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kd = 0; kd < NKD; ++kd) {
        bf16x8 bk = *reinterpret_cast<const bf16x8*>(&K_lds[n * 16 + cl][kd * 32 + rg * 8]);
        s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa[kd], bk, s[n], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + online softmax ----
    // interior tiles need no masking (wave rows all >= row0):
    const int off = skv - sq;
    bool tile_full = (k0 + KVBLK <= skv) && (row0 + 15 < sq);
    if (causal) tile_full &= (k0 + KVBLK - 1 <= row0 + off);
    if (window > 0) tile_full &= (k0 >= row0 + 15 + off - window + 1);
    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) pmax[r] = -1e30f;
    if (tile_full) {
#pragma unroll
      for (int n = 0; n < KVBLK / 16; ++n) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float val = s[n][r] * scale;
          s[n][r] = val;
          pmax[r] = fmaxf(pmax[r], val);
        }
      }
    } else {
#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n) {
      int col = k0 + n * 16 + cl;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + rg * 4 + r;
        bool ok = (col < skv) && (row < sq);
        if (causal) ok &= (col <= row + off);
        if (window > 0) ok &= (col > row + off - window);
        float val = ok ? s[n][r] * scale : -1e30f;
        s[n][r] = val;
        pmax[r] = fmaxf(pmax[r], val);
      }
    }
    }
    // row-reduce max over the 16 lanes of each group
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) pmax[r] = fmaxf(pmax[r], __shfl_xor(pmax[r], off, 64));
    }
    float alpha[4], psum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_run[r], pmax[r]);
      alpha[r] = __expf(m_run[r] - m_new);   // exp(-inf - -inf)=exp(0-0) safe: m_run>=-1e30
      m_run[r] = m_new;
      psum[r] = 0.f;
    }
    if (tile_full) {
#pragma unroll
      for (int n = 0; n < KVBLK / 16; ++n) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = exp2f((s[n][r] - m_run[r]) * LOG2E);
          s[n][r] = p;
          psum[r] += p;
        }
      }
    } else {
#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = exp2f((s[n][r] - m_run[r]) * LOG2E);
        if (s[n][r] <= -1e29f) p = 0.f;  // fully-masked guard
        s[n][r] = p;
        psum[r] += p;
      }
    }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) psum[r] += __shfl_xor(psum[r], off, 64);
      l_run[r] = l_run[r] * alpha[r] + psum[r];
#pragma unroll
      for (int n = 0; n < ND; ++n) oacc[n][r] *= alpha[r];
    }

    // ---- P to LDS (D-layout -> A-layout round trip) ----
#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) P_lds[wid][rg * 4 + r][n * 16 + cl] = f2sbf(s[n][r]);
    }
    // P_lds is wave-private: same-wave LDS RAW needs only the compiler's
    // lgkmcnt (same-array dependency), not a block-wide barrier.
    __builtin_amdgcn_s_waitcnt(0 /* vmcnt0 lgkmcnt0 expcnt0 */);

    // ---- O += P V ----
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < KVBLK / 32; ++ks) {
      short8 ptmp = *reinterpret_cast<const short8*>(&P_lds[wid][cl][ks * 32 + rg * 8]);
      bf16x8 pa = *reinterpret_cast<bf16x8*>(&ptmp);
#pragma unroll
      for (int n = 0; n < ND; ++n) {
        bf16x8 bv = *reinterpret_cast<const bf16x8*>(&Vt_lds[n * 16 + cl][ks * 32 + rg * 8]);
        oacc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, bv, oacc[n], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  // ---- epilogue ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = row0 + rg * 4 + r;
    if (row >= sq) continue;
    float inv = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
    bf16* orow = obase + (long)row * q_srow;
#pragma unroll
    for (int n = 0; n < ND; ++n) orow[n * 16 + cl] = f2bf(oacc[n][r] * inv);
    if (cl == 0) lse[((long)batch * hq + head) * sq + row] = m_run[r] + __logf(fmaxf(l_run[r], 1e-30f));
  }
}

// PV B-fragment note: B[k][col] with col = d index, k = kv index ->
// V[k0+k][dcol] = Vt_lds[dcol][k]; read above as Vt_lds[n*16+cl][ks*32+rg*8+j]. OK.

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    bool causal, double scale, long window) {
  TORCH_CHECK(q.dim() == 4 && q.dtype() == torch::kBFloat16);
  int sq = q.size(0), b = q.size(1), hq = q.size(2), d = q.size(3);
  int skv = k.size(0), hkv = k.size(2);
  TORCH_CHECK(d == 64 || d == 128, "head dim must be 64 or 128");
  TORCH_CHECK(hq % hkv == 0);
  auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
  auto out = torch::empty_like(qc);
  auto lse = torch::empty({b, hq, sq}, q.options().dtype(torch::kFloat32));
  dim3 grid((sq + 63) / 64, b * hq);
  auto stream = at::cuda::getCurrentHIPStream();
  if (d == 128)
    hipLaunchKernelGGL((attn_fwd_kernel<128>), grid, dim3(256), 0, stream,
                       (const bf16*)qc.data_ptr(), (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
                       (bf16*)out.data_ptr(), lse.data_ptr<float>(),
                       sq, skv, b, hq, hkv, (float)scale, causal ? 1 : 0, (int)window);
  else
    hipLaunchKernelGGL((attn_fwd_kernel<64>), grid, dim3(256), 0, stream,
                       (const bf16*)qc.data_ptr(), (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
                       (bf16*)out.data_ptr(), lse.data_ptr<float>(),
                       sq, skv, b, hq, hkv, (float)scale, causal ? 1 : 0, (int)window);
  return {out, lse};
}
