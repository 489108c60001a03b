// Fused (flash-style) attention FORWARD for CDNA4 (K1 in SURVEY.md §2.3).
//
// Layout: q [s, b, hq, d], k/v [s, b, hkv, d] bf16, d in {64, 128};
// causal (with kv-offset for decode) or full; GQA (hq multiple of hkv);
// optional sliding window.  Outputs: o [s, b, hq, d] bf16, lse [b, hq, s] f32.
//
// v2 structure (technique stack per the CDNA4 guide §5.5 catalog):
//   * 8 waves, Q-block 256 rows (32/wave as 2x16 sub-tiles): K/V staging is
//     amortized over 4x the MFMA work of the 64-row v1.
//   * swapped QK^T - mfma(A=K, B=Q) gives S^T[kv][q] with q = lane&15, so the
//     softmax reduction is 2 shuffles (xor 16/32) and Q stays in registers
//     for the whole kernel (T12 structure).
//   * reg-staged issue-early/write-late K+V prefetch (T14): global loads for
//     tile t+1 issue before tile t's compute, drain (vmcnt 0) + LDS write +
//     one barrier at tile end. Double-buffered LDS.
//   * XOR chunk swizzle on K/Vt/P LDS tiles (T2): row-bit xor into the 16B
//     chunk index keeps ds_read_b128 bank-uniform with unpadded power-of-2
//     strides (16B alignment preserved).
//   * defer-max rescale (T13, THR=8): skip the O/l rescale pass while the
//     running max grows by < THR (exp headroom is harmless in f32 accum).
//   * per-wave causal tile skip + block-level tile range; XCD-bijective
//     workgroup remap so all q-chunks of one (batch,head) share one XCD's L2.
//   * s_setprio around MFMA clusters (T5).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define LOG2E 1.44269504088896340736f
#define WAIT_VM0 0x0F70    // vmcnt=0, expcnt/lgkmcnt = no-wait
#define RESCALE_THR 8.0f

// 16B-chunk XOR swizzle: row-major [R][C] bf16 tile, C a multiple of 8.
__device__ __forceinline__ int swz8(int row, int chunk) { return chunk ^ (row & 7); }

// ABL: 0=full, 1=no-softmax, 2=no-PV, 3=no-staging (ablation probes; wrong
// numerics for 1-3, perf triage only)
template <int D, int ABL = 0>
__global__ __launch_bounds__(512) void attn_fwd_kernel_v2(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    bf16* __restrict__ o, float* __restrict__ lse,
    int sq, int skv, int b, int hq, int hkv, float scale,
    int causal, int window, int nqc) {
  constexpr int QBLK = 256, KVBLK = 64;
  constexpr int NKD = D / 32;      // QK k-steps over d
  constexpr int ND = D / 16;       // d sub-tiles
  constexpr int KCH = D / 8;       // 16B chunks per K row
  constexpr int SREG = KVBLK * D / 8 / 512;  // staged short8 per lane (2 at D=128)

  // single-buffered: the T14 staging registers are the second buffer; the
  // extra barrier costs less than the halved occupancy of 2x LDS buffers
  __shared__ __align__(16) short K_lds[KVBLK][D];      // [kv][d], chunk-swizzled
  __shared__ __align__(16) short Vt_lds[D][KVBLK];     // [d][kv], chunk-swizzled

  // ---- XCD-bijective workgroup remap (T1; bijective per m204) ----
  const int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    int q8 = nwg >> 3, r8 = nwg & 7;
    int xcd = bid & 7, idx = bid >> 3;
    bid = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + idx;
  }
  const int bh = bid / nqc;       // batch * hq + head (contiguous per XCD)
  const int q0 = (bid % nqc) * QBLK;
  const int batch = bh / hq;
  const int head = bh % hq;
  const int kv_head = head / (hq / hkv);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int g = lane >> 4;              // 16-lane group (0..3)
  const int c = lane & 15;

  const long q_srow = (long)b * hq * D;
  const long k_srow = (long)b * hkv * D;
  const bf16* qbase = q + ((long)batch * hq + head) * D;
  const bf16* kbase = k + ((long)batch * hkv + kv_head) * D;
  const bf16* vbase = v + ((long)batch * hkv + kv_head) * D;
  bf16* obase = o + ((long)batch * hq + head) * D;

  const int q0w = q0 + wid * 32;        // this wave's first q row

  // ---- Q tile -> B fragments (kept in registers for the whole kernel) ----
  // B[col=q][k=d]: lane (g,c) holds Q[q0w + qt*16 + c][ks*32 + g*8 ..]
  bf16x8 qb[2][NKD];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt) {
    int qrow = min(q0w + qt * 16 + c, sq - 1);
    const bf16* qr = qbase + (long)qrow * q_srow;
#pragma unroll
    for (int ks = 0; ks < NKD; ++ks) {
      short8 tmp = *reinterpret_cast<const short8*>(qr + ks * 32 + g * 8);
      qb[qt][ks] = *reinterpret_cast<bf16x8*>(&tmp);
    }
  }

  f32x4 oacc[2][ND];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt)
#pragma unroll
    for (int n = 0; n < ND; ++n) oacc[qt][n] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[2] = {-1e30f, -1e30f}, l_run[2] = {0.f, 0.f};

  const int off = skv - sq;
  int t_end = causal ? min((q0 + QBLK - 1 + off) / KVBLK, (skv - 1) / KVBLK)
                     : (skv - 1) / KVBLK;
  int t_start = 0;
  if (window > 0) t_start = max(0, (q0 + off - window + 1) / KVBLK);

  // staging lane->element maps. K: row-linear (coalesced global loads,
  // vector LDS stores). V: kv-fast (global loads scatter across rows -- the
  // tile still covers every byte, L2 absorbs the ordering -- but the
  // transposed scalar LDS stores then spread over all 32 banks; the
  // row-linear map would leave only 4 distinct banks per store: 16-way
  // conflict, measured 3.2e8 SQ_LDS_BANK_CONFLICT vs 1.6e8 SQ_BUSY).
  int st_row[SREG], st_c8[SREG], sv_row[SREG], sv_c8[SREG];
#pragma unroll
  for (int i = 0; i < SREG; ++i) {
    int idx = (int)threadIdx.x + i * 512;
    st_row[i] = idx / KCH;
    st_c8[i] = idx % KCH;
    sv_row[i] = idx & (KVBLK - 1);
    sv_c8[i] = idx / KVBLK;
  }

  short8 kreg[SREG], vreg[SREG];
  // ---- prologue: stage tile t_start ----
  {
    const int k0 = t_start * KVBLK;
#pragma unroll
    for (int i = 0; i < SREG; ++i) {
      int krow = k0 + st_row[i];
      kreg[i] = short8{0, 0, 0, 0, 0, 0, 0, 0};
      vreg[i] = short8{0, 0, 0, 0, 0, 0, 0, 0};
      if (krow < skv)
        kreg[i] = *reinterpret_cast<const short8*>(kbase + (long)krow * k_srow + st_c8[i] * 8);
      int vrow = k0 + sv_row[i];
      if (vrow < skv)
        vreg[i] = *reinterpret_cast<const short8*>(vbase + (long)vrow * k_srow + sv_c8[i] * 8);
    }
#pragma unroll
    for (int i = 0; i < SREG; ++i) {
      *reinterpret_cast<short8*>(&K_lds[st_row[i]][swz8(st_row[i], st_c8[i]) * 8]) = kreg[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int d = sv_c8[i] * 8 + j;
        Vt_lds[d][swz8(d, sv_row[i] >> 3) * 8 + (sv_row[i] & 7)] = vreg[i][j];
      }
    }
    __syncthreads();
  }

  for (int t = t_start; t <= t_end; ++t) {
    const int k0 = t * KVBLK;
    // ---- issue next tile's global loads early (T14) ----
    const bool have_next = (ABL != 3) && (t + 1 <= t_end);
    if (have_next) {
      const int k0n = k0 + KVBLK;
#pragma unroll
      for (int i = 0; i < SREG; ++i) {
        int krow = k0n + st_row[i];
        kreg[i] = short8{0, 0, 0, 0, 0, 0, 0, 0};
        vreg[i] = short8{0, 0, 0, 0, 0, 0, 0, 0};
        if (krow < skv)
          kreg[i] = *reinterpret_cast<const short8*>(kbase + (long)krow * k_srow + st_c8[i] * 8);
        int vrow = k0n + sv_row[i];
        if (vrow < skv)
          vreg[i] = *reinterpret_cast<const short8*>(vbase + (long)vrow * k_srow + sv_c8[i] * 8);
      }
    }

    // per-wave skip: tile entirely masked for this wave's 32 q rows
    bool wave_skip = false;
    if (causal && k0 > q0w + 31 + off) wave_skip = true;
    if (window > 0 && k0 + KVBLK - 1 < q0w + off - window + 1) wave_skip = true;

    if (!wave_skip) {
      // ---- S^T = scale * K Q^T : D_S[row=kv][col=q] ----
      f32x4 st[2][KVBLK / 16];
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kvt = 0; kvt < KVBLK / 16; ++kvt) {
#pragma unroll
        for (int qt = 0; qt < 2; ++qt) st[qt][kvt] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < NKD; ++ks) {
          int krow = kvt * 16 + c;
          short8 tmp = *reinterpret_cast<const short8*>(
              &K_lds[krow][swz8(krow, ks * 4 + g) * 8]);
          bf16x8 ak = *reinterpret_cast<bf16x8*>(&tmp);
#pragma unroll
          for (int qt = 0; qt < 2; ++qt)
            st[qt][kvt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak, qb[qt][ks], st[qt][kvt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- mask + online softmax (per lane: q = q0w + qt*16 + c) ----
      if (ABL != 1) {
      bool tile_full = (k0 + KVBLK <= skv) && (q0w + 31 < sq);
      if (causal) tile_full &= (k0 + KVBLK - 1 <= q0w + off);
      if (window > 0) tile_full &= (k0 >= q0w + 31 + off - window + 1);

      // m/p tracked in RAW score domain (scale folded into the exp fma)
      float pmax[2] = {-1e30f, -1e30f};
      if (tile_full) {
#pragma unroll
        for (int qt = 0; qt < 2; ++qt)
#pragma unroll
          for (int kvt = 0; kvt < KVBLK / 16; ++kvt)
#pragma unroll
            for (int r = 0; r < 4; ++r)
              pmax[qt] = fmaxf(pmax[qt], st[qt][kvt][r]);
      } else {
#pragma unroll
        for (int qt = 0; qt < 2; ++qt) {
          int qrow = q0w + qt * 16 + c;
#pragma unroll
          for (int kvt = 0; kvt < KVBLK / 16; ++kvt)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              int kvcol = k0 + kvt * 16 + g * 4 + r;
              bool ok = (kvcol < skv) && (qrow < sq);
              if (causal) ok &= (kvcol <= qrow + off);
              if (window > 0) ok &= (kvcol > qrow + off - window);
              if (!ok) st[qt][kvt][r] = -1e30f;
              pmax[qt] = fmaxf(pmax[qt], st[qt][kvt][r]);
            }
        }
      }
      // reduce max over the 4 lanes holding this q column (xor 16, 32)
#pragma unroll
      for (int qt = 0; qt < 2; ++qt) {
        pmax[qt] = fmaxf(pmax[qt], __shfl_xor(pmax[qt], 16, 64));
        pmax[qt] = fmaxf(pmax[qt], __shfl_xor(pmax[qt], 32, 64));
      }

      // defer-max (T13): only rescale when the max grew by > THR (raw units)
      const float thr_raw = RESCALE_THR / scale;
      bool need = (pmax[0] > m_run[0] + thr_raw) || (pmax[1] > m_run[1] + thr_raw) ||
                  (m_run[0] == -1e30f);
      if (__any(need)) {
        float alpha[2];
#pragma unroll
        for (int qt = 0; qt < 2; ++qt) {
          float m_new = fmaxf(m_run[qt], pmax[qt]);
          alpha[qt] = __expf((m_run[qt] - m_new) * scale);
          m_run[qt] = m_new;
          l_run[qt] *= alpha[qt];
        }
        // broadcast alpha from q=c layout to O's q=g*4+r layout
#pragma unroll
        for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float a = __shfl(alpha[qt], (lane & 48) + g * 4 + r, 64);
#pragma unroll
            for (int n = 0; n < ND; ++n) oacc[qt][n][r] *= a;
          }
        }
      }

      // exp + row-sum: p = exp2(sv*c1 - m*c1), one fma + one exp each
      const float c1 = scale * LOG2E;
#pragma unroll
      for (int qt = 0; qt < 2; ++qt) {
        float ps = 0.f;
        const float mc = m_run[qt] * c1;
#pragma unroll
        for (int kvt = 0; kvt < KVBLK / 16; ++kvt)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float sv = st[qt][kvt][r];
            float p = exp2f(__builtin_fmaf(sv, c1, -mc));
            if (sv <= -1e29f) p = 0.f;
            st[qt][kvt][r] = p;
            ps += p;
          }
        l_run[qt] += ps;
      }

      }  // ABL != 1
      if (ABL == 2) {  // keep QK live, skip PV
#pragma unroll
        for (int qt = 0; qt < 2; ++qt)
#pragma unroll
          for (int kvt = 0; kvt < KVBLK / 16; ++kvt)
#pragma unroll
            for (int r = 0; r < 4; ++r) oacc[qt][0][r & 3] += st[qt][kvt][r];
      } else {
      // ---- pack P to bf16 pairs in registers ----
      // pk[qt][kvt][p] at lane (g,c): kv pair (16*kvt + 4g + 2p, +1) of q col c
      int pk[2][KVBLK / 16][2];
#pragma unroll
      for (int qt = 0; qt < 2; ++qt)
#pragma unroll
        for (int kvt = 0; kvt < KVBLK / 16; ++kvt)
#pragma unroll
          for (int p = 0; p < 2; ++p) {
            unsigned lo = (unsigned short)f2sbf(st[qt][kvt][2 * p]);
            unsigned hi = (unsigned short)f2sbf(st[qt][kvt][2 * p + 1]);
            pk[qt][kvt][p] = (int)(lo | (hi << 16));
          }

      // ---- O += P V : D_O[row=q][col=d], A=P (built in-register), B=Vt ----
      // The D-fragment -> A-fragment relayout is exactly two register-pair
      // swaps (T12 family, pure VALU, no LDS):
      //   (U, W)   = permlane32_swap(pk[2ks2][p], pk[2ks2+1][p])
      //   (A_p, A_{p+2}) = permlane16_swap(U, W)
      // giving A-frag pair u at lane (g,c) = kv pair 16*ks2*2 + 4g' ... =
      // pair index 16*ks2 + 4g + u as the A layout requires.
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks2 = 0; ks2 < KVBLK / 32; ++ks2) {
        bf16x8 pa[2];
#pragma unroll
        for (int qt = 0; qt < 2; ++qt) {
          int au[4];
#pragma unroll
          for (int p = 0; p < 2; ++p) {
            auto uw = __builtin_amdgcn_permlane32_swap(
                pk[qt][2 * ks2][p], pk[qt][2 * ks2 + 1][p], false, false);
            auto aa = __builtin_amdgcn_permlane16_swap(uw[0], uw[1], false, false);
            au[p] = aa[0];
            au[p + 2] = aa[1];
          }
          int4 av = make_int4(au[0], au[1], au[2], au[3]);
          pa[qt] = *reinterpret_cast<bf16x8*>(&av);
        }
#pragma unroll
        for (int n = 0; n < ND; ++n) {
          int vrow = n * 16 + c;
          short8 tmp = *reinterpret_cast<const short8*>(
              &Vt_lds[vrow][swz8(vrow, ks2 * 4 + g) * 8]);
          bf16x8 bv = *reinterpret_cast<bf16x8*>(&tmp);
#pragma unroll
          for (int qt = 0; qt < 2; ++qt)
            oacc[qt][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa[qt], bv, oacc[qt][n], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      }  // ABL != 2
    }

    // ---- overwrite the LDS tile with staged tile t+1 ----
    if (have_next) {
      __syncthreads();  // everyone done reading tile t
      __builtin_amdgcn_s_waitcnt(WAIT_VM0);
      __builtin_amdgcn_sched_barrier(0);
#pragma unroll
      for (int i = 0; i < SREG; ++i) {
        *reinterpret_cast<short8*>(&K_lds[st_row[i]][swz8(st_row[i], st_c8[i]) * 8]) = kreg[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int d = sv_c8[i] * 8 + j;
          Vt_lds[d][swz8(d, sv_row[i] >> 3) * 8 + (sv_row[i] & 7)] = vreg[i][j];
        }
      }
      __syncthreads();  // tile t+1 visible
    }
  }

  // ---- epilogue ----
  // finish l: reduce the lane-partials over the 4 lanes per q column
  float linv_c[2];
#pragma unroll
  for (int qt = 0; qt < 2; ++qt) {
    l_run[qt] += __shfl_xor(l_run[qt], 16, 64);
    l_run[qt] += __shfl_xor(l_run[qt], 32, 64);
    linv_c[qt] = (l_run[qt] > 0.f) ? 1.f / l_run[qt] : 0.f;
    int qrow = q0w + qt * 16 + c;
    if (g == 0 && qrow < sq)
      lse[((long)batch * hq + head) * sq + qrow] =
          m_run[qt] * scale + __logf(fmaxf(l_run[qt], 1e-30f));
  }
#pragma unroll
  for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float inv = __shfl(linv_c[qt], (lane & 48) + g * 4 + r, 64);
      int qrow = q0w + qt * 16 + g * 4 + r;
      if (qrow >= sq) continue;
      bf16* orow = obase + (long)qrow * q_srow;
#pragma unroll
      for (int n = 0; n < ND; ++n) orow[n * 16 + c] = f2bf(oacc[qt][n][r] * inv);
    }
  }
}

// ---------------------------------------------------------------------------
// v1 kernel (64-row blocks) kept for short sequences (sq < 256).
// ---------------------------------------------------------------------------

template <int D>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k, const bf16* __restrict__ v,
    bf16* __restrict__ o, float* __restrict__ lse,
    int sq, int skv, int b, int hq, int hkv, float scale,
    int causal, int window) {
  constexpr int QBLK = 64, KVBLK = 64;
  constexpr int KPAD = D + 8;
  constexpr int VPAD = KVBLK + 8;
  constexpr int ND = D / 16;
  constexpr int NKD = D / 32;

  __shared__ short K_lds[KVBLK][KPAD];
  __shared__ short Vt_lds[D][VPAD];
  __shared__ short P_lds[4][16][KVBLK + 8];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int q0 = blockIdx.x * QBLK;
  const int bh = blockIdx.y;
  const int batch = bh / hq;
  const int head = bh % hq;
  const int kv_head = head / (hq / hkv);

  const long q_srow = (long)b * hq * D;
  const long k_srow = (long)b * hkv * D;
  const bf16* qbase = q + ((long)batch * hq + head) * D;
  const bf16* kbase = k + ((long)batch * hkv + kv_head) * D;
  const bf16* vbase = v + ((long)batch * hkv + kv_head) * D;
  bf16* obase = o + ((long)batch * hq + head) * D;

  const int row0 = q0 + wid * 16;
  const int rg = lane >> 4;
  const int cl = lane & 15;

  bf16x8 qa[NKD];
  {
    int qrow = row0 + cl;
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

  int t_end = causal ? min((q0 + QBLK - 1 + (skv - sq)) / KVBLK, (skv - 1) / KVBLK)
                     : (skv - 1) / KVBLK;
  int t_start = 0;
  if (window > 0) t_start = max(0, (q0 + (skv - sq) - window + 1) / KVBLK);

  for (int t = t_start; t <= t_end; ++t) {
    const int k0 = t * KVBLK;
    {
      constexpr int G = KVBLK * D / 8;
      for (int idx = threadIdx.x; idx < G; idx += 256) {
        int r = idx / (D / 8), c8 = idx % (D / 8);
        int krow = k0 + r;
        short8 kv8 = short8{0, 0, 0, 0, 0, 0, 0, 0};
        if (krow < skv)
          kv8 = *reinterpret_cast<const short8*>(kbase + (long)krow * k_srow + c8 * 8);
        *reinterpret_cast<short8*>(&K_lds[r][c8 * 8]) = kv8;
      }
      for (int idx = threadIdx.x; idx < G; idx += 256) {
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
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off2 = 1; off2 < 16; off2 <<= 1) pmax[r] = fmaxf(pmax[r], __shfl_xor(pmax[r], off2, 64));
    }
    float alpha[4], psum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_run[r], pmax[r]);
      alpha[r] = __expf(m_run[r] - m_new);
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
        if (s[n][r] <= -1e29f) p = 0.f;
        s[n][r] = p;
        psum[r] += p;
      }
    }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off2 = 1; off2 < 16; off2 <<= 1) psum[r] += __shfl_xor(psum[r], off2, 64);
      l_run[r] = l_run[r] * alpha[r] + psum[r];
#pragma unroll
      for (int n = 0; n < ND; ++n) oacc[n][r] *= alpha[r];
    }

#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) P_lds[wid][rg * 4 + r][n * 16 + cl] = f2sbf(s[n][r]);
    }
    __builtin_amdgcn_s_waitcnt(0);

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
  auto stream = at::cuda::getCurrentHIPStream();
  static const bool force_v1 = getenv("MEGATRON_AMD_ATTN_V1") != nullptr;
  static const char* abl_env = getenv("MEGA_ATTN_ABL");
  static const int abl = abl_env ? atoi(abl_env) : 0;
  if (sq >= 256 && !force_v1) {
    int nqc = (sq + 255) / 256;
    dim3 grid(nqc * b * hq);
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, grid, dim3(512), 0, stream,
                         (const bf16*)qc.data_ptr(), (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
                         (bf16*)out.data_ptr(), lse.data_ptr<float>(),
                         sq, skv, b, hq, hkv, (float)scale, causal ? 1 : 0, (int)window, nqc);
    };
    if (d == 128) {
      if (abl == 1) launch(attn_fwd_kernel_v2<128, 1>);
      else if (abl == 2) launch(attn_fwd_kernel_v2<128, 2>);
      else if (abl == 3) launch(attn_fwd_kernel_v2<128, 3>);
      else launch(attn_fwd_kernel_v2<128, 0>);
    } else {
      launch(attn_fwd_kernel_v2<64, 0>);
    }
    return {out, lse};
  }
  dim3 grid((sq + 63) / 64, b * hq);
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
