// Common helpers for megatron_amd CDNA4 (gfx950) HIP kernels.
// Wave width is 64 on CDNA4 (hard-coded per the HIP guide).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

#define HIP_CHECK(cmd)                                                              \
  do {                                                                              \
    hipError_t e = (cmd);                                                           \
    if (e != hipSuccess) {                                                          \
      throw std::runtime_error(std::string("HIP error: ") + hipGetErrorString(e) +  \
                               " at " __FILE__ ":" + std::to_string(__LINE__));     \
    }                                                                               \
  } while (0)

using bf16 = __hip_bfloat16;

// vector types for coalesced loads: 8 x bf16 = 16 B/lane (guide G13)
typedef short  short8  __attribute__((ext_vector_type(8)));
typedef float  float4v __attribute__((ext_vector_type(4)));
typedef float  float8v __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf2f(bf16 x) { return __bfloat162float(x); }
__device__ __forceinline__ bf16 f2bf(float x) { return __float2bfloat16(x); }

__device__ __forceinline__ float sbf2f(short s) {
  union { unsigned int u; float f; } cvt;
  cvt.u = ((unsigned int)(unsigned short)s) << 16;
  return cvt.f;
}
__device__ __forceinline__ short f2sbf(float f) {
  // round-to-nearest-even bf16 truncation
  union { float f; unsigned int u; } cvt;
  cvt.f = f;
  unsigned int lsb = (cvt.u >> 16) & 1;
  cvt.u += 0x7fff + lsb;
  return (short)(cvt.u >> 16);
}

// D-fragment -> transposed-A-fragment relayout (CDNA4, pure VALU).
//
// Input: a 16x64 MFMA D-tile of M held as 4 sub-tiles of f32x4 packed to
// bf16 pairs: pk[t][p] at lane (g = lane>>4, c = lane&15) holds the pair
// (M[16t + 4g + 2p][c], M[16t + 4g + 2p + 1][c]).
// Output: the A-fragment (bf16x8 as int4) of M^T for k-step `ks` (k = 32
// wide): lane (g,c) gets M^T[row=c][k = 32ks + 8g .. +7].
// Exactly two register-pair swaps per packed stream:
//   (U, W) = permlane32_swap(pk[2ks][p], pk[2ks+1][p])
//   (A_p, A_{p+2}) = permlane16_swap(U, W)
__device__ __forceinline__ void dfrag_to_at(const int pk0[2], const int pk1[2], int* av4) {
  auto uw0 = __builtin_amdgcn_permlane32_swap(pk0[0], pk1[0], false, false);
  auto aa0 = __builtin_amdgcn_permlane16_swap(uw0[0], uw0[1], false, false);
  auto uw1 = __builtin_amdgcn_permlane32_swap(pk0[1], pk1[1], false, false);
  auto aa1 = __builtin_amdgcn_permlane16_swap(uw1[0], uw1[1], false, false);
  av4[0] = aa0[0];
  av4[1] = aa1[0];
  av4[2] = aa0[1];
  av4[3] = aa1[1];
}

// pack two f32 into one dword of 2 bf16 (lo = first)
__device__ __forceinline__ int pack_bf16x2(float lo, float hi) {
  union { float f; unsigned u; } a, b;
  a.f = lo; b.f = hi;
  unsigned la = (a.u + (0x7fff + ((a.u >> 16) & 1))) >> 16;
  unsigned lb = (b.u + (0x7fff + ((b.u >> 16) & 1))) >> 16;
  return (int)((la & 0xffffu) | (lb << 16));
}

// full-wave reduction (64 lanes)
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}
__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block reduction: one value per thread -> one value broadcast to all.
// BLOCK must be a multiple of 64 and <= 1024.
template <int BLOCK>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds /*[BLOCK/64]*/) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  float r = (threadIdx.x < NW) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1) r += __shfl_xor(r, off, 64);
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  return lds[0];
}

static inline int grid_for(long total_threads_needed, int block) {
  long blocks = (total_threads_needed + block - 1) / block;
  // memory-bound grid sizing: cap ~2048 blocks, grid-stride the rest (G11)
  if (blocks > 2048) blocks = 2048;
  return (int)blocks;
}
