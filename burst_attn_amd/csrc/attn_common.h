// Shared helpers for the gfx950 BurstAttention tile kernels.
//
// Design notes (CDNA4 / MI355X):
//  * wave64; MFMA shape 32x32x16 (bf16/f16 in, fp32 accumulate).
//  * "Swapped" operand scheme: both the QK^T and PV contractions are
//    computed transposed (S^T = mfma(K, Q); O^T = mfma(V^T, P^T)) so that
//    every per-row softmax quantity (running max, row sum, rescale factor,
//    output normaliser) is LANE-LOCAL: the 32x32 MFMA D-layout gives each
//    lane one output COLUMN (col = lane&31), so with q on the column axis
//    the whole online-softmax state lives in registers of the lane that
//    owns that q row.  No cross-lane traffic per tile except one
//    shfl_xor(32) for the max/sum halves.
//  * P (f32, D-layout) is converted to MFMA A/B fragments in-register via
//    pack-to-2xT + v_permlane32_swap (lane halves exchange) — no LDS
//    round trip for P.
//  * K (and V) tiles are staged in LDS with a 16-byte XOR swizzle
//    (byte ^= (row & SWZ) << 4) so the per-lane row-slice ds_read_b128 of
//    the MFMA fragments is bank-conflict-free.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(8))) _Float16 f16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;
typedef __attribute__((ext_vector_type(2))) int i32x2_t;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_t;

#define BA_LOG2E 1.44269504088896340736f
#define BA_LN2 0.69314718055994530942f
#define BA_NEG_BIG (-1e30f)

template <typename T>
struct mfma_traits;

template <>
struct mfma_traits<_Float16> {
  using frag = f16x8_t;
  static __device__ __forceinline__ f32x16_t mma(frag a, frag b, f32x16_t c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
  }
  static __device__ __forceinline__ unsigned int bits(float x) {
    _Float16 h = (_Float16)x;
    return (unsigned int)__builtin_bit_cast(unsigned short, h);
  }
};

template <>
struct mfma_traits<__bf16> {
  using frag = bf16x8_t;
  static __device__ __forceinline__ f32x16_t mma(frag a, frag b, f32x16_t c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
  static __device__ __forceinline__ unsigned int bits(float x) {
    __bf16 h = (__bf16)x;
    return (unsigned int)__builtin_bit_cast(unsigned short, h);
  }
};

// 32x32 MFMA C/D register->row map: row = (r&3) + 8*(r>>2) + 4*hi,
// col = lane&31  (cdna_hip_programming.md §3)
__device__ __forceinline__ constexpr int ba_crow(int r, int hi) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

// LDS byte-offset swizzle (16B granules).  SWZ spreads a column access
// over slots within the row; SWZ2 adds a (row>>3) term so that the
// transposed-image staging writes (rows 8 apart across lanes) also land
// on distinct banks: 32-way -> ~4-way write conflicts, reads stay
// conflict-free (hand-checked for both ds_read_b128 lane groups).
template <int SWZ, int SWZ2 = 0>
__device__ __forceinline__ int ba_swz(int byte, int row) {
  return byte ^ ((((row & SWZ) ^ ((row >> 3) & SWZ2))) << 4);
}

// Build the two 16-deep MFMA fragments (k-slices u=0,1 of a 32-wide axis)
// from 16 lane-local f32 values in D-layout.  p[r] sits at axis position
// ba_crow(r, hi); the fragment wants position 8*hi + j contiguous:
//   w(frag u, dword t) pairs = permlane32_swap(pack(p[8u+2t], p[8u+2t+1]),
//                                              pack(p[8u+2t+4], p[8u+2t+5]))
template <typename T>
__device__ __forceinline__ void ba_build_frag_pair(
    const f32x16_t& p, typename mfma_traits<T>::frag out[2]) {
  using MT = mfma_traits<T>;
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    unsigned int a0 = MT::bits(p[8 * u + 0]) | (MT::bits(p[8 * u + 1]) << 16);
    unsigned int a1 = MT::bits(p[8 * u + 2]) | (MT::bits(p[8 * u + 3]) << 16);
    unsigned int b0 = MT::bits(p[8 * u + 4]) | (MT::bits(p[8 * u + 5]) << 16);
    unsigned int b1 = MT::bits(p[8 * u + 6]) | (MT::bits(p[8 * u + 7]) << 16);
    i32x2_t r0 = __builtin_amdgcn_permlane32_swap((int)a0, (int)b0, false, false);
    i32x2_t r1 = __builtin_amdgcn_permlane32_swap((int)a1, (int)b1, false, false);
    u32x4_t w = {(unsigned int)r0[0], (unsigned int)r1[0],
                 (unsigned int)r0[1], (unsigned int)r1[1]};
    out[u] = __builtin_bit_cast(typename mfma_traits<T>::frag, w);
  }
}

// 16B row-slice read: 8 contiguous elements of one row of a [rows][RS]
// row-major swizzled tile (RS = row size in elements).
template <typename T, int RS, int SWZ, int SWZ2 = 0>
__device__ __forceinline__ typename mfma_traits<T>::frag ba_ld_rowslice(
    const T* lds, int row, int elem0) {
  int byte = ba_swz<SWZ, SWZ2>(row * (2 * RS) + 2 * elem0, row);
  u32x4_t wv = *(const u32x4_t*)((const char*)lds + byte);
  return __builtin_bit_cast(typename mfma_traits<T>::frag, wv);
}

// transposed staging write: scatter one 8-element row chunk (elements
// d0..d0+7 of source row `src_row`) into a [RS_T rows][.] transposed image
// at rows d0..d0+7, column src_row.
template <typename T, int RS_T, int SWZ_T, int SWZ2_T = 7>
__device__ __forceinline__ void ba_st_transposed(T* lds, int src_row, int d0,
                                                 const u32x4_t& chunk) {
  const T* e = (const T*)&chunk;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int row = d0 + j;
    const int byte = ba_swz<SWZ_T, SWZ2_T>(row * (2 * RS_T) + 2 * src_row, row);
    *(T*)((char*)lds + byte) = e[j];
  }
}

#define BA_CHECK_LAUNCH()                         \
  do {                                            \
    hipError_t e_ = hipGetLastError();            \
    if (e_ != hipSuccess) return (int)e_;         \
  } while (0)
