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

// raw v_exp_f32: exp2 without libm's denormal fixup (ldexp + selects per
// call); sub-2^-126 softmax terms flush to 0, which is exactly right here
__device__ __forceinline__ float ba_exp2(float x) {
  return __builtin_amdgcn_exp2f(x);
}

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
  // one v_cvt_pkrtz_f16_f32 instead of two converts + shift/or
  static __device__ __forceinline__ unsigned int pack2(float lo, float hi) {
    return __builtin_bit_cast(unsigned int, __builtin_amdgcn_cvt_pkrtz(lo, hi));
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
  static __device__ __forceinline__ unsigned int pack2(float lo, float hi) {
    return bits(lo) | (bits(hi) << 16);  // hipcc fuses to v_cvt_pk_bf16_f32
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
    unsigned int a0 = MT::pack2(p[8 * u + 0], p[8 * u + 1]);
    unsigned int a1 = MT::pack2(p[8 * u + 2], p[8 * u + 3]);
    unsigned int b0 = MT::pack2(p[8 * u + 4], p[8 * u + 5]);
    unsigned int b1 = MT::pack2(p[8 * u + 6], p[8 * u + 7]);
    i32x2_t r0 = __builtin_amdgcn_permlane32_swap((int)a0, (int)b0, false, false);
    i32x2_t r1 = __builtin_amdgcn_permlane32_swap((int)a1, (int)b1, false, false);
    u32x4_t w = {(unsigned int)r0[0], (unsigned int)r1[0],
                 (unsigned int)r0[1], (unsigned int)r1[1]};
    out[u] = __builtin_bit_cast(typename mfma_traits<T>::frag, w);
  }
}

// 16-value max via 3-input nesting (clang fuses fmaxf chains to v_max3)
__device__ __forceinline__ float ba_max16(const f32x16_t& v) {
  float a = fmaxf(fmaxf(v[0], v[1]), v[2]);
  float b = fmaxf(fmaxf(v[3], v[4]), v[5]);
  float c = fmaxf(fmaxf(v[6], v[7]), v[8]);
  float d = fmaxf(fmaxf(v[9], v[10]), v[11]);
  float e = fmaxf(fmaxf(v[12], v[13]), v[14]);
  float f = fmaxf(fmaxf(a, b), v[15]);
  return fmaxf(fmaxf(fmaxf(c, d), e), f);
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

typedef __attribute__((ext_vector_type(4))) short s16x4_t;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_t;

// ---- hardware transpose-read path (ds_read_tr16_b64, gfx950) ---------
// Probed semantics (profiles/r01/tr16_probe.txt): with l = 16h + 4q + i,
//   result[j](l) = lds_u16[ addr16(16h + 4j + q) + i ],  j = 0..3
// i.e. each lane reads 8B at its own address; a 4x4 transpose happens
// across lanes {16h+4j+q : j} -> {16h+4q+i : i}.  So an MFMA A-operand
// fragment A[row = dbase + (l&31)][k = kv0 + 8*(l>>5) + j] comes from a
// ROW-MAJOR [kv][D] image when each lane addresses
//   (kv0 + 8*(l>>5) + ((l>>2)&3)) * D + dbase + 16*((l>>4)&1) + 4*(l&3)
// Image swizzle: byte ^= (kv & 7) << 3 (8B granules) makes both the
// tr16 reads and the b64 staging writes (near-)conflict-free.
template <typename T, int D>
__device__ __forceinline__ int ba_tr16_byte(int lane, int kv0, int dbase) {
  const int kvr = kv0 + 8 * (lane >> 5) + ((lane >> 2) & 3);
  const int e16 = kvr * D + dbase + 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
  return (2 * e16) ^ ((kvr & 7) << 3);
}

template <typename T, int D>
__device__ __forceinline__ typename mfma_traits<T>::frag ba_ld_tr16_frag(
    const T* lds, int lane, int kv0, int dbase) {
  auto p0 = (__attribute__((address_space(3))) s16x4_t*)((
      __attribute__((address_space(3))) char*)(lds) +
      ba_tr16_byte<T, D>(lane, kv0, dbase));
  auto p1 = (__attribute__((address_space(3))) s16x4_t*)((
      __attribute__((address_space(3))) char*)(lds) +
      ba_tr16_byte<T, D>(lane, kv0 + 4, dbase));
  s16x4_t lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p0);
  s16x4_t hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p1);
  u32x2_t l2 = __builtin_bit_cast(u32x2_t, lo);
  u32x2_t h2 = __builtin_bit_cast(u32x2_t, hi);
  u32x4_t w = {l2[0], l2[1], h2[0], h2[1]};
  return __builtin_bit_cast(typename mfma_traits<T>::frag, w);
}

// staging write for the tr16 image: one 8-elem row chunk as two swizzled
// 8B stores (row-major + (kv&7)<<3 XOR)
template <typename T, int D>
__device__ __forceinline__ void ba_st_tr16row(T* lds, int kv, int d0,
                                              const u32x4_t& chunk) {
  const int b0 = (2 * (kv * D + d0)) ^ ((kv & 7) << 3);
  const int b1 = (2 * (kv * D + d0 + 4)) ^ ((kv & 7) << 3);
  u32x2_t lohalf = {chunk.x, chunk.y};
  u32x2_t hihalf = {chunk.z, chunk.w};
  *(u32x2_t*)((char*)lds + b0) = lohalf;
  *(u32x2_t*)((char*)lds + b1) = hihalf;
}

#define BA_CHECK_LAUNCH()                         \
  do {                                            \
    hipError_t e_ = hipGetLastError();            \
    if (e_ != hipSuccess) return (int)e_;         \
  } while (0)
