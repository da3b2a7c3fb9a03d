// MFMA-pipe worksheet for the round-3 asm decision.
//
// Question: does a 1-wave/SIMD hand-ordered stream (counted lgkmcnt,
// reads issued AHEAD of their MFMAs) beat the production structure's
// 2-waves/SIMD compiler schedule on the exact dependency shape of the
// attention inner loop — chains of ds_read_b128 -> v_mfma_f32_32x32x16
// with rotating accumulators?
//
// Variants (all: one 144 KB-LDS workgroup per CU so occupancy is forced;
// 16 reads + 16 MFMAs per step, CHAINS rotating accumulators):
//   cpp_2w : NT=512 (8 waves = 2/SIMD), compiler schedule — models the
//            production kernels.
//   cpp_1w : NT=256 (4 waves = 1/SIMD), compiler schedule — models the
//            rejected BA_FWD_NT=256 structure.
//   asm_1w : NT=256, the 16-step body in ordered inline asm with a
//            4-deep read-ahead ring and counted lgkmcnt(3) waits.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/mfma_pipe_probe.hip -o tools/mfma_pipe_probe
#include <hip/hip_runtime.h>
#include <stdio.h>
#include <stdlib.h>

typedef __attribute__((ext_vector_type(8))) _Float16 f16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_t;

#define CK(x)                                                        \
  do {                                                               \
    hipError_t e_ = (x);                                             \
    if (e_ != hipSuccess) {                                          \
      printf("HIP error %s @%d\n", hipGetErrorString(e_), __LINE__); \
      return 1;                                                      \
    }                                                                \
  } while (0)

constexpr int LDS_E = 72 * 1024;  // 144 KB of f16 -> one workgroup per CU

template <int NT, int CHAINS>
__global__ __launch_bounds__(NT) void pipe_cpp(float* out, int iters) {
  __shared__ _Float16 lds[LDS_E];
  for (int i = threadIdx.x; i < LDS_E; i += NT)
    lds[i] = (_Float16)((i & 7) * 0.125f);
  __syncthreads();
  const int lane = threadIdx.x & 63;
  f16x8_t b = *(const f16x8_t*)&lds[(lane & 31) * 8];
  f32x16_t acc[CHAINS];
#pragma unroll
  for (int c = 0; c < CHAINS; ++c) acc[c] = (f32x16_t)(0.f);
  // per-lane base, 16 B aligned, wanders per iteration (defeats hoisting)
  int base = (threadIdx.x * 16) & (2 * LDS_E - 16);
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int byte = (base + j * 2048) & (2 * LDS_E - 16);
      f16x8_t a = *(const f16x8_t*)((const char*)lds + byte);
      acc[j % CHAINS] =
          __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, acc[j % CHAINS], 0, 0, 0);
    }
    base = (base + 4096) & (2 * LDS_E - 16);
  }
  float s = 0.f;
#pragma unroll
  for (int c = 0; c < CHAINS; ++c)
#pragma unroll
    for (int r = 0; r < 16; ++r) s += acc[c][r];
  if (s == 1234.5678f) out[blockIdx.x] = s;  // keep everything live
}

// softmax-dependency variant: after every 16-MFMA group, a serial VALU
// block (max tree + 16 exp2 + packs, as the production softmax) that
// DEPENDS on one accumulator chain and FEEDS the next group's B operand
// — the production QK->softmax->PV structure.
template <int NT>
__global__ __launch_bounds__(NT) void pipe_cpp_dep(float* out, int iters) {
  __shared__ _Float16 lds[LDS_E];
  for (int i = threadIdx.x; i < LDS_E; i += NT)
    lds[i] = (_Float16)((i & 7) * 0.125f);
  __syncthreads();
  const int lane = threadIdx.x & 63;
  f16x8_t b = *(const f16x8_t*)&lds[(lane & 31) * 8];
  f32x16_t acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) acc[c] = (f32x16_t)(0.f);
  int base = (threadIdx.x * 16) & (2 * LDS_E - 16);
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int byte = (base + j * 2048) & (2 * LDS_E - 16);
      f16x8_t a = *(const f16x8_t*)((const char*)lds + byte);
      acc[j % 4] =
          __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, acc[j % 4], 0, 0, 0);
    }
    // softmax-shaped serial VALU on chain 0, feeding b (the dependency)
    float m = acc[0][0];
#pragma unroll
    for (int r = 1; r < 16; ++r) m = fmaxf(m, acc[0][r]);
    m = fmaxf(m, __shfl_xor(m, 32));
    float rs = 0.f;
    _Float16 h[8];
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      const float e0 = __builtin_amdgcn_exp2f(acc[0][2 * r] - m);
      const float e1 = __builtin_amdgcn_exp2f(acc[0][2 * r + 1] - m);
      rs += e0 + e1;
      h[r] = (_Float16)(e0 * 0.001f);
    }
    rs += __shfl_xor(rs, 32);
    b = *(f16x8_t*)h;  // feeds the next group's MFMAs
    acc[0] = (f32x16_t)(rs * 1e-30f);  // chain restarts (like st per subtile)
    base = (base + 4096) & (2 * LDS_E - 16);
  }
  float s = 0.f;
#pragma unroll
  for (int c = 0; c < 4; ++c)
#pragma unroll
    for (int r = 0; r < 16; ++r) s += acc[c][r];
  if (s == 1234.5678f) out[blockIdx.x] = s;
}

// asm variant: 4 accumulator chains, 4-slot read ring, counted waits.
// The whole 16-step body is ONE ordered asm block; the compiler only
// allocates registers.
__global__ __launch_bounds__(256) void pipe_asm(float* out, int iters) {
  __shared__ _Float16 lds[LDS_E];
  for (int i = threadIdx.x; i < LDS_E; i += 256)
    lds[i] = (_Float16)((i & 7) * 0.125f);
  __syncthreads();
  const int lane = threadIdx.x & 63;
  f16x8_t b = *(const f16x8_t*)&lds[(lane & 31) * 8];
  f32x16_t a0 = (f32x16_t)(0.f), a1 = (f32x16_t)(0.f), a2 = (f32x16_t)(0.f),
           a3 = (f32x16_t)(0.f);
  f16x8_t r0, r1, r2, r3;
  // 4 wandering addresses (byte offsets into the LDS object)
  int p0 = (threadIdx.x * 16) & (2 * LDS_E - 16);
  int p1 = (p0 + 2048) & (2 * LDS_E - 16);
  int p2 = (p0 + 4096) & (2 * LDS_E - 16);
  int p3 = (p0 + 6144) & (2 * LDS_E - 16);
  // LDS base as a flat-cast pointer for ds_read addressing
  auto lp = (__attribute__((address_space(3))) char*)lds;
  // prologue: fill the 4-deep ring
  asm volatile(
      "ds_read_b128 %0, %4\n\t"
      "ds_read_b128 %1, %5\n\t"
      "ds_read_b128 %2, %6\n\t"
      "ds_read_b128 %3, %7\n\t"
      : "=v"(r0), "=v"(r1), "=v"(r2), "=v"(r3)
      : "v"(lp + p0), "v"(lp + p1), "v"(lp + p2), "v"(lp + p3));
  for (int it = 0; it < iters; ++it) {
    // steady state: wait the OLDEST read only (3 younger stay in
    // flight), MFMA it, re-issue its slot 4 ahead.  16 steps = 4 full
    // ring turns; accumulator chain c = slot index (4 chains).
    asm volatile(
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %0, %4, %8, %0\n\t"
        "ds_read_b128 %4, %9 offset:512\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %1, %5, %8, %1\n\t"
        "ds_read_b128 %5, %10 offset:512\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %2, %6, %8, %2\n\t"
        "ds_read_b128 %6, %11 offset:512\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %3, %7, %8, %3\n\t"
        "ds_read_b128 %7, %12 offset:512\n\t"

        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %0, %4, %8, %0\n\t"
        "ds_read_b128 %4, %9 offset:1024\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %1, %5, %8, %1\n\t"
        "ds_read_b128 %5, %10 offset:1024\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %2, %6, %8, %2\n\t"
        "ds_read_b128 %6, %11 offset:1024\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %3, %7, %8, %3\n\t"
        "ds_read_b128 %7, %12 offset:1024\n\t"

        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %0, %4, %8, %0\n\t"
        "ds_read_b128 %4, %9 offset:1536\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %1, %5, %8, %1\n\t"
        "ds_read_b128 %5, %10 offset:1536\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %2, %6, %8, %2\n\t"
        "ds_read_b128 %6, %11 offset:1536\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %3, %7, %8, %3\n\t"
        "ds_read_b128 %7, %12 offset:1536\n\t"

        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %0, %4, %8, %0\n\t"
        "ds_read_b128 %4, %9\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %1, %5, %8, %1\n\t"
        "ds_read_b128 %5, %10\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %2, %6, %8, %2\n\t"
        "ds_read_b128 %6, %11\n\t"
        "s_waitcnt lgkmcnt(3)\n\t"
        "v_mfma_f32_32x32x16_f16 %3, %7, %8, %3\n\t"
        "ds_read_b128 %7, %12\n\t"
        "s_nop 7\n\t"
        : "+v"(a0), "+v"(a1), "+v"(a2), "+v"(a3), "+v"(r0), "+v"(r1),
          "+v"(r2), "+v"(r3)
        : "v"(b), "v"(lp + p0), "v"(lp + p1), "v"(lp + p2), "v"(lp + p3));
  }
  asm volatile("s_waitcnt lgkmcnt(0)\n\ts_nop 7" ::: "memory");
  float s = 0.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) s += a0[r] + a1[r] + a2[r] + a3[r];
  // keep the ring reads live too
  s += (float)r0[0] + (float)r1[0] + (float)r2[0] + (float)r3[0];
  if (s == 1234.5678f) out[blockIdx.x] = s;
}

// asm + gap-filler variant: same 16-MFMA/16-read ring, but each MFMA gap
// carries 4 softmax-shaped VALU instructions (max3 / exp2 / fma / pack)
// on a side register set — 64 VALU per group, the production softmax
// volume.  Measures whether softmax WORK rides the 1-wave MFMA stream.
__global__ __launch_bounds__(256) void pipe_asm_sm(float* out, int iters) {
  __shared__ _Float16 lds[LDS_E];
  for (int i = threadIdx.x; i < LDS_E; i += 256)
    lds[i] = (_Float16)((i & 7) * 0.125f);
  __syncthreads();
  const int lane = threadIdx.x & 63;
  f16x8_t b = *(const f16x8_t*)&lds[(lane & 31) * 8];
  f32x16_t a0 = (f32x16_t)(0.f), a1 = (f32x16_t)(0.f), a2 = (f32x16_t)(0.f),
           a3 = (f32x16_t)(0.f);
  f16x8_t r0, r1, r2, r3;
  float s0 = 0.1f * lane, s1 = 0.2f, s2 = 0.3f, s3 = 0.4f;
  int p0 = (threadIdx.x * 16) & (2 * LDS_E - 16);
  int p1 = (p0 + 2048) & (2 * LDS_E - 16);
  int p2 = (p0 + 4096) & (2 * LDS_E - 16);
  int p3 = (p0 + 6144) & (2 * LDS_E - 16);
  auto lp = (__attribute__((address_space(3))) char*)lds;
  asm volatile(
      "ds_read_b128 %0, %4\n\t"
      "ds_read_b128 %1, %5\n\t"
      "ds_read_b128 %2, %6\n\t"
      "ds_read_b128 %3, %7\n\t"
      : "=v"(r0), "=v"(r1), "=v"(r2), "=v"(r3)
      : "v"(lp + p0), "v"(lp + p1), "v"(lp + p2), "v"(lp + p3));
  for (int it = 0; it < iters; ++it) {
    // one macro step = wait, MFMA, read re-issue, 4 VALU fillers
#define STEP(ACC, RD, AD, OFF)                          \
  "s_waitcnt lgkmcnt(3)\n\t"                            \
  "v_mfma_f32_32x32x16_f16 " ACC ", " RD ", %12, " ACC  \
  "\n\t"                                                \
  "ds_read_b128 " RD ", " AD " offset:" OFF "\n\t"      \
  "v_max3_f32 %8, %8, %9, %10\n\t"                      \
  "v_exp_f32 %9, %9\n\t"                                \
  "v_fma_f32 %10, %10, %11, %8\n\t"                     \
  "v_cvt_pk_f16_f32 %11, %10, %9\n\t"
    asm volatile(
        STEP("%0", "%4", "%13", "512")
        STEP("%1", "%5", "%14", "512")
        STEP("%2", "%6", "%15", "512")
        STEP("%3", "%7", "%16", "512")
        STEP("%0", "%4", "%13", "1024")
        STEP("%1", "%5", "%14", "1024")
        STEP("%2", "%6", "%15", "1024")
        STEP("%3", "%7", "%16", "1024")
        STEP("%0", "%4", "%13", "1536")
        STEP("%1", "%5", "%14", "1536")
        STEP("%2", "%6", "%15", "1536")
        STEP("%3", "%7", "%16", "1536")
        STEP("%0", "%4", "%13", "0")
        STEP("%1", "%5", "%14", "0")
        STEP("%2", "%6", "%15", "0")
        STEP("%3", "%7", "%16", "0")
        "s_nop 7\n\t"
        : "+v"(a0), "+v"(a1), "+v"(a2), "+v"(a3), "+v"(r0), "+v"(r1),
          "+v"(r2), "+v"(r3), "+v"(s0), "+v"(s1), "+v"(s2), "+v"(s3)
        : "v"(b), "v"(lp + p0), "v"(lp + p1), "v"(lp + p2), "v"(lp + p3));
#undef STEP
  }
  asm volatile("s_waitcnt lgkmcnt(0)\n\ts_nop 7" ::: "memory");
  float s = s0 + s1 + s2 + s3;
#pragma unroll
  for (int r = 0; r < 16; ++r) s += a0[r] + a1[r] + a2[r] + a3[r];
  s += (float)r0[0] + (float)r1[0] + (float)r2[0] + (float)r3[0];
  if (s == 1234.5678f) out[blockIdx.x] = s;
}

// attention-shaped asm variant: the REAL phase schedule of the planned
// asm forward.  Per phase: 8 QK MFMAs on a SINGLE accumulator chain
// alternating with 8 PV MFMAs on 4 rotating chains (so the single
// chain is never back-to-back with itself); 3 softmax-model VALU per
// gap ending in the P packs; K and V each stream through a 4-slot read
// ring at a steady counted lgkmcnt(7).  Two phases unrolled so the
// stQ/stP states ping-pong positionally.  (The P->PV data link is
// order-guaranteed by the stream; its value flows through a fixed
// fragment here — the timing shape is what is measured.)
__global__ __launch_bounds__(256) void pipe_asm_att(float* out, int iters) {
  __shared__ _Float16 lds[LDS_E];
  for (int i = threadIdx.x; i < LDS_E; i += 256)
    lds[i] = (_Float16)((i & 7) * 0.125f);
  __syncthreads();
  const int lane = threadIdx.x & 63;
  f16x8_t qb = *(const f16x8_t*)&lds[(lane & 31) * 8];
  f16x8_t pb = *(const f16x8_t*)&lds[(lane & 31) * 8 + 256];
  f32x16_t o0 = (f32x16_t)(0.f), o1 = (f32x16_t)(0.f), o2 = (f32x16_t)(0.f),
           o3 = (f32x16_t)(0.f);
  f32x16_t sx = (f32x16_t)(0.f), sy = (f32x16_t)(0.f);
  f16x8_t kr0, kr1, kr2, kr3, vr0, vr1, vr2, vr3;
  float t0 = 0.1f * lane, t1 = 0.2f, t2 = 0.3f, t3 = 0.4f;
  int pk = (threadIdx.x * 16) & (2 * LDS_E - 16);
  int pv = (pk + 8192) & (2 * LDS_E - 16);
  auto lp = (__attribute__((address_space(3))) char*)lds;
  asm volatile(  // prologue: fill both rings (interleaved K,V order)
      "ds_read_b128 %0, %8\n\t"
      "ds_read_b128 %4, %9\n\t"
      "ds_read_b128 %1, %8 offset:512\n\t"
      "ds_read_b128 %5, %9 offset:512\n\t"
      "ds_read_b128 %2, %8 offset:1024\n\t"
      "ds_read_b128 %6, %9 offset:1024\n\t"
      "ds_read_b128 %3, %8 offset:1536\n\t"
      "ds_read_b128 %7, %9 offset:1536\n\t"
      : "=v"(kr0), "=v"(kr1), "=v"(kr2), "=v"(kr3), "=v"(vr0), "=v"(vr1),
        "=v"(vr2), "=v"(vr3)
      : "v"(lp + pk), "v"(lp + pv));
  for (int it = 0; it < iters; ++it) {
// one step pair: QK MFMA on the phase chain ST + PV MFMA on rotating OT
#define PH_STEP(ST, OT, KR, VR, KOFF, VOFF)            \
  "s_waitcnt lgkmcnt(7)\n\t"                           \
  "v_mfma_f32_32x32x16_f16 " ST ", " KR ", %18, " ST   \
  "\n\t"                                               \
  "ds_read_b128 " KR ", %20 offset:" KOFF "\n\t"       \
  "v_max3_f32 %14, %14, %15, %16\n\t"                  \
  "v_exp_f32 %15, %15\n\t"                             \
  "v_fma_f32 %16, %16, %17, %14\n\t"                   \
  "s_waitcnt lgkmcnt(7)\n\t"                           \
  "v_mfma_f32_32x32x16_f16 " OT ", " VR ", %19, " OT   \
  "\n\t"                                               \
  "ds_read_b128 " VR ", %21 offset:" VOFF "\n\t"       \
  "v_max3_f32 %17, %17, %14, %15\n\t"                  \
  "v_exp_f32 %14, %14\n\t"                             \
  "v_add_f32 %15, %15, %16\n\t"
#define PHASE(ST)                                           \
  PH_STEP(ST, "%0", "%6", "%10", "512", "2560")             \
  PH_STEP(ST, "%1", "%7", "%11", "1024", "3072")            \
  PH_STEP(ST, "%2", "%8", "%12", "1536", "3584")            \
  PH_STEP(ST, "%3", "%9", "%13", "2048", "4096")            \
  PH_STEP(ST, "%0", "%6", "%10", "0", "2048")               \
  PH_STEP(ST, "%1", "%7", "%11", "512", "2560")             \
  PH_STEP(ST, "%2", "%8", "%12", "1024", "3072")            \
  PH_STEP(ST, "%3", "%9", "%13", "1536", "3584")            \
  "v_cvt_pk_f16_f32 %14, %14, %15\n\t"                     \
  "v_cvt_pk_f16_f32 %16, %16, %17\n\t"
    asm volatile(
        PHASE("%4")
        PHASE("%5")
        "s_nop 7\n\t"
        : "+v"(o0), "+v"(o1), "+v"(o2), "+v"(o3), "+v"(sx), "+v"(sy),
          "+v"(kr0), "+v"(kr1), "+v"(kr2), "+v"(kr3), "+v"(vr0), "+v"(vr1),
          "+v"(vr2), "+v"(vr3), "+v"(t0), "+v"(t1), "+v"(t2), "+v"(t3)
        : "v"(qb), "v"(pb), "v"(lp + pk), "v"(lp + pv));
#undef PHASE
#undef PH_STEP
  }
  asm volatile("s_waitcnt lgkmcnt(0)\n\ts_nop 7" ::: "memory");
  float s = t0 + t1 + t2 + t3;
#pragma unroll
  for (int r = 0; r < 16; ++r)
    s += o0[r] + o1[r] + o2[r] + o3[r] + sx[r] + sy[r];
  s += (float)kr0[0] + (float)vr0[0];
  if (s == 1234.5678f) out[blockIdx.x] = s;
}

template <typename K>
static float bench(const char* name, K kern, int nt, float* out, int iters,
                   int waves_per_wg, int mfmas_per_iter) {
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  const int blocks = 256;  // one per CU
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(nt), 0, 0, out, iters);
  (void)hipDeviceSynchronize();
  hipError_t le = hipGetLastError();
  if (le != hipSuccess) {
    printf("%-10s LAUNCH ERROR: %s\n", name, hipGetErrorString(le));
    return 0;
  }
  (void)hipEventRecord(e0);
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(nt), 0, 0, out, iters);
  (void)hipEventRecord(e1);
  (void)hipEventSynchronize(e1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, e0, e1);
  // FLOPs: per wave-step MFMA = 2*32*32*16
  double fl = (double)blocks * waves_per_wg * iters * mfmas_per_iter * 2.0 *
              32 * 32 * 16;
  double tf = fl / (ms / 1e3) / 1e12;
  printf("%-10s %8.2f ms  %8.1f TF (%4.1f%% of 2.5PF)\n", name, ms, tf,
         tf / 2500 * 100);
  return tf;
}

int main(int argc, char** argv) {
  const int iters = (argc > 1) ? atoi(argv[1]) : 200000;
  float* out;
  CK(hipMalloc(&out, 1024 * 4));
  bench("cpp_2w_c1", pipe_cpp<512, 1>, 512, out, iters, 8, 16);
  bench("cpp_2w_c2", pipe_cpp<512, 2>, 512, out, iters, 8, 16);
  bench("cpp_2w_c4", pipe_cpp<512, 4>, 512, out, iters, 8, 16);
  bench("cpp_1w_c4", pipe_cpp<256, 4>, 256, out, iters, 4, 16);
  bench("cpp_1w_c8", pipe_cpp<256, 8>, 256, out, iters, 4, 16);
  bench("cpp_2w_dep", pipe_cpp_dep<512>, 512, out, iters, 8, 16);
  bench("cpp_1w_dep", pipe_cpp_dep<256>, 256, out, iters, 4, 16);
  bench("asm_1w_c4", pipe_asm, 256, out, iters, 4, 16);
  bench("asm_1w_sm", pipe_asm_sm, 256, out, iters, 4, 16);
  bench("asm_1w_att", pipe_asm_att, 256, out, iters, 4, 32);
  return 0;
}
