// BurstAttention backward tile kernels for gfx950 (MI355X, CDNA4).
//
// Replaces the flash-attn backward the reference calls at
// burst_attn/burst_utils.py:211-248, with the tile math of the reference's
// own restatement (burst_utils.py:77-101):
//     p   = exp(q k^T * scale - lse)
//     dv += p^T do;   dp = do v^T;   ds = p * (dp - delta) * scale
//     dq += ds k;     dk += ds^T q
//
// dq/dk/dv are fp32 ACCUMULATORS: every kernel adds its tile contribution
// in place (strided), which lets the ring layer accumulate across rounds
// with no elementwise-add passes or per-round allocations.
//
// Two kernel plans behind bahip_attn_bwd:
//   deterministic=1 (atomic-free, 8 tile GEMMs — flash-attn executes 5):
//     1. bwd_preprocess: delta = rowsum(o * do) fp32  (flash's preprocess)
//     2. dq kernel  (8 waves, q-block resident, streams k/v):
//          S^T = mfma(K,Q); dP^T = mfma(V,dO); dS^T in-lane;
//          dQ^T += mfma(K^T, dS^T)          — all softmax state lane-local
//     3. dkdv kernels (kv-block resident, streams q/do):
//          MODE 0 (dV): S = mfma(Q,K^T); dV^T += mfma(dO^T, P)
//          MODE 1 (dK): + dP = mfma(dO,V^T); dK^T += mfma(Q^T, dS)
//   deterministic=0 (default, 6 tile GEMMs — the flash-attn plan):
//     1. bwd_preprocess
//     2. bwd_dkq kernel (kv-resident, FLIPPED orientation: S^T = mfma(K,Q)
//        with kv on the MFMA row axis and q on the lane axis, so lse/delta
//        are lane-local — no per-row constant folding, no serial LDS->MFMA
//        prologue):  S^T, dP^T, then
//          dK^T += mfma(Q^T, dS)   (dS fragments via a per-wave LDS
//                                   transpose of the dS^T D-layout)
//          dQ   += mfma(dS^T, K^T) (fragments built in-register; partials
//                                   LDS-reduced across waves, flushed with
//                                   one global fp32 atomic-add per
//                                   workgroup q-subtile)
//     3. dkdv MODE 0 (dV) as above.
// Operand scheme and LDS swizzle: see attn_common.h.

#include "attn_common.h"
#include "../../include/burst_attn_hip.h"

#include <stdio.h>
#include <stdlib.h>

namespace {

// ====================== delta preprocess ==============================
template <typename T, int D>
__global__ __launch_bounds__(256) void bwd_preprocess_kernel(
    const T* __restrict__ o, const T* __restrict__ dout,
    float* __restrict__ delta, int S, int N,
    int64_t o_sb, int64_t o_ss, int64_t o_sh,
    int64_t g_sb, int64_t g_ss, int64_t g_sh) {
  // one wave: 8 rows x 8 lanes/row, 16B..32B per lane
  constexpr int EPL = D / 8;  // elements per lane (16 for D=128)
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in = wave * 8 + (lane >> 3);
  const int sub = lane & 7;
  const int s = blockIdx.x * 32 + row_in;
  const int n = blockIdx.y, b = blockIdx.z;
  if (s >= S) return;
  const T* op = o + b * o_sb + (int64_t)s * o_ss + n * o_sh + sub * EPL;
  const T* gp = dout + b * g_sb + (int64_t)s * g_ss + n * g_sh + sub * EPL;
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < EPL; ++j) acc += (float)op[j] * (float)gp[j];
  acc += __shfl_xor(acc, 1);
  acc += __shfl_xor(acc, 2);
  acc += __shfl_xor(acc, 4);
  if (sub == 0) delta[((int64_t)b * N + n) * S + s] = acc;
}

// ====================== dq kernel (q-resident) ========================
constexpr int KVBLK = 64;

// DQP=1 (env BA_DQ_PIPE): register-relief + explicit read-ahead
// experiment driven by the r2 wait-taxonomy finding: the
// dO fragments move from 32 registers to a persistent LDS image, and
// the freed registers hold a 2-deep software pipeline of the k/v
// fragment reads so each ds_read has >= 2 MFMAs of cover.
template <typename T, int D, int DQP = 0>
__global__ __launch_bounds__(512) void bwd_dq_kernel(
    const T* __restrict__ dout, const T* __restrict__ q,
    const T* __restrict__ k, const T* __restrict__ v,
    const float* __restrict__ delta, const float* __restrict__ lse,
    float* __restrict__ dq, int Sq, int Sk, int N,
    int64_t g_sb, int64_t g_ss, int64_t g_sh,
    int64_t q_sb, int64_t q_ss, int64_t q_sh,
    int64_t k_sb, int64_t k_ss, int64_t k_sh,
    int64_t v_sb, int64_t v_ss, int64_t v_sh,
    int64_t d_sb, int64_t d_sh, int64_t l_sb, int64_t l_sh,
    int64_t dq_sb, int64_t dq_ss, int64_t dq_sh,
    float scale, int causal) {
  using MT = mfma_traits<T>;
  using frag = typename MT::frag;
  constexpr int SWZ = (D == 128) ? 15 : 7;   // row-major images (2*D B rows)
  constexpr int SWZ_T = 7;                   // transposed images (128 B rows)
  constexpr int NT = 512;
  constexpr int PT = (KVBLK * D / 8) / NT;

  // [2 buffers][K row-major | V row-major | K^T transposed]
  // (+ DQP: persistent dO image for the workgroup's 256 q rows)
  __shared__ T lds[2 * 3 * KVBLK * D + (DQP ? 256 * D : 0)];
  auto ldsK = [&](int buf) -> T* { return lds + buf * (3 * KVBLK * D); };
  auto ldsV = [&](int buf) -> T* { return lds + buf * (3 * KVBLK * D) + KVBLK * D; };
  auto ldsKT = [&](int buf) -> T* { return lds + buf * (3 * KVBLK * D) + 2 * KVBLK * D; };
  T* ldsDO = lds + 2 * 3 * KVBLK * D;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31, hi = lane >> 5;
  const int n = blockIdx.y, b = blockIdx.z;
  const int qb = blockIdx.x * 256 + wave * 32;
  const int q_row = qb + l31;

  const T* qp = q + b * q_sb + (int64_t)n * q_sh;
  const T* gp = dout + b * g_sb + (int64_t)n * g_sh;
  const T* kp = k + b * k_sb + (int64_t)n * k_sh;
  const T* vp = v + b * v_sb + (int64_t)n * v_sh;

  frag qf[D / 16], gf[DQP ? 1 : D / 16];
#pragma unroll
  for (int s = 0; s < D / 16; ++s) {
    if (q_row < Sq) {
      qf[s] = __builtin_bit_cast(
          frag, *(const u32x4_t*)(qp + (int64_t)q_row * q_ss + 16 * s + 8 * hi));
      if (!DQP)
        gf[s] = __builtin_bit_cast(
            frag,
            *(const u32x4_t*)(gp + (int64_t)q_row * g_ss + 16 * s + 8 * hi));
    } else {
      u32x4_t z = {0, 0, 0, 0};
      qf[s] = __builtin_bit_cast(frag, z);
      if (!DQP) gf[s] = __builtin_bit_cast(frag, z);
    }
  }
  if (DQP) {
    // one-time: stage the workgroup's dO block [256 q][D] row-major
    // (clamped rows contribute garbage x 0 through the q_row mask)
    constexpr int PTD = (256 * D / 8) / NT;
#pragma unroll
    for (int c = 0; c < PTD; ++c) {
      const int flat = tid + c * NT;
      const int row = flat / (D / 8), col8 = flat % (D / 8);
      const int qg = blockIdx.x * 256 + row;
      const int qc = qg < Sq ? qg : (Sq - 1);
      u32x4_t ch = *(const u32x4_t*)(gp + (int64_t)qc * g_ss + col8 * 8);
      const int byte = ba_swz<SWZ>(row * (2 * D) + col8 * 16, row);
      *(u32x4_t*)((char*)ldsDO + byte) = ch;
    }
  }
  const float c2 = scale * BA_LOG2E;
  const float lse2 =
      (q_row < Sq) ? lse[b * l_sb + n * l_sh + q_row] * BA_LOG2E : 0.f;
  const float dlt =
      (q_row < Sq) ? delta[b * d_sb + n * d_sh + q_row] : 0.f;

  f32x16_t dqt[D / 32];
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt) dqt[dt] = (f32x16_t)(0.f);

  const int kv_limit = causal ? min(Sk, (int)(blockIdx.x + 1) * 256) : Sk;
  const int nt = (kv_limit + KVBLK - 1) / KVBLK;

  auto issue_loads = [&](int tile, u32x4_t* kreg, u32x4_t* vreg) {
    const int kv0 = tile * KVBLK;
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      const int row = flat / (D / 8), col8 = flat % (D / 8);
      // clamped instead of guarded (see attn_fwd.hip): out-of-range kv
      // rows re-load row Sk-1; their S values are masked to p = 0
      const int kvg = kv0 + row;
      const int kvc = kvg < Sk ? kvg : (Sk - 1);
      kreg[c] = *(const u32x4_t*)(kp + (int64_t)kvc * k_ss + col8 * 8);
      vreg[c] = *(const u32x4_t*)(vp + (int64_t)kvc * v_ss + col8 * 8);
    }
  };
  auto write_lds = [&](int buf, const u32x4_t* kreg, const u32x4_t* vreg) {
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      const int row = flat / (D / 8), col8 = flat % (D / 8);
      const int byte = ba_swz<SWZ>(row * (2 * D) + col8 * 16, row);
      *(u32x4_t*)((char*)ldsK(buf) + byte) = kreg[c];
      *(u32x4_t*)((char*)ldsV(buf) + byte) = vreg[c];
      ba_st_transposed<T, KVBLK, SWZ_T, 7>(ldsKT(buf), row, col8 * 8, kreg[c]);
    }
  };

  {
    u32x4_t kreg[PT], vreg[PT];
    issue_loads(0, kreg, vreg);
    write_lds(0, kreg, vreg);
    __syncthreads();
  }
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static form (younger half)

  int cur = 0;
  for (int t = 0; t < nt; ++t) {
    const int kv0 = t * KVBLK;
    const bool has_next = (t + 1) < nt;
    u32x4_t kreg[PT], vreg[PT];
    if (has_next) issue_loads(t + 1, kreg, vreg);

    const bool active = !causal || (kv0 <= qb + 31);
    if (active) {
#pragma unroll
      for (int kvs = 0; kvs < 2; ++kvs) {
        f32x16_t st = (f32x16_t)(0.f), dpt = (f32x16_t)(0.f);
        if (DQP) {
          // explicit 2-deep read-ahead: fragment reads for slice s+1
          // issue before slice s's MFMAs, so each ds_read is covered by
          // >= 2 MFMAs instead of the allocator's read-wait-use
          frag kfA = ba_ld_rowslice<T, D, SWZ>(ldsK(cur), kvs * 32 + l31,
                                               8 * hi);
          frag vfA = ba_ld_rowslice<T, D, SWZ>(ldsV(cur), kvs * 32 + l31,
                                               8 * hi);
          frag gfA = ba_ld_rowslice<T, D, SWZ>(ldsDO, wave * 32 + l31,
                                               8 * hi);
#pragma unroll
          for (int s = 0; s < D / 16; ++s) {
            frag kfB = kfA, vfB = vfA, gfB = gfA;
            if (s + 1 < D / 16) {
              kfA = ba_ld_rowslice<T, D, SWZ>(ldsK(cur), kvs * 32 + l31,
                                              16 * (s + 1) + 8 * hi);
              vfA = ba_ld_rowslice<T, D, SWZ>(ldsV(cur), kvs * 32 + l31,
                                              16 * (s + 1) + 8 * hi);
              gfA = ba_ld_rowslice<T, D, SWZ>(ldsDO, wave * 32 + l31,
                                              16 * (s + 1) + 8 * hi);
            }
            st = MT::mma(kfB, qf[s], st);
            dpt = MT::mma(vfB, gfB, dpt);
          }
        } else {
#pragma unroll
        for (int s = 0; s < D / 16; ++s) {
          frag kf = ba_ld_rowslice<T, D, SWZ>(ldsK(cur), kvs * 32 + l31,
                                              16 * s + 8 * hi);
          frag vf = ba_ld_rowslice<T, D, SWZ>(ldsV(cur), kvs * 32 + l31,
                                              16 * s + 8 * hi);
          st = MT::mma(kf, qf[s], st);
          dpt = MT::mma(vf, gf[DQP ? 0 : s], dpt);
        }
        }
        // p^T = exp2(S^T*c2 - lse2);  dS^T = p^T*(dP^T - delta)*scale
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv_g = kv0 + kvs * 32 + ba_crow(r, 0) + 4 * hi;
          const bool valid =
              kv_g < Sk && (!causal || kv_g <= q_row) && q_row < Sq;
          const float e = valid ? __builtin_fmaf(st[r], c2, -lse2) : BA_NEG_BIG;
          const float p = ba_exp2(e);
          st[r] = p * (dpt[r] - dlt) * scale;
        }
        frag dsf[2];
        ba_build_frag_pair<T>(st, dsf);
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt) {
          const int col = dt * 32 + l31;
#pragma unroll
          for (int u = 0; u < 2; ++u) {
            frag ktf = ba_ld_rowslice<T, KVBLK, SWZ_T, 7>(
                ldsKT(cur), col, kvs * 32 + 16 * u + 8 * hi);
            dqt[dt] = MT::mma(ktf, dsf[u], dqt[dt]);
          }
        }
      }
    }
    if (has_next) write_lds(cur ^ 1, kreg, vreg);
    __syncthreads();
    cur ^= 1;
  }

  if (q_row < Sq) {
    // accumulate epilogue: each (b, q_row, n) row is written by exactly
    // one lane, so a plain read-add-write stays deterministic
    float* drow = dq + b * dq_sb + (int64_t)q_row * dq_ss + n * dq_sh;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        drow[dt * 32 + ba_crow(r, hi)] += dqt[dt][r];
  }
}

// ============== dk and dv kernels (kv-resident, q/do streamed) =========
// Two single-purpose 8-wave kernels instead of one fused dk+dv kernel:
// the fused form needs 4 x 64 accumulator registers per lane and spilled
// (~90 VGPRs) at the 2-waves/SIMD budget; split, each kernel is register
// -clean and runs at the dq kernel's rate.  Cost: S is recomputed (the
// backward executes 8 tile GEMMs total vs flash-attn's 5).
// MODE: 0 = dV (dv^T += mfma(dO^T, P)); 1 = dK (dk^T += mfma(Q^T, dS));
//       2 = fused dK+dV (both accumulators; 8-wave; may spill — A/B)
// QBLK: streamed q-tile rows (128 for dV measured -8%: 27-VGPR spills).
template <typename T, int D, int MODE, int QBLK = 64>
__global__ __launch_bounds__(512) void bwd_dkv_kernel(
    const T* __restrict__ dout, const T* __restrict__ q,
    const T* __restrict__ k, const T* __restrict__ v,
    const float* __restrict__ delta, const float* __restrict__ lse,
    float* __restrict__ dout_acc, float* __restrict__ dout_acc2,
    int Sq, int Sk, int N,
    int64_t g_sb, int64_t g_ss, int64_t g_sh,
    int64_t q_sb, int64_t q_ss, int64_t q_sh,
    int64_t k_sb, int64_t k_ss, int64_t k_sh,
    int64_t v_sb, int64_t v_ss, int64_t v_sh,
    int64_t d_sb, int64_t d_sh, int64_t l_sb, int64_t l_sh,
    int64_t o_sb, int64_t o_ss, int64_t o_sh,
    float scale, int causal) {
  using MT = mfma_traits<T>;
  using frag = typename MT::frag;
  constexpr int SWZ = (D == 128) ? 15 : 7;
  constexpr int SWZ_T = 7;
  constexpr int NT = 512;
  constexpr int PT = (QBLK * D / 8) / NT;
  // images per buffer: MODE_DV: [Q row-major | dO^T]; MODE_DK:
  // [Q row-major | dO row-major | Q^T]; MODE 2: all four
  constexpr int IMGS = MODE == 0 ? 2 : (MODE == 1 ? 3 : 4);

  // one LDS object (a second __shared__ forces vmcnt(0) on every ds_read
  // — cdna guide §5 trap 4a).  lse/delta do NOT go through LDS: a
  // float-cast tail here defeated alias analysis and made hipcc emit
  // lgkmcnt(0) before EVERY ds_read of the stream images (50 full
  // drains per tile vs the dq kernel's counted waits — the round-2 wait
  // taxonomy's 58% parked dK).  They ride per-lane global loads,
  // prefetched one tile ahead, instead.
  __shared__ T lds[2 * IMGS * QBLK * D];
  auto ldsQ = [&](int buf) -> T* { return lds + buf * (IMGS * QBLK * D); };
  // MODE_DV: transposed dO; MODE_DK: row-major dO
  auto ldsG = [&](int buf) -> T* {
    return lds + buf * (IMGS * QBLK * D) + QBLK * D;
  };
  auto ldsQT = [&](int buf) -> T* {
    return lds + buf * (IMGS * QBLK * D) + 2 * QBLK * D;
  };
  auto ldsGT2 = [&](int buf) -> T* {  // MODE 2 only: dO^T as 4th image
    return lds + buf * (IMGS * QBLK * D) + 3 * QBLK * D;
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31, hi = lane >> 5;
  const int n = blockIdx.y, b = blockIdx.z;
  const int kvb = blockIdx.x * 256 + wave * 32;
  const int kv_col = kvb + l31;

  const T* qp = q + b * q_sb + (int64_t)n * q_sh;
  const T* gp = dout + b * g_sb + (int64_t)n * g_sh;
  const float* dp_ = delta + b * d_sb + n * d_sh;
  const float* lp_ = lse + b * l_sb + n * l_sh;

  // resident K (always, for S); resident V when dP is needed
  frag kf[D / 16], vf[MODE >= 1 ? D / 16 : 1];
  {
    const T* kp = k + b * k_sb + (int64_t)n * k_sh;
    const T* vp = v + b * v_sb + (int64_t)n * v_sh;
#pragma unroll
    for (int s = 0; s < D / 16; ++s) {
      if (kv_col < Sk) {
        kf[s] = __builtin_bit_cast(
            frag,
            *(const u32x4_t*)(kp + (int64_t)kv_col * k_ss + 16 * s + 8 * hi));
        if (MODE >= 1)
          vf[s] = __builtin_bit_cast(
              frag,
              *(const u32x4_t*)(vp + (int64_t)kv_col * v_ss + 16 * s + 8 * hi));
      } else {
        u32x4_t z = {0, 0, 0, 0};
        kf[s] = __builtin_bit_cast(frag, z);
        if (MODE >= 1) vf[s] = __builtin_bit_cast(frag, z);
      }
    }
  }
  const float c2 = scale * BA_LOG2E;

  f32x16_t acc[D / 32];
  f32x16_t acc2[MODE == 2 ? D / 32 : 1];  // MODE 2: dV accumulator
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt) acc[dt] = (f32x16_t)(0.f);
  if (MODE == 2)
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt) acc2[dt] = (f32x16_t)(0.f);

  // causal: q tiles wholly before this workgroup's kv rows are masked
  const int t0 = causal ? (blockIdx.x * 256) / QBLK : 0;
  const int nt = (Sq + QBLK - 1) / QBLK;

  // per-lane lse*log2e and delta for both subtiles of a tile (l2[qs] =
  // {lse2, delta}), loaded straight from global — clamped, branchless
  auto issue_loads = [&](int tile, u32x4_t* qreg, u32x4_t* greg,
                         float2* l2) {
    const int q0 = tile * QBLK;
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      const int row = flat / (D / 8), col8 = flat % (D / 8);
      // clamped instead of guarded (see attn_fwd.hip): out-of-range q
      // rows re-load row Sq-1; the q_g < Sq mask zeroes their P/dS, so
      // they contribute garbage x 0 to the dV/dK contractions
      const int qg = q0 + row;
      const int qc = qg < Sq ? qg : (Sq - 1);
      qreg[c] = *(const u32x4_t*)(qp + (int64_t)qc * q_ss + col8 * 8);
      greg[c] = *(const u32x4_t*)(gp + (int64_t)qc * g_ss + col8 * 8);
    }
#pragma unroll
    for (int qs = 0; qs < QBLK / 32; ++qs) {
      const int qg = q0 + qs * 32 + l31;
      const int qc = qg < Sq ? qg : (Sq - 1);
      // clamped rows contaminate nothing: the q_g < Sq mask zeroes p/dS
      l2[qs].x = lp_[qc] * BA_LOG2E;
      l2[qs].y = (MODE >= 1) ? dp_[qc] : 0.f;
    }
  };
  auto write_lds = [&](int buf, const u32x4_t* qreg, const u32x4_t* greg) {
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      const int row = flat / (D / 8), col8 = flat % (D / 8);
      const int byte = ba_swz<SWZ>(row * (2 * D) + col8 * 16, row);
      *(u32x4_t*)((char*)ldsQ(buf) + byte) = qreg[c];
      if (MODE == 0) {
        ba_st_transposed<T, QBLK, SWZ_T, 7>(ldsG(buf), row, col8 * 8, greg[c]);
      } else {
        // MODE_DK keeps the scatter-write transposed image: the tr16
        // address set spills ~47 VGPRs here (measured -14% fwd+bwd)
        *(u32x4_t*)((char*)ldsG(buf) + byte) = greg[c];
        ba_st_transposed<T, QBLK, SWZ_T, 7>(ldsQT(buf), row, col8 * 8, qreg[c]);
        if (MODE == 2)
          ba_st_tr16row<T, D>(ldsGT2(buf), row, col8 * 8, greg[c]);
      }
    }
  };

  float2 l2c[QBLK / 32];
  {
    u32x4_t qreg[PT], greg[PT];
    issue_loads(t0, qreg, greg, l2c);
    write_lds(0, qreg, greg);
    __syncthreads();
  }
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static form (younger half)

  int cur = 0;
  for (int t = t0; t < nt; ++t) {
    const int q0 = t * QBLK;
    const bool has_next = (t + 1) < nt;
    u32x4_t qreg[PT], greg[PT];
    float2 l2n[QBLK / 32] = {};
    if (has_next) issue_loads(t + 1, qreg, greg, l2n);

    const bool active = !causal || (q0 + QBLK - 1 >= kvb);
    if (active) {
      const float2* ld2s = l2c;  // this tile's lse2/delta (prefetched)
#pragma unroll
      for (int qs = 0; qs < QBLK / 32; ++qs) {
        // ---- augmentation fold: the per-row constants ride the MFMA.
        // One extra k-slice per GEMM with B = [1,1,0...] (lanes of the
        // low half) and A carrying a row constant split hi/lo into two
        // fp16/bf16 elements (the split keeps the exponent-domain error
        // at fp32 roundoff): S accumulates -lse*log2e/c2 so the exp
        // argument is just st*c2; dP accumulates -delta so dS is just
        // p*dpt*scale.  Replaces 32 per-row LDS reads + fma/sub chains
        // (68% WAIT_ANY and 24 spilled VGPRs on the dK kernel).
        const float2 ld2 = ld2s[qs];
        frag ones01, ones0, a_lse, a_dlt;
        {
          u32x4_t z = {0, 0, 0, 0};
          ones01 = __builtin_bit_cast(frag, z);
          a_lse = __builtin_bit_cast(frag, z);
          a_dlt = __builtin_bit_cast(frag, z);
          if (hi == 0) {
            const float av = -ld2.x / c2;
            const T ah = (T)av;
            a_lse[0] = ah;
            a_lse[1] = (T)(av - (float)ah);
            const T dh = (T)(-ld2.y);
            a_dlt[0] = dh;
            a_dlt[1] = (T)(-ld2.y - (float)dh);
            ones01[0] = (T)1.f;
            ones01[1] = (T)1.f;
          }
          ones0 = ones01;
        }
        f32x16_t st = MT::mma(a_lse, ones01, (f32x16_t)(0.f));
        f32x16_t dpt = (f32x16_t)(0.f);
        if (MODE >= 1) dpt = MT::mma(a_dlt, ones0, dpt);
#pragma unroll
        for (int s = 0; s < D / 16; ++s) {
          frag qfr = ba_ld_rowslice<T, D, SWZ>(ldsQ(cur), qs * 32 + l31,
                                               16 * s + 8 * hi);
          st = MT::mma(qfr, kf[s], st);
          if (MODE >= 1) {
            frag gfr = ba_ld_rowslice<T, D, SWZ>(ldsG(cur), qs * 32 + l31,
                                                 16 * s + 8 * hi);
            dpt = MT::mma(gfr, vf[s], dpt);
          }
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int q_loc = qs * 32 + ba_crow(r, 0) + 4 * hi;
          const int q_g = q0 + q_loc;
          const bool valid =
              q_g < Sq && kv_col < Sk && (!causal || q_g >= kv_col);
          const float e = valid ? st[r] * c2 : BA_NEG_BIG;
          const float p = ba_exp2(e);
          if (MODE == 0) {
            st[r] = p;  // P for dV
          } else if (MODE == 1) {
            st[r] = p * dpt[r] * scale;  // dS for dK
          } else {
            dpt[r] = p * dpt[r] * scale;  // dS
            st[r] = p;                    // P
          }
        }
        frag f01[2];
        ba_build_frag_pair<T>(st, f01);
        frag f01b[MODE == 2 ? 2 : 1];
        if (MODE == 2) ba_build_frag_pair<T>(dpt, (frag*)f01b);
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt) {
          const int drow = dt * 32 + l31;
#pragma unroll
          for (int u = 0; u < 2; ++u) {
            // MODE_DV: A = dO^T row-slice; MODE_DK: A = Q^T row-slice
            const T* timg = (MODE == 0) ? ldsG(cur) : ldsQT(cur);
            frag tf = ba_ld_rowslice<T, QBLK, SWZ_T, 7>(
                timg, drow, qs * 32 + 16 * u + 8 * hi);
            if (MODE == 2) {
              // acc = dK (dS frags), acc2 = dV (P frags via dO^T)
              acc[dt] = MT::mma(tf, f01b[u], acc[dt]);
              frag gtf = ba_ld_tr16_frag<T, D>(ldsGT2(cur), lane,
                                               qs * 32 + 16 * u, dt * 32);
              acc2[dt] = MT::mma(gtf, f01[u], acc2[dt]);
            } else {
              acc[dt] = MT::mma(tf, f01[u], acc[dt]);
            }
          }
        }
      }
    }
    if (has_next) write_lds(cur ^ 1, qreg, greg);
    __syncthreads();
    cur ^= 1;
#pragma unroll
    for (int qs = 0; qs < QBLK / 32; ++qs) l2c[qs] = l2n[qs];
  }

  if (kv_col < Sk) {
    // accumulate epilogue (single-writer per row: read-add-write)
    float* row = dout_acc + b * o_sb + (int64_t)kv_col * o_ss + n * o_sh;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        row[dt * 32 + ba_crow(r, hi)] += acc[dt][r];
    if (MODE == 2) {
      float* row2 = dout_acc2 + b * o_sb + (int64_t)kv_col * o_ss + n * o_sh;
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r)
          row2[dt * 32 + ba_crow(r, hi)] += acc2[dt][r];
    }
  }
}

// ============== fused dK + dQ kernel (kv-resident, flipped) ============
// FLIPPED orientation: S^T = mfma(K, Q) puts kv on the MFMA row axis and
// q on the lane axis, so lse/delta are LANE-LOCAL scalars (read straight
// from global, coalesced) — the split dK kernel's per-q-row constants
// (the "augmentation fold" MFMAs and their serial LDS->MFMA prologue,
// 65% SQ_WAIT in the round-1 PMCs) disappear.
//   dS^T sits in D-layout [kv rows][q cols].  From it:
//   * dQ fragments [row=q][contr=kv] come from ba_build_frag_pair
//     IN-REGISTER; dQ += mfma(dS^T, K^T) against a persistent K^T LDS
//     image (K is kv-block resident, staged transposed once).  Per-wave
//     partials (each wave owns 32 kv) are LDS-reduced (ds_add) across the
//     workgroup's 256 kv and flushed with ONE global fp32 atomic-add per
//     q-subtile — flash-attn's accumulation plan (hence deterministic=0
//     only).
//   * dK fragments [row=kv][contr=q] need the other orientation: a
//     per-wave LDS transpose (16 b16 scatter writes + 2 b128 reads, no
//     barrier — wave-private scratch); dK^T += mfma(Q^T, dS) stays
//     register-resident exactly like the split kernel.
// FUSE_DQ=0 degenerates to a flipped dK-only kernel (atomic-free — legal
// for deterministic=1; kept for A/B against the split MODE 1).
//
// Atomic-contention control: the grid is launched HEAD-FIRST
// (blockIdx.x = head) so concurrent workgroups flush different heads'
// dq regions, and non-causal workgroups start their q-tile stream at a
// per-kv-block offset (wrapping) so same-head workgroups flush
// DIFFERENT q tiles at any instant — without both, the 256 same-head
// workgroups march the same 16 KB of dq lines in lockstep and the
// line-serialised L2 RMWs dominate (measured 18x).
// DBG (env BA_DKQ_DBG, perf probes only — results wrong for >0):
//   1 = skip the global atomic flush;  2 = plain LDS writes, no ds_add;
//   3 = skip the whole dQ section (KT staging/LDS size/barriers kept);
//   4 = dQ MFMAs with dsT as both operands (no K^T LDS reads);
//   5 = dQ MFMAs + K^T reads, but no ds_add and no flush.
// SCRW: dS scratch row width (64 = conflict-free reads, 32 = half LDS).
//
// FUSE_DV=1 (requires FUSE_DQ=0): the 4-GEMM fused dK+dV kernel — adds
// P (kept beside dS), a streamed dO^T image, and a second accumulator:
//   dV^T? no — dV[kv][d] += mfma(P-frags [row=kv][contr=q],
//                                dO^T image [row=d][contr=q])
// P and dS share ONE per-wave scratch region sequentially (DS ops retire
// in order within a wave, so the write-read-write-read chain is safe
// without barriers).  This replaces BOTH split dkv kernels with 4 tile
// GEMMs instead of 5, atomic-free and deterministic — the measured
// verdict on gfx950 is that fp32 atomics (LDS ds_add ~2.6 s, global L2
// ~1.8 s at this density, each atomic dropping its L2 line) rule out
// flash-attn's atomic dq plan entirely (tools/dkq_probe.hip).
template <typename T, int D, int FUSE_DQ, int DBG = 0, int SCRW = 64,
          int FUSE_DV = 0>
__global__ __launch_bounds__(512) void bwd_dkq_kernel(
    const T* __restrict__ dout, const T* __restrict__ q,
    const T* __restrict__ k, const T* __restrict__ v,
    const float* __restrict__ delta, const float* __restrict__ lse,
    float* __restrict__ dq, float* __restrict__ dk, float* __restrict__ dv,
    int Sq, int Sk, int N,
    int64_t g_sb, int64_t g_ss, int64_t g_sh,
    int64_t q_sb, int64_t q_ss, int64_t q_sh,
    int64_t k_sb, int64_t k_ss, int64_t k_sh,
    int64_t v_sb, int64_t v_ss, int64_t v_sh,
    int64_t d_sb, int64_t d_sh, int64_t l_sb, int64_t l_sh,
    int64_t dq_sb, int64_t dq_ss, int64_t dq_sh,
    int64_t dk_sb, int64_t dk_ss, int64_t dk_sh,
    int64_t dv_sb, int64_t dv_ss, int64_t dv_sh,
    float scale, int causal) {
  static_assert(!(FUSE_DQ && FUSE_DV), "dq and dv fusions are exclusive");
  using MT = mfma_traits<T>;
  using frag = typename MT::frag;
  constexpr int QBLK = 32;    // streamed q rows per tile (1 subtile)
  constexpr int KVWG = 256;   // kv columns per workgroup (8 waves x 32)
  constexpr int NT = 512;
  constexpr int NIMG = 3 + FUSE_DV;         // Q, dO, Q^T [, dO^T]
  constexpr int SWZ = (D == 128) ? 15 : 7;  // Q/dO row-major images
  constexpr int SWZ_QT = 3;                 // Q^T image rows are 64 B
  constexpr int SWZ_SC = (SCRW == 64) ? 7 : 3;
  constexpr int CHUNKS = QBLK * (D / 8);    // staged 8-elem chunks per img

  // one LDS object (a second __shared__ forces vmcnt(0) per ds_read):
  // [2 buf x {Q row-major | dO row-major | Q^T | (dO^T)}] [K^T persistent]
  // [V row-major persistent (FUSE_DV — keeps the V fragments out of the
  // register file; kv-resident so staged once)] [8 x per-wave P/dS
  // scratch] [fp32 dq reduce buffer]
  constexpr int STREAM = 2 * NIMG * QBLK * D;
  constexpr int KT_E = FUSE_DQ ? D * KVWG : 0;
  constexpr int V_E = FUSE_DV ? D * KVWG : 0;
  constexpr int SCR_E = 8 * 32 * SCRW;
  constexpr int SCR2_E = FUSE_DV ? SCR_E : 0;  // separate P scratch
  constexpr int REDU_E = FUSE_DQ ? QBLK * D * (int)(4 / sizeof(T)) : 0;
  static_assert((STREAM + KT_E + V_E + SCR_E + SCR2_E + REDU_E) *
                        (int)sizeof(T) <=
                    163840,
                "LDS budget exceeded");
  __shared__ T lds[STREAM + KT_E + V_E + SCR_E + SCR2_E + REDU_E];
  auto ldsQ = [&](int buf) -> T* { return lds + buf * (NIMG * QBLK * D); };
  auto ldsG = [&](int buf) -> T* {
    return lds + buf * (NIMG * QBLK * D) + QBLK * D;
  };
  auto ldsQT = [&](int buf) -> T* {
    return lds + buf * (NIMG * QBLK * D) + 2 * QBLK * D;
  };
  auto ldsGT = [&](int buf) -> T* {  // FUSE_DV only
    return lds + buf * (NIMG * QBLK * D) + 3 * QBLK * D;
  };
  T* ldsKT = lds + STREAM;
  T* ldsV = lds + STREAM + KT_E;
  auto ldsSC = [&](int w) -> T* {
    return lds + STREAM + KT_E + V_E + w * 32 * SCRW;
  };
  auto ldsSC2 = [&](int w) -> T* {  // FUSE_DV: P scratch
    return lds + STREAM + KT_E + V_E + SCR_E + w * 32 * SCRW;
  };
  float* redu = (float*)(lds + STREAM + KT_E + V_E + SCR_E + SCR2_E);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31, hi = lane >> 5;
  const int n = blockIdx.x, b = blockIdx.z;     // head-first dispatch
  const int kvblk = blockIdx.y;
  const int kvb = kvblk * KVWG + wave * 32;     // wave's kv block
  const int kv_col = kvb + l31;

  const T* qp = q + b * q_sb + (int64_t)n * q_sh;
  const T* gp = dout + b * g_sb + (int64_t)n * g_sh;
  const T* kp = k + b * k_sb + (int64_t)n * k_sh;
  const T* vp = v + b * v_sb + (int64_t)n * v_sh;
  const float* dp_ = delta + b * d_sb + n * d_sh;
  const float* lp_ = lse + b * l_sb + n * l_sh;

  // resident K fragments [row=kv][contr=d] (A operand of S^T); V the
  // same way in registers UNLESS FUSE_DV (register budget: it moves to
  // the persistent LDS image instead)
  frag kf[D / 16], vf[FUSE_DV ? 1 : D / 16];
#pragma unroll
  for (int s = 0; s < D / 16; ++s) {
    if (kv_col < Sk) {
      kf[s] = __builtin_bit_cast(
          frag, *(const u32x4_t*)(kp + (int64_t)kv_col * k_ss + 16 * s + 8 * hi));
      if (!FUSE_DV)
        vf[s] = __builtin_bit_cast(
            frag,
            *(const u32x4_t*)(vp + (int64_t)kv_col * v_ss + 16 * s + 8 * hi));
    } else {
      u32x4_t z = {0, 0, 0, 0};
      kf[s] = __builtin_bit_cast(frag, z);
      if (!FUSE_DV) vf[s] = __builtin_bit_cast(frag, z);
    }
  }
  const float c2 = scale * BA_LOG2E;

  f32x16_t acc[D / 32];  // dK^T accumulator [d rows][kv col = lane]
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt) acc[dt] = (f32x16_t)(0.f);
  f32x16_t acc2[FUSE_DV ? D / 32 : 1];  // dV accumulator [kv rows][d col]
#pragma unroll
  for (int dt = 0; dt < (FUSE_DV ? D / 32 : 1); ++dt)
    acc2[dt] = (f32x16_t)(0.f);

  // one-time: zero the reduce buffer; stage K^T [D rows][KVWG] transposed
  if (FUSE_DQ) {
    for (int i = tid; i < QBLK * D; i += NT) redu[i] = 0.f;
    constexpr int PTK = (KVWG * D / 8) / NT;
#pragma unroll
    for (int c = 0; c < PTK; ++c) {
      const int flat = tid + c * NT;
      const int kv = flat / (D / 8), col8 = flat % (D / 8);
      const int kvg = kvblk * KVWG + kv;
      const int kvc = kvg < Sk ? kvg : (Sk - 1);
      u32x4_t ch = *(const u32x4_t*)(kp + (int64_t)kvc * k_ss + col8 * 8);
      ba_st_transposed<T, KVWG, 15, 0>(ldsKT, kv, col8 * 8, ch);
    }
  }
  // one-time (FUSE_DV): stage the workgroup's V block row-major
  if (FUSE_DV) {
    constexpr int PTV = (KVWG * D / 8) / NT;
#pragma unroll
    for (int c = 0; c < PTV; ++c) {
      const int flat = tid + c * NT;
      const int kv = flat / (D / 8), col8 = flat % (D / 8);
      const int kvg = kvblk * KVWG + kv;
      const int kvc = kvg < Sk ? kvg : (Sk - 1);
      u32x4_t ch = *(const u32x4_t*)(vp + (int64_t)kvc * v_ss + col8 * 8);
      const int byte = ba_swz<SWZ>(kv * (2 * D) + col8 * 16, kv);
      *(u32x4_t*)((char*)ldsV + byte) = ch;
    }
  }

  const int t0 = causal ? (kvblk * KVWG) / QBLK : 0;
  const int nt = (Sq + QBLK - 1) / QBLK;
  const int span = nt - t0;
  // non-causal stagger: spread same-head workgroups evenly over the
  // q-tile sequence (causal blocks are naturally staggered by t0)
  const int stag =
      (FUSE_DQ && !causal && gridDim.y > 1)
          ? (int)(((int64_t)kvblk * span) / gridDim.y)
          : 0;

  // per-tile prefetch: Q and dO chunks + this lane's lse*log2e and delta
  auto issue_loads = [&](int tile, u32x4_t* qreg, u32x4_t* greg, float* lse2,
                         float* dlt) {
    const int q0 = tile * QBLK;
    constexpr int PT = (CHUNKS + NT - 1) / NT;
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      if (CHUNKS % NT == 0 || flat < CHUNKS) {
        const int row = flat / (D / 8), col8 = flat % (D / 8);
        const int qg = q0 + row;
        const int qc = qg < Sq ? qg : (Sq - 1);  // clamped (see attn_fwd.hip)
        qreg[c] = *(const u32x4_t*)(qp + (int64_t)qc * q_ss + col8 * 8);
        greg[c] = *(const u32x4_t*)(gp + (int64_t)qc * g_ss + col8 * 8);
      }
    }
    const int qg = q0 + l31;
    const int qc = qg < Sq ? qg : (Sq - 1);
    *lse2 = lp_[qc] * BA_LOG2E;  // lane-local; invalid rows masked later
    *dlt = dp_[qc];
  };
  auto write_lds = [&](int buf, const u32x4_t* qreg, const u32x4_t* greg) {
    constexpr int PT = (CHUNKS + NT - 1) / NT;
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      if (CHUNKS % NT == 0 || flat < CHUNKS) {
        const int row = flat / (D / 8), col8 = flat % (D / 8);
        const int byte = ba_swz<SWZ>(row * (2 * D) + col8 * 16, row);
        *(u32x4_t*)((char*)ldsQ(buf) + byte) = qreg[c];
        *(u32x4_t*)((char*)ldsG(buf) + byte) = greg[c];
        ba_st_transposed<T, QBLK, SWZ_QT, 3>(ldsQT(buf), row, col8 * 8,
                                             qreg[c]);
        if (FUSE_DV)
          ba_st_transposed<T, QBLK, SWZ_QT, 3>(ldsGT(buf), row, col8 * 8,
                                               greg[c]);
      }
    }
  };

  float lse2_c, dlt_c;
  {
    u32x4_t qreg[(CHUNKS + NT - 1) / NT], greg[(CHUNKS + NT - 1) / NT];
    issue_loads(t0 + stag, qreg, greg, &lse2_c, &dlt_c);
    write_lds(0, qreg, greg);
    __syncthreads();
  }
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static form (younger half)

  int cur = 0;
  int t = t0 + stag;  // wrapped iteration order over [t0, nt)
  for (int i = 0; i < span; ++i) {
    const int q0 = t * QBLK;
    int tn = t + 1;
    if (tn >= nt) tn = t0;
    const bool has_next = (i + 1) < span;
    u32x4_t qreg[(CHUNKS + NT - 1) / NT], greg[(CHUNKS + NT - 1) / NT];
    float lse2_n = 0.f, dlt_n = 0.f;
    if (has_next) issue_loads(tn, qreg, greg, &lse2_n, &dlt_n);

    // waves whose kv block is wholly after this q tile are masked anyway
    const bool active = !causal || (q0 + QBLK - 1 >= kvb);
    if (active) {
      // ---- S^T = mfma(K, Q), dP^T = mfma(V, dO): [kv rows][q cols]
      f32x16_t st = (f32x16_t)(0.f), dpt = (f32x16_t)(0.f);
#pragma unroll
      for (int s = 0; s < D / 16; ++s) {
        frag qfr = ba_ld_rowslice<T, D, SWZ>(ldsQ(cur), l31, 16 * s + 8 * hi);
        frag gfr = ba_ld_rowslice<T, D, SWZ>(ldsG(cur), l31, 16 * s + 8 * hi);
        frag vop;
        if (FUSE_DV)
          vop = ba_ld_rowslice<T, D, SWZ>(ldsV, wave * 32 + l31,
                                          16 * s + 8 * hi);
        else
          vop = vf[s];
        st = MT::mma(kf[s], qfr, st);
        dpt = MT::mma(vop, gfr, dpt);
      }
      // ---- dS^T = p * (dP^T - delta) * scale, lse/delta LANE-LOCAL.
      // FUSE_DV streams BOTH P and dS straight into their own per-wave
      // scratch regions inside this loop (no register homes, no
      // write-read-write serialisation): the dV/dK fragments read them
      // back transposed.
      T* sc = ldsSC(wave);
      T* sc2 = ldsSC2(wave);
      auto scr_byte = [&](int r) {
        const int row = ba_crow(r, 0) + 4 * hi;  // kv-local
        return ba_swz<SWZ_SC, 0>(row * (2 * SCRW) + 2 * l31, row);
      };
      const int q_g = q0 + l31;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv_g = kvb + ba_crow(r, 0) + 4 * hi;
        const bool valid =
            q_g < Sq && kv_g < Sk && (!causal || q_g >= kv_g);
        const float e = valid ? __builtin_fmaf(st[r], c2, -lse2_c) : BA_NEG_BIG;
        const float p = ba_exp2(e);
        if (FUSE_DV) *(T*)((char*)sc2 + scr_byte(r)) = (T)p;
        st[r] = p * (dpt[r] - dlt_c) * scale;
      }
      auto scr_put = [&](const f32x16_t& vv) {
#pragma unroll
        for (int r = 0; r < 16; ++r)
          *(T*)((char*)sc + scr_byte(r)) = (T)vv[r];
      };
      // ---- per-wave scratch transpose: dS^T D-layout -> dS fragments
      // (own region — the dV section's P reads don't serialise against
      // these writes)
      scr_put(st);
      if (FUSE_DQ && DBG != 3) {
        // ---- dQ: fragments [row=q][contr=kv] in-register; one dt tile
        // at a time to keep the partial's register span at 16
        frag dsT[2];
        ba_build_frag_pair<T>(st, dsT);
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt) {
          f32x16_t dqp = (f32x16_t)(0.f);
#pragma unroll
          for (int u = 0; u < 2; ++u) {
            frag ktf;
            if (DBG == 4)  // perf probe: no K^T LDS reads
              ktf = dsT[u];
            else
              ktf = ba_ld_rowslice<T, KVWG, 15, 0>(
                  ldsKT, dt * 32 + l31, wave * 32 + 16 * u + 8 * hi);
            dqp = MT::mma(dsT[u], ktf, dqp);
          }
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int q_loc = ba_crow(r, 0) + 4 * hi;
            if (DBG == 5) {         // perf probe: no LDS reduce at all
              if (dqp[r] == 1234.5678f) redu[0] = dqp[r];  // keep dqp live
            } else if (DBG == 2)    // perf probe: write, not LDS-atomic
              redu[q_loc * D + dt * 32 + l31] = dqp[r];
            else
              atomicAdd(&redu[q_loc * D + dt * 32 + l31], dqp[r]);  // ds_add
          }
        }
      }
      if (FUSE_DV) {
        // ---- dV += mfma(P, dO^T): P read back from its own scratch
        frag pf[2];
        pf[0] = ba_ld_rowslice<T, SCRW, SWZ_SC, 0>(sc2, l31, 8 * hi);
        pf[1] = ba_ld_rowslice<T, SCRW, SWZ_SC, 0>(sc2, l31, 16 + 8 * hi);
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
          for (int u = 0; u < 2; ++u) {
            frag gt = ba_ld_rowslice<T, QBLK, SWZ_QT, 3>(
                ldsGT(cur), dt * 32 + l31, 16 * u + 8 * hi);
            acc2[dt] = MT::mma(pf[u], gt, acc2[dt]);
          }
        }
      }
      // ---- dK^T += mfma(Q^T, dS)
      frag dsf[2];
      dsf[0] = ba_ld_rowslice<T, SCRW, SWZ_SC, 0>(sc, l31, 8 * hi);
      dsf[1] = ba_ld_rowslice<T, SCRW, SWZ_SC, 0>(sc, l31, 16 + 8 * hi);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
        for (int u = 0; u < 2; ++u) {
          frag tf = ba_ld_rowslice<T, QBLK, SWZ_QT, 3>(
              ldsQT(cur), dt * 32 + l31, 16 * u + 8 * hi);
          acc[dt] = MT::mma(tf, dsf[u], acc[dt]);
        }
      }
    }
    if (!FUSE_DQ) {
      if (has_next) write_lds(cur ^ 1, qreg, greg);
      __syncthreads();
    } else {
      __syncthreads();  // A: every wave's ds_adds for this tile landed
      // flush the reduced dq partial: ONE atomic per element per
      // workgroup (the ds_adds above summed the 8 waves' 32-kv
      // partials).  Issued BEFORE write_lds so the staging's counted
      // vmcnt wait does not drain the fire-and-forget atomics.
      const int base = tid * (QBLK * D / NT);  // 8 consecutive floats
      const int q_loc = base / D, d0 = base % D;
      const int q_g2 = q0 + q_loc;
      if (q_g2 < Sq && DBG != 1 && DBG != 5) {
        float* gdst = dq + b * dq_sb + (int64_t)q_g2 * dq_ss + n * dq_sh + d0;
#pragma unroll
        for (int j = 0; j < QBLK * D / NT; ++j) {
          unsafeAtomicAdd(&gdst[j], redu[base + j]);
          redu[base + j] = 0.f;
        }
      } else {
#pragma unroll
        for (int j = 0; j < QBLK * D / NT; ++j) redu[base + j] = 0.f;
      }
      if (has_next) write_lds(cur ^ 1, qreg, greg);
      __syncthreads();  // B: redu zeroed + staging visible
    }
    cur ^= 1;
    lse2_c = lse2_n;
    dlt_c = dlt_n;
    t = tn;
  }

  if (kv_col < Sk) {
    float* row = dk + b * dk_sb + (int64_t)kv_col * dk_ss + n * dk_sh;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        row[dt * 32 + ba_crow(r, hi)] += acc[dt][r];
  }
  if (FUSE_DV) {
    // dV accumulator sits [kv rows][d col = lane]: per (r, dt) the 32
    // lanes of a half write one kv row's 128 B contiguously
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv_g = kvb + ba_crow(r, 0) + 4 * hi;
      if (kv_g < Sk) {
        float* row2 = dv + b * dv_sb + (int64_t)kv_g * dv_ss + n * dv_sh;
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt)
          row2[dt * 32 + l31] += acc2[dt][r];
      }
    }
  }
}

}  // namespace

// ====================== launchers =====================================
extern "C" int bahip_attn_bwd_preprocess(
    const void* o, const void* dout, float* delta, int64_t B, int64_t S,
    int64_t N, int64_t D, const int64_t o_strides[3],
    const int64_t do_strides[3], int o_dtype, int do_dtype, void* stream) {
  if (o_dtype != do_dtype) return 1003;
  dim3 grid((unsigned)((S + 31) / 32), (unsigned)N, (unsigned)B);
#define LAUNCH_PRE(T, DD)                                                   \
  bwd_preprocess_kernel<T, DD><<<grid, 256, 0, (hipStream_t)stream>>>(      \
      (const T*)o, (const T*)dout, delta, (int)S, (int)N, o_strides[0],     \
      o_strides[1], o_strides[2], do_strides[0], do_strides[1],             \
      do_strides[2])
  if (D == 128 && o_dtype == BAHIP_BF16) LAUNCH_PRE(__bf16, 128);
  else if (D == 128 && o_dtype == BAHIP_F16) LAUNCH_PRE(_Float16, 128);
  else if (D == 64 && o_dtype == BAHIP_BF16) LAUNCH_PRE(__bf16, 64);
  else if (D == 64 && o_dtype == BAHIP_F16) LAUNCH_PRE(_Float16, 64);
  else return 1002;
#undef LAUNCH_PRE
  BA_CHECK_LAUNCH();
  return 0;
}

// plan selection (all plans accumulate into dq/dk/dv; plans 0 and 1 are
// bitwise deterministic — the `deterministic` flag is honoured by
// construction):
//   plan 0 "split8" (DEFAULT): dq kernel + split dV (MODE 0) + split dK
//     (MODE 1) — 8 tile GEMMs, two-pass dq, atomic-free.  Fastest
//     measured plan on gfx950 (tools/dkq_probe.hip round 2).
//     BA_BWD_DK_FLIP=1 swaps MODE 1 for the flipped FUSE_DQ=0 kernel
//     (measured equal, 142 vs 137 ms at s=65536).
//   plan 1 "dkvf7" (BA_BWD_FUSED=1): dq kernel + fused flipped dK+dV —
//     7 tile GEMMs, atomic-free, but the QBLK=32 single-workgroup
//     structure + P/dS scratch round trips measured SLOWER than split8
//     (260 vs 230 ms for the dK+dV part at s=65536); kept as the
//     documented experiment.
//   plan 2 "atomic6" (BA_BWD_FUSED=2): flash-attn's plan — fused dK+dQ
//     with fp32 atomic dq + split dV.  6 GEMMs but MEASURED DEAD on
//     gfx950 (tools/dkq_probe.hip: LDS ds_add ~2.6 s, global fp32
//     atomics ~1.8 s at this density — atomics drop their L2 line);
//     kept for the record and for re-testing on future silicon.
template <typename T, int D>
static int launch_bwd(const void* dout, const void* q, const void* k,
                      const void* v, const float* delta, const float* lse,
                      float* dq, float* dk, float* dv, int64_t B, int64_t Sq,
                      int64_t Sk, int64_t N, const int64_t* gs,
                      const int64_t* qs, const int64_t* ks, const int64_t* vs,
                      const int64_t* ds, const int64_t* ls,
                      const int64_t* dqs, const int64_t* dks,
                      const int64_t* dvs, float scale, int causal,
                      int deterministic, void* stream) {
  // read per call (cheap at one bwd invocation) so tests can flip plans
  const char* ef = getenv("BA_BWD_FUSED");
  const int env_fused = ef ? atoi(ef) : -1;
  const char* df = getenv("BA_BWD_DK_FLIP");
  const int dk_flip = df ? atoi(df) : 0;
  const char* dbg = getenv("BA_DKQ_DBG");
  const int dkq_dbg = dbg ? atoi(dbg) : 0;
  (void)deterministic;  // plans 0/1 are deterministic by construction
  const int plan = env_fused >= 0 ? env_fused : 0;
  dim3 grid_kv((unsigned)((Sk + 255) / 256), (unsigned)N, (unsigned)B);
  // head-first grid for the dkq/dkvf kernels
  dim3 grid_dkq((unsigned)N, (unsigned)((Sk + 255) / 256), (unsigned)B);
  dim3 grid_dq((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B);

#define DKQ_ARGS                                                              \
  (const T*)dout, (const T*)q, (const T*)k, (const T*)v, delta, lse, dq, dk, \
      dv, (int)Sq, (int)Sk, (int)N, gs[0], gs[1], gs[2], qs[0], qs[1],        \
      qs[2], ks[0], ks[1], ks[2], vs[0], vs[1], vs[2], ds[0], ds[1], ls[0],   \
      ls[1], dqs[0], dqs[1], dqs[2], dks[0], dks[1], dks[2], dvs[0], dvs[1],  \
      dvs[2], scale, causal

  const char* dqp_e = getenv("BA_DQ_PIPE");
  const int dqp = dqp_e ? atoi(dqp_e) : 0;
  if (plan != 2) {
#define DQ_ARGS                                                               \
  (const T*)dout, (const T*)q, (const T*)k, (const T*)v, delta, lse, dq,      \
      (int)Sq, (int)Sk, (int)N, gs[0], gs[1], gs[2], qs[0], qs[1], qs[2],     \
      ks[0], ks[1], ks[2], vs[0], vs[1], vs[2], ds[0], ds[1], ls[0], ls[1],   \
      dqs[0], dqs[1], dqs[2], scale, causal
    if (dqp)
      bwd_dq_kernel<T, D, 1><<<grid_dq, 512, 0, (hipStream_t)stream>>>(DQ_ARGS);
    else
      bwd_dq_kernel<T, D, 0><<<grid_dq, 512, 0, (hipStream_t)stream>>>(DQ_ARGS);
#undef DQ_ARGS
    BA_CHECK_LAUNCH();
  }
  if (plan == 1) {
    bwd_dkq_kernel<T, D, 0, 0, 32, 1>
        <<<grid_dkq, 512, 0, (hipStream_t)stream>>>(DKQ_ARGS);
    BA_CHECK_LAUNCH();
    return 0;
  }
  bwd_dkv_kernel<T, D, 0, 64><<<grid_kv, 512, 0, (hipStream_t)stream>>>(
      (const T*)dout, (const T*)q, (const T*)k, (const T*)v, delta, lse, dv,
      nullptr, (int)Sq, (int)Sk, (int)N, gs[0], gs[1], gs[2], qs[0], qs[1],
      qs[2], ks[0], ks[1], ks[2], vs[0], vs[1], vs[2], ds[0], ds[1], ls[0],
      ls[1], dvs[0], dvs[1], dvs[2], scale, causal);
  BA_CHECK_LAUNCH();
  if (plan == 2) {
    if (dkq_dbg == 1)
      bwd_dkq_kernel<T, D, 1, 1>
          <<<grid_dkq, 512, 0, (hipStream_t)stream>>>(DKQ_ARGS);
    else if (dkq_dbg == 2)
      bwd_dkq_kernel<T, D, 1, 2>
          <<<grid_dkq, 512, 0, (hipStream_t)stream>>>(DKQ_ARGS);
    else
      bwd_dkq_kernel<T, D, 1, 0>
          <<<grid_dkq, 512, 0, (hipStream_t)stream>>>(DKQ_ARGS);
    BA_CHECK_LAUNCH();
  } else if (dk_flip) {
    bwd_dkq_kernel<T, D, 0><<<grid_dkq, 512, 0, (hipStream_t)stream>>>(DKQ_ARGS);
    BA_CHECK_LAUNCH();
  } else {
    bwd_dkv_kernel<T, D, 1><<<grid_kv, 512, 0, (hipStream_t)stream>>>(
        (const T*)dout, (const T*)q, (const T*)k, (const T*)v, delta, lse,
        dk, nullptr, (int)Sq, (int)Sk, (int)N, gs[0], gs[1], gs[2], qs[0],
        qs[1], qs[2], ks[0], ks[1], ks[2], vs[0], vs[1], vs[2], ds[0], ds[1],
        ls[0], ls[1], dks[0], dks[1], dks[2], scale, causal);
    BA_CHECK_LAUNCH();
  }
#undef DKQ_ARGS
  return 0;
}

extern "C" int bahip_attn_bwd(
    const void* dout, const void* q, const void* k, const void* v,
    const float* delta, const float* lse, float* dq, float* dk, float* dv,
    int64_t B, int64_t Sq, int64_t Sk, int64_t N, int64_t D,
    const int64_t do_strides[3], const int64_t q_strides[3],
    const int64_t k_strides[3], const int64_t v_strides[3],
    const int64_t delta_strides[2], const int64_t lse_strides[2],
    const int64_t dq_strides[3], const int64_t dk_strides[3],
    const int64_t dv_strides[3], float softmax_scale, int causal,
    int deterministic, int dtype, void* stream) {
  if (causal && Sq != Sk) return 1001;
#define LAUNCH_BWD(T, DD)                                                     \
  launch_bwd<T, DD>(dout, q, k, v, delta, lse, dq, dk, dv, B, Sq, Sk, N,      \
                    do_strides, q_strides, k_strides, v_strides,              \
                    delta_strides, lse_strides, dq_strides, dk_strides,       \
                    dv_strides, softmax_scale, causal, deterministic, stream)
  if (D == 128 && dtype == BAHIP_BF16) return LAUNCH_BWD(__bf16, 128);
  if (D == 128 && dtype == BAHIP_F16) return LAUNCH_BWD(_Float16, 128);
  if (D == 64 && dtype == BAHIP_BF16) return LAUNCH_BWD(__bf16, 64);
  if (D == 64 && dtype == BAHIP_F16) return LAUNCH_BWD(_Float16, 64);
#undef LAUNCH_BWD
  return 1002;
}
