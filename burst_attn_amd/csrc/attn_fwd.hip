// BurstAttention forward tile kernel for gfx950 (MI355X, CDNA4).
//
// Computes one flash-attention tile o = softmax(q k^T * scale) v plus its
// log-sum-exp — the role the flash-attn CUDA extension plays for the
// reference at burst_attn/burst_utils.py:149-177 — as a single hand-written
// HIP kernel (online softmax in the exp2 domain, fused, no S matrix in HBM).
//
// Structure (see attn_common.h header comment for the operand scheme):
//   * workgroup = 8 waves (512 threads); wave w owns q rows
//     [blk*256 + w*32, +32); grid = (ceil(Sq/256), N, B).
//   * kv tiles of KVBLK=64 rows staged in LDS, double-buffered, one barrier
//     per tile; next tile's global loads are issued before the current
//     tile's MFMAs (async-stage split).  K is stored row-major [kv][D]
//     (XOR-swizzled); V is stored TRANSPOSED [D][kv] at staging time so
//     that both operands' MFMA fragments are plain 16B row-slice
//     ds_read_b128s — no per-element transposed gathers (those made the
//     compiler hoist & spill hundreds of swizzled addresses).
//   * per kv tile and wave: S^T = mfma(K, Q) (2 x D/16 MFMAs),
//     online-softmax update (lane-local m/l in exp2 domain),
//     P->fragments in-register (pack + permlane32_swap),
//     O^T += mfma(V^T, P^T) (2 x 2 x D/32 MFMAs).
//   * epilogue: o = O^T / l (fp32 out), lse = ln2*(m2 + log2(l)).

#include "attn_common.h"
#include "../../include/burst_attn_hip.h"

#include <stdio.h>
#include <stdlib.h>

namespace {


// OUT_STATE=0: write normalised o (fp32) + lse — the stateless tile.
// OUT_STATE=1: carry-in/carry-out accumulator state (acc = unnormalised
// O, m = running max in the exp2 domain, l = running sum) — the fused
// in-kernel merge replacing the reference's per-round
// cuda_scale_out_lse_helper pass (burst_utils.py:20-33; lao.py's
// carry-in design, lao.py:108-114).
// VPATH: 0 = V transposed image + b128 row-slice reads;
//         1 = V row-major (tr16-swizzled) + ds_read_tr16_b64 fragments
// SUBT: 0 = joint softmax over the 64-kv tile; 1 = per-32-subtile online
// updates (lets subtile-0 PV MFMAs overlap subtile-1 QK/softmax);
// 2 = att[2] double-pipeline (guide T15): the softmax FINISH (mask, max,
// rescale, exp2, pack) and PV of subtile j-1 are issued while subtile
// j's QK MFMAs fill — an explicit two-stage software pipeline carried
// ACROSS tile boundaries (requires NBUF=4 so the previous tile's V
// image survives one extra subtile, and a barrier every tile);
// 3 = THREE-stage pipeline at ONE WAVE PER SIMD (worksheet-driven,
// profiles/r02/mfma_pipe_worksheet.txt): phase k issues QK(k) MFMAs,
// the softmax of subtile k-1 (producing the P fragments), and PV(k-2)
// — three independent chains per wave, K/V fragments prefetched a full
// phase ahead into registers.  Requires NT=256 and NBUF=4 (128 KB LDS
// forces one 4-wave workgroup per CU = 1 wave/SIMD, where the ~500
// register budget holds the pipeline state).  The ot rescale of
// softmax(k-1) is ordered AFTER PV(k-2) lands (scale consistency).
// NT: threads per workgroup (512 = 8 waves x 1 block/CU;
//     256 = 4 waves x 2 blocks/CU — decoupled barrier groups)
// NBUF: LDS tile buffers. 2 = stage one tile ahead, barrier every tile.
// 4 = stage two tiles ahead, barrier every OTHER tile (a buffer is
// rewritten two tiles after its last read, so one barrier in any two
// consecutive tile boundaries separates every write from its readers
// and every reader from the overwrite).
// KREG: 1 = K tiles go straight from HBM into the MFMA fragment
// registers (no LDS round trip): the QK cluster is PURE-REG (guide T16's
// load-K->reg cluster), and tile t+1's K loads are issued the moment
// tile t's QK MFMAs consume the block (WAR on the same registers), so
// softmax+PV+staging+barrier (~700+ cycles) cover the HBM latency.
// Only the V^T image stays in LDS.  Requires SUBT=1, NBUF=2, VPATH=0.
// (Measured -47%: the per-lane strided K loads swamp vmem issue.)
// KREG=2 = cross-barrier K-FRAGMENT PREFETCH: one 8-fragment register
// set is re-filled from LDS a full SUBTILE ahead of its QK use — legal
// across tile boundaries because NBUF=4 staging (AHEAD=2) wrote tile
// t+1's buffer two tiles ago and a barrier every tile orders it.  The
// QK cluster's lgkm park (SQ_WAIT_ANY 38% in the r2 PMC taxonomy)
// becomes overlap.  Requires SUBT=1, NBUF=4, VPATH=0.
template <int N>
struct ba_ic {
  static constexpr int value = N;
};
struct ba_dyn {
  int value;
};
template <int N>
__device__ __forceinline__ int ba_buf_of(ba_ic<N>) { return N; }
__device__ __forceinline__ int ba_buf_of(ba_dyn d) { return d.value; }

template <typename T, int D, int KVBLK, int OUT_STATE, int VPATH, int SUBT,
          int NT = 512, int NBUF = 2, int KREG = 0>
__global__ __launch_bounds__(NT) void attn_fwd_kernel(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    float* __restrict__ o, float* __restrict__ lse,
    int Sq, int Sk, int N,
    int64_t q_sb, int64_t q_ss, int64_t q_sh,
    int64_t k_sb, int64_t k_ss, int64_t k_sh,
    int64_t v_sb, int64_t v_ss, int64_t v_sh,
    float scale, int causal,
    // state (OUT_STATE=1): acc [B,Sq,N,D] strided, m/l [B,N,Sq] strided
    float* __restrict__ st_acc, float* __restrict__ st_m,
    float* __restrict__ st_l,
    int64_t a_sb, int64_t a_ss, int64_t a_sh,
    int64_t ml_sb, int64_t ml_sh, int carry_in) {
  using MT = mfma_traits<T>;
  using frag = typename MT::frag;
  constexpr int SWZ_K = (D == 128) ? 15 : 7;  // K image rows are 2*D bytes
  constexpr int SWZ_V = 7;                    // V^T image rows are 128 bytes
  constexpr int PT = (KVBLK * D / 8) / NT;
  constexpr int QROWS = NT / 2;  // q rows per workgroup (32 per wave)
  static_assert(PT >= 1, "tile must fill at least one chunk per thread");

  static_assert(KREG != 1 || (SUBT == 1 && NBUF == 2 && VPATH == 0),
                "KREG=1 path: SUBT=1, NBUF=2, V^T image only");
  static_assert(KREG != 2 || (SUBT == 1 && NBUF == 4 && VPATH == 0),
                "KREG=2 path: SUBT=1, NBUF=4 (cross-barrier prefetch)");
  // single LDS object: [NBUF buffers][K row-major | V transposed][KVBLK*D]
  // (KREG=1: V transposed only)
  constexpr int IMGS_F = (KREG == 1) ? 1 : 2;
  __shared__ T lds[NBUF * IMGS_F * KVBLK * D];
  auto ldsK = [&](int buf) -> T* { return lds + buf * (IMGS_F * KVBLK * D); };
  auto ldsVT = [&](int buf) -> T* {
    return lds + buf * (IMGS_F * KVBLK * D) + (KREG == 1 ? 0 : KVBLK * D);
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int l31 = lane & 31;
  const int hi = lane >> 5;
  const int n = blockIdx.y;
  const int b = blockIdx.z;
  const int qb = blockIdx.x * QROWS + wave * 32;
  const int q_row = qb + l31;

  const T* qp = q + (int64_t)b * q_sb + (int64_t)n * q_sh;
  const T* kp = k + (int64_t)b * k_sb + (int64_t)n * k_sh;
  const T* vp = v + (int64_t)b * v_sb + (int64_t)n * v_sh;

  // Q fragments: B-operand of S^T = mfma(K, Q); lane holds q row q_row,
  // elements d = 16*s + 8*hi + j  (8 contiguous -> one 16B load)
  frag qf[D / 16];
#pragma unroll
  for (int s = 0; s < D / 16; ++s) {
    if (q_row < Sq) {
      const T* src = qp + (int64_t)q_row * q_ss + 16 * s + 8 * hi;
      qf[s] = __builtin_bit_cast(frag, *(const u32x4_t*)src);
    } else {
      u32x4_t z = {0, 0, 0, 0};
      qf[s] = __builtin_bit_cast(frag, z);
    }
  }

  const float c2 = scale * BA_LOG2E;  // exp2-domain scale
  float m2 = BA_NEG_BIG;
  float lsum = 0.f;
  f32x16_t ot[D / 32];
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt) ot[dt] = (f32x16_t)(0.f);
  if (OUT_STATE && carry_in && q_row < Sq) {
    m2 = st_m[b * ml_sb + n * ml_sh + q_row];
    lsum = st_l[b * ml_sb + n * ml_sh + q_row];
    const float* arow = st_acc + b * a_sb + (int64_t)q_row * a_ss + n * a_sh;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r) ot[dt][r] = arow[dt * 32 + ba_crow(r, hi)];
  }

  const int kv_limit =
      causal ? min(Sk, (int)(blockIdx.x + 1) * QROWS) : Sk;
  const int nt = (kv_limit + KVBLK - 1) / KVBLK;
  static_assert(SUBT != 2 || (NBUF == 4 && VPATH == 0),
                "the T15 pipeline needs NBUF=4 and the V^T image");

  // ---- SUBT=2 pipeline state: the previous subtile's raw scores and
  // its metadata, finished during the next subtile's QK cluster
  f32x16_t stP;
  int p_kv0 = -1;   // -1 = pipeline empty
  int p_cur = 0;
  bool p_full = false;
  constexpr float DEFER_THR2 = 8.f;
  auto finish_subtile = [&]() {
    if (!p_full) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv_g = p_kv0 + ba_crow(r, 0) + 4 * hi;
        if (!(kv_g < Sk && (!causal || kv_g <= q_row))) stP[r] = BA_NEG_BIG;
      }
    }
    float tm = ba_max16(stP);
    tm = fmaxf(tm, __shfl_xor(tm, 32));
    tm *= c2;
    if (!__all(tm - m2 <= DEFER_THR2)) {
      const float mnew = fmaxf(m2, tm);
      const float alpha = ba_exp2(m2 - mnew);
      m2 = mnew;
      lsum *= alpha;
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r) ot[dt][r] *= alpha;
    }
    float rowsum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      stP[r] = ba_exp2(__builtin_fmaf(stP[r], c2, -m2));
      rowsum += stP[r];
    }
    rowsum += __shfl_xor(rowsum, 32);
    lsum += rowsum;
    frag pfp[2];
    ba_build_frag_pair<T>(stP, pfp);
    const int kvs_l = (p_kv0 / 32) & 1;  // subtile index within its tile
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt) {
      const int drow = dt * 32 + l31;
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        frag vv = ba_ld_rowslice<T, KVBLK, SWZ_V, 7>(
            ldsVT(p_cur), drow, kvs_l * 32 + 16 * u + 8 * hi);
        ot[dt] = MT::mma(vv, pfp[u], ot[dt]);
      }
    }
  };

  // ---- SUBT=3 additional state: stage 2 (P fragments of subtile k-2
  // awaiting PV) and the register-prefetched K/V operands
  static_assert(SUBT != 3 || (NBUF == 4 && VPATH == 0 && NT == 256),
                "SUBT=3: NT=256 (1 wave/SIMD), NBUF=4, V^T image");
  frag pvf[2];                     // P fragments awaiting PV
  frag kfP[SUBT == 3 ? D / 16 : 1];  // prefetched K frags (QK of phase k)
  frag vvP[SUBT == 3 ? D / 16 : 1];  // prefetched V rowslices (PV operands)
  int v_kv0 = -1;                  // -1 = stage 2 empty

  // one SUBT=3 pipeline phase for subtile (t, kvs) using buffer `cur3`:
  // QK(k) on prefetched kfP -> kf prefetch for k+1 -> softmax part 1 of
  // stP (mask+max) -> PV(k-2) on prefetched vvP -> softmax part 2
  // (rescale ordered AFTER the PV, exp2, rowsum, pack -> pvf) -> vv
  // prefetch for the new pf -> rotate stage 1.
  // `steady` folds the pipeline-occupancy guards away at the two
  // steady-state call sites (t >= 1: both stages provably occupied) —
  // round-3 prerequisite: the steady loop must be straight-line for the
  // hand-scheduled .s body
  auto phase3 = [&](bool steady, int t, int kvs, int cur3, bool tile_full) {
    // --- QK(k) on the prefetched K fragments, then re-fill them for
    // the next subtile (WAR on kfP; a full phase of flight).
    // NOTE (measured): pinning this cadence with per-op
    // volatile asm does NOT help — the accumulators cross the C/asm
    // boundary through 100-470 register copies per phase and the C
    // softmax is not gap-packed.  The plain form below is the honest
    // compiler-scheduled 3-stage pipeline; the hand-placed version is
    // the round-3 full-asm body.
    f32x16_t stQ = (f32x16_t)(0.f);
#pragma unroll
    for (int s = 0; s < D / 16; ++s) stQ = MT::mma(kfP[s], qf[s], stQ);
    {
      const int nt_t = (kvs == 0) ? t : t + 1;
      const int nkvs = kvs ^ 1;
      if (nt_t < nt) {
#pragma unroll
        for (int s = 0; s < D / 16; ++s)
          kfP[s] = ba_ld_rowslice<T, D, SWZ_K>(ldsK(nt_t % NBUF),
                                               nkvs * 32 + l31,
                                               16 * s + 8 * hi);
      }
    }
    // --- softmax part 1 on stP: mask + row max (no ot/m2 writes)
    float tm = BA_NEG_BIG;
    if (steady || p_kv0 >= 0) {
      if (!p_full) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv_g = p_kv0 + ba_crow(r, 0) + 4 * hi;
          if (!(kv_g < Sk && (!causal || kv_g <= q_row))) stP[r] = BA_NEG_BIG;
        }
      }
      tm = ba_max16(stP);
      tm = fmaxf(tm, __shfl_xor(tm, 32));
      tm *= c2;
    }
    // --- PV(k-2) on the prefetched operands
    if (steady || v_kv0 >= 0) {
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
        for (int u = 0; u < 2; ++u)
          ot[dt] = MT::mma(vvP[dt * 2 + u], pvf[u], ot[dt]);
      }
    }
    // --- softmax part 2: rescale (safe now: the PV above has landed),
    // exp2, rowsum, pack; then prefetch the new pf's V operands
    if (steady || p_kv0 >= 0) {
      if (!__all(tm - m2 <= DEFER_THR2)) {
        const float mnew = fmaxf(m2, tm);
        const float alpha = ba_exp2(m2 - mnew);
        m2 = mnew;
        lsum *= alpha;
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) ot[dt][r] *= alpha;
      }
      float rowsum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        stP[r] = ba_exp2(__builtin_fmaf(stP[r], c2, -m2));
        rowsum += stP[r];
      }
      rowsum += __shfl_xor(rowsum, 32);
      lsum += rowsum;
      ba_build_frag_pair<T>(stP, pvf);
      v_kv0 = p_kv0;
      const int kvs_l = (p_kv0 / 32) & 1;
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
        for (int u = 0; u < 2; ++u)
          vvP[dt * 2 + u] = ba_ld_rowslice<T, KVBLK, SWZ_V, 7>(
              ldsVT(p_cur), dt * 32 + l31, kvs_l * 32 + 16 * u + 8 * hi);
      }
    }
    // --- rotate stage 1
    stP = stQ;
    p_kv0 = t * KVBLK + kvs * 32;
    p_cur = cur3;
    p_full = tile_full;
  };

  // staging lane offsets are loop-invariant; computing them once turns an
  // in-range tile's addresses into (scalar tile base) + (u32 lane offset),
  // which the compiler emits as saddr-form global_loads — the per-tile
  // 64-bit address recompute (~32 VALU ops) leaves the steady loop, and
  // the issue port is the oversubscribed resource there (ACTIVE+STALL
  // ~124% of SIMD issue slots at 2 waves, profiles/r02/sq_wait_taxonomy)
  uint32_t k_loff[PT], v_loff[PT];
#pragma unroll
  for (int c = 0; c < PT; ++c) {
    const int flat = tid + c * NT;
    const int row = flat / (D / 8);
    const int col8 = flat % (D / 8);
    k_loff[c] = (uint32_t)((row * k_ss + col8 * 8) * (int64_t)sizeof(T));
    v_loff[c] = (uint32_t)((row * v_ss + col8 * 8) * (int64_t)sizeof(T));
  }
  const bool stage_lin =
      (int64_t)(KVBLK - 1) * k_ss * (int64_t)sizeof(T) + 16 <= 0xffffffffLL &&
      (int64_t)(KVBLK - 1) * v_ss * (int64_t)sizeof(T) + 16 <= 0xffffffffLL;
  auto issue_loads = [&](int tile, u32x4_t* kreg, u32x4_t* vreg) {
    const int kv0 = tile * KVBLK;
    // the fast path's invariant offsets cost live registers the SUBT=2/3
    // pipelined variants cannot spare (they spill); production SUBT=1 only
    if (SUBT == 1 && stage_lin && kv0 + KVBLK <= Sk) {
      const char* kb = (const char*)(kp + (int64_t)kv0 * k_ss);
      const char* vb = (const char*)(vp + (int64_t)kv0 * v_ss);
#pragma unroll
      for (int c = 0; c < PT; ++c) {
        if (KREG != 1) kreg[c] = *(const u32x4_t*)(kb + k_loff[c]);
        vreg[c] = *(const u32x4_t*)(vb + v_loff[c]);
      }
      return;
    }
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      const int row = flat / (D / 8);
      const int col8 = flat % (D / 8);
      // clamped index instead of a guarded load: the guard compiles to
      // an exec-mask branch around every load pair (serialising the
      // staging); clamped rows re-load row Sk-1, whose values are
      // nullified by the kv-range mask (p = 0) in the softmax
      const int kvg = kv0 + row;
      const int kvc = kvg < Sk ? kvg : (Sk - 1);
      if (KREG != 1)
        kreg[c] = *(const u32x4_t*)(kp + (int64_t)kvc * k_ss + col8 * 8);
      vreg[c] = *(const u32x4_t*)(vp + (int64_t)kvc * v_ss + col8 * 8);
    }
  };
  auto write_lds = [&](int buf, const u32x4_t* kreg, const u32x4_t* vreg) {
#pragma unroll
    for (int c = 0; c < PT; ++c) {
      const int flat = tid + c * NT;
      const int row = flat / (D / 8);
      const int col8 = flat % (D / 8);
      if (KREG != 1) {
        const int byte = ba_swz<SWZ_K>(row * (2 * D) + col8 * 16, row);
        *(u32x4_t*)((char*)ldsK(buf) + byte) = kreg[c];
      }
      if (VPATH == 0)
        ba_st_transposed<T, KVBLK, SWZ_V, 7>(ldsVT(buf), row, col8 * 8, vreg[c]);
      else
        ba_st_tr16row<T, D>(ldsVT(buf), row, col8 * 8, vreg[c]);
    }
  };
  // KREG: one K subtile straight into the A-fragment registers (lane
  // reads its own kv row; clamped like the staging)
  auto load_k_sub = [&](int tile, int kvs, frag* dst) {
#pragma unroll
    for (int s = 0; s < D / 16; ++s) {
      const int kvg = tile * KVBLK + kvs * 32 + l31;
      const int kvc = kvg < Sk ? kvg : (Sk - 1);
      dst[s] = __builtin_bit_cast(
          frag, *(const u32x4_t*)(kp + (int64_t)kvc * k_ss + 16 * s + 8 * hi));
    }
  };

  frag kfr[KREG == 1 ? 2 : 1][D / 16];  // KREG: resident K fragments
  {  // prologue: tiles 0..NBUF-2
    u32x4_t kreg[PT], vreg[PT];
    issue_loads(0, kreg, vreg);
    write_lds(0, kreg, vreg);
    if (KREG == 1) {
      load_k_sub(0, 0, kfr[0]);
      load_k_sub(0, 1, kfr[KREG == 1 ? 1 : 0]);
    }
    if (NBUF == 4 && nt > 1) {
      issue_loads(1, kreg, vreg);
      write_lds(1, kreg, vreg);
    }
    __syncthreads();
    if (KREG == 2) {  // prime the prefetch pipeline: (tile 0, subtile 0)
#pragma unroll
      for (int s2 = 0; s2 < D / 16; ++s2)
        kfr[0][s2] =
            ba_ld_rowslice<T, D, SWZ_K>(ldsK(0), l31, 16 * s2 + 8 * hi);
    }
    if (SUBT == 3) {  // prime the 3-stage pipeline's K operands
#pragma unroll
      for (int s2 = 0; s2 < D / 16; ++s2)
        kfP[s2] =
            ba_ld_rowslice<T, D, SWZ_K>(ldsK(0), l31, 16 * s2 + 8 * hi);
    }
  }
  // static priority for the younger dispatch half (T5 static form):
  // wave-uniform condition via readfirstlane, one s_setprio, no flips
  if (NT == 512 && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  constexpr int AHEAD = NBUF == 4 ? 2 : 1;
  // the tile loop is hand-unrolled by the buffer period so `cur` (and the
  // staging write buffer) are compile-time per copy: the LDS buffer offset
  // folds into the ds_read/ds_write offset immediates instead of costing a
  // v_or per access — ~40 issue slots per tile in an issue-bound loop
  auto tile_body = [&](int t, auto curc) {
    const int cur = ba_buf_of(curc);  // compile-time literal for ba_ic
    const int kv0 = t * KVBLK;
    const bool has_next = (t + AHEAD) < nt;
    u32x4_t kreg[PT], vreg[PT];
    if (has_next) issue_loads(t + AHEAD, kreg, vreg);

    const bool active = !causal || (kv0 <= qb + 31);
    if (active && SUBT == 3) {
      const bool tile_full3 =
          (kv0 + KVBLK <= Sk) && (!causal || (kv0 + KVBLK - 1 <= qb));
      // NOTE: a C-level warmup peel (separate guarded/steady calls)
      // measured -48% — four inlined phase copies bloat the loop body.
      // The peel belongs at the .s level (round 3), where the warmup
      // runs before the loop label.
      [[clang::always_inline]] phase3(false, t, 0, cur, tile_full3);
      [[clang::always_inline]] phase3(false, t, 1, cur, tile_full3);
    } else if (active && SUBT == 2) {
      // ---- T15 pipeline: QK(j) fills while FINISH+PV(j-1) retire
      const bool tile_full =
          (kv0 + KVBLK <= Sk) && (!causal || (kv0 + KVBLK - 1 <= qb));
#pragma unroll
      for (int kvs = 0; kvs < 2; ++kvs) {
        f32x16_t stQ = (f32x16_t)(0.f);
#pragma unroll
        for (int s2 = 0; s2 < D / 16; ++s2) {
          frag kf = ba_ld_rowslice<T, D, SWZ_K>(ldsK(cur), kvs * 32 + l31,
                                                16 * s2 + 8 * hi);
          stQ = MT::mma(kf, qf[s2], stQ);
        }
        if (p_kv0 >= 0) [[clang::always_inline]] finish_subtile();
        stP = stQ;
        p_kv0 = kv0 + kvs * 32;
        p_cur = cur;
        p_full = tile_full;
      }
    } else if (active && SUBT == 1) {
      // ---- per-subtile pipeline: {QK, softmax, PV} x 2, independent
      // chains so the scheduler overlaps PV(0) with QK(1)
      const bool tile_full =
          (kv0 + KVBLK <= Sk) && (!causal || (kv0 + KVBLK - 1 <= qb));
      constexpr float DEFER_THR = 8.f;
#pragma unroll
      for (int kvs = 0; kvs < 2; ++kvs) {
        f32x16_t st = (f32x16_t)(0.f);
#pragma unroll
        for (int s2 = 0; s2 < D / 16; ++s2) {
          frag kf = KREG ? kfr[KREG == 1 ? kvs : 0][s2]
                         : ba_ld_rowslice<T, D, SWZ_K>(
                               ldsK(cur), kvs * 32 + l31, 16 * s2 + 8 * hi);
          st = MT::mma(kf, qf[s2], st);
        }
        // KREG=1: this subtile's K block is consumed — re-issue its
        // global loads for the next tile at once (WAR on the same
        // registers; HBM latency hides under softmax + PV + staging)
        if (KREG == 1 && has_next)
          load_k_sub(t + AHEAD, kvs, kfr[KREG == 1 ? kvs : 0]);
        // KREG=2: re-fill the fragment set from LDS one subtile ahead
        // ((t,0)->(t,1)->(t+1,0); tile t+1's buffer was staged two
        // tiles ago, ordered by the per-tile barrier)
        if (KREG == 2) {
          const int pn_t = (kvs == 0) ? t : t + 1;
          if (pn_t < nt) {
            const int pkvs = kvs ^ 1;
#pragma unroll
            for (int s2 = 0; s2 < D / 16; ++s2)
              kfr[0][s2] = ba_ld_rowslice<T, D, SWZ_K>(
                  ldsK(pn_t % NBUF), pkvs * 32 + l31, 16 * s2 + 8 * hi);
          }
        }
        // fold the softmax scale into the exp argument (exp2+fma): the
        // row max is taken on RAW scores (max commutes with c2 > 0), the
        // scale costs one multiply on the max instead of 16 per subtile
        if (!tile_full) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kv_g = kv0 + kvs * 32 + ba_crow(r, 0) + 4 * hi;
            if (!(kv_g < Sk && (!causal || kv_g <= q_row))) st[r] = BA_NEG_BIG;
          }
        }
        float tm = ba_max16(st);
        tm = fmaxf(tm, __shfl_xor(tm, 32));
        tm *= c2;  // into the exp2 domain
        if (!__all(tm - m2 <= DEFER_THR)) {
          const float mnew = fmaxf(m2, tm);
          const float alpha = ba_exp2(m2 - mnew);
          m2 = mnew;
          lsum *= alpha;
#pragma unroll
          for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
            for (int r = 0; r < 16; ++r) ot[dt][r] *= alpha;
        }
        float rowsum = 0.f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          st[r] = ba_exp2(__builtin_fmaf(st[r], c2, -m2));
          rowsum += st[r];
        }
        rowsum += __shfl_xor(rowsum, 32);
        lsum += rowsum;
        frag pf[2];
        ba_build_frag_pair<T>(st, pf);
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt) {
          const int drow = dt * 32 + l31;
#pragma unroll
          for (int u = 0; u < 2; ++u) {
            frag vv;
            if (VPATH == 0)
              vv = ba_ld_rowslice<T, KVBLK, SWZ_V, 7>(ldsVT(cur), drow,
                                                      kvs * 32 + 16 * u + 8 * hi);
            else
              vv = ba_ld_tr16_frag<T, D>(ldsVT(cur), lane,
                                         kvs * 32 + 16 * u, dt * 32);
            ot[dt] = MT::mma(vv, pf[u], ot[dt]);
          }
        }
      }
    } else if (active) {

      // ---- S^T = mfma(K, Q): two 32-kv subtiles
      f32x16_t st0 = (f32x16_t)(0.f), st1 = (f32x16_t)(0.f);
#pragma unroll
      for (int s = 0; s < D / 16; ++s) {
        frag k0 = ba_ld_rowslice<T, D, SWZ_K>(ldsK(cur), l31, 16 * s + 8 * hi);
        frag k1 =
            ba_ld_rowslice<T, D, SWZ_K>(ldsK(cur), 32 + l31, 16 * s + 8 * hi);
        st0 = MT::mma(k0, qf[s], st0);
        st1 = MT::mma(k1, qf[s], st1);
      }
      // ---- scale into exp2 domain (+ mask only on boundary tiles)
      const bool tile_full =
          (kv0 + KVBLK <= Sk) && (!causal || (kv0 + KVBLK - 1 <= qb));
      if (tile_full) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          st0[r] *= c2;
          st1[r] *= c2;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv_g0 = kv0 + ba_crow(r, 0) + 4 * hi;
          const int kv_g1 = kv_g0 + 32;
          st0[r] = (kv_g0 < Sk && (!causal || kv_g0 <= q_row)) ? st0[r] * c2
                                                               : BA_NEG_BIG;
          st1[r] = (kv_g1 < Sk && (!causal || kv_g1 <= q_row)) ? st1[r] * c2
                                                               : BA_NEG_BIG;
        }
      }
      // ---- online softmax update (lane-local), defer-max (T13):
      // skip the O/l rescale while the running max grows by <= THR
      // (exp2 domain); P is then bounded by 2^THR instead of 1, which
      // the fp32 accumulate absorbs.  Decision taken BEFORE this tile's
      // P is exponentiated (the textbook-safe order); wave-uniform.
      constexpr float DEFER_THR = 8.f;
      float tm = BA_NEG_BIG;
#pragma unroll
      for (int r = 0; r < 16; ++r) tm = fmaxf(tm, fmaxf(st0[r], st1[r]));
      tm = fmaxf(tm, __shfl_xor(tm, 32));
      const bool rescale = !__all(tm - m2 <= DEFER_THR);
      if (rescale) {
        const float mnew = fmaxf(m2, tm);
        const float alpha = ba_exp2(m2 - mnew);
        m2 = mnew;
        lsum *= alpha;
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) ot[dt][r] *= alpha;
      }
      float rowsum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        st0[r] = ba_exp2(st0[r] - m2);
        st1[r] = ba_exp2(st1[r] - m2);
        rowsum += st0[r] + st1[r];
      }
      rowsum += __shfl_xor(rowsum, 32);
      lsum += rowsum;

      // ---- P -> fragments, O^T += mfma(V^T, P^T)
      frag pf0[2], pf1[2];
      ba_build_frag_pair<T>(st0, pf0);
      ba_build_frag_pair<T>(st1, pf1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt) {
        const int drow = dt * 32 + l31;
#pragma unroll
        for (int u = 0; u < 2; ++u) {
          frag v0, v1;
          if (VPATH == 0) {
            v0 = ba_ld_rowslice<T, KVBLK, SWZ_V, 7>(ldsVT(cur), drow,
                                                    16 * u + 8 * hi);
            v1 = ba_ld_rowslice<T, KVBLK, SWZ_V, 7>(ldsVT(cur), drow,
                                                    32 + 16 * u + 8 * hi);
          } else {
            v0 = ba_ld_tr16_frag<T, D>(ldsVT(cur), lane, 16 * u, dt * 32);
            v1 = ba_ld_tr16_frag<T, D>(ldsVT(cur), lane, 32 + 16 * u, dt * 32);
          }
          ot[dt] = MT::mma(v0, pf0[u], ot[dt]);
          ot[dt] = MT::mma(v1, pf1[u], ot[dt]);
        }
      }
    }

    if (has_next) write_lds((cur + AHEAD) % NBUF, kreg, vreg);
    // SUBT=2 reads the previous tile's V one subtile late: barrier every
    // tile so the rewrite (2 buffers ahead) never crosses those reads
    if (SUBT == 2 || SUBT == 3 || KREG == 2 || NBUF == 2 || (t & 1) ||
        t + 1 >= nt)
      __syncthreads();
  };
  if constexpr (SUBT == 1) {
    for (int tb = 0; tb < nt; tb += NBUF) {
      [[clang::always_inline]] tile_body(tb, ba_ic<0>{});
      if (tb + 1 < nt) [[clang::always_inline]] tile_body(tb + 1, ba_ic<1 % NBUF>{});
      if (NBUF > 2) {
        if (tb + 2 < nt) [[clang::always_inline]] tile_body(tb + 2, ba_ic<2 % NBUF>{});
        if (tb + 3 < nt) [[clang::always_inline]] tile_body(tb + 3, ba_ic<3 % NBUF>{});
      }
    }
  } else {
    // the pipelined variants (SUBT=2/3) carry register state across tile
    // boundaries; unrolled copies quadruple it into scratch spills, so
    // they keep the rolled loop with a runtime buffer index
    for (int t = 0; t < nt; ++t)
      [[clang::always_inline]] tile_body(t, ba_dyn{t % NBUF});
  }
  if (SUBT == 2 && p_kv0 >= 0) finish_subtile();  // drain the pipeline
  if (SUBT == 3) {  // drain both pending stages (PV first: scale order)
    if (v_kv0 >= 0) {
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int u = 0; u < 2; ++u)
          ot[dt] = MT::mma(vvP[dt * 2 + u], pvf[u], ot[dt]);
    }
    if (p_kv0 >= 0) finish_subtile();
  }

  // ---- epilogue
  if (q_row < Sq) {
    if (OUT_STATE) {
      float* arow = st_acc + b * a_sb + (int64_t)q_row * a_ss + n * a_sh;
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r)
          arow[dt * 32 + ba_crow(r, hi)] = ot[dt][r];
      if (hi == 0) {
        st_m[b * ml_sb + n * ml_sh + q_row] = m2;
        st_l[b * ml_sb + n * ml_sh + q_row] = lsum;
      }
    } else {
      const float inv_l = 1.f / lsum;
      float* orow = o + (((int64_t)b * Sq + q_row) * N + n) * D;
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r)
          orow[dt * 32 + ba_crow(r, hi)] = ot[dt][r] * inv_l;
      if (hi == 0)
        lse[((int64_t)b * N + n) * Sq + q_row] = BA_LN2 * (m2 + log2f(lsum));
    }
  }
}

// finalize: o = acc / l (cast to T), lse = ln2 * (m + log2(l)).
// acc/m/l are the FULL chunk (contiguous); one wave row-group per row.
template <typename T, int D>
__global__ void attn_fwd_finalize_kernel(
    const float* __restrict__ acc, const float* __restrict__ m,
    const float* __restrict__ l, T* __restrict__ o, float* __restrict__ lse,
    int S, int N) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  constexpr int LPR = D / 16;           // lanes per row (8 for D=128)
  const int rows_per_wave = 64 / LPR;   // 8
  const int row_in = wave * rows_per_wave + lane / LPR;
  const int sub = lane % LPR;
  const int s = blockIdx.x * 4 * rows_per_wave + row_in;
  const int n = blockIdx.y, b = blockIdx.z;
  if (s >= S) return;
  const float lv = l[((int64_t)b * N + n) * S + s];
  const float inv_l = 1.f / lv;
  const float* arow = acc + (((int64_t)b * S + s) * N + n) * D + sub * 16;
  T* orow = o + (((int64_t)b * S + s) * N + n) * D + sub * 16;
#pragma unroll
  for (int j = 0; j < 16; ++j) orow[j] = (T)(arow[j] * inv_l);
  if (sub == 0)
    lse[((int64_t)b * N + n) * S + s] =
        BA_LN2 * (m[((int64_t)b * N + n) * S + s] + log2f(lv));
}

// ---- MFMA layout probe (test support): D = A*B for one 32x32x16 tile,
// with A,B,D moved per the layout assumptions above.  a: [32][16] row-major,
// b: [16][32] row-major, d: [32][32] row-major, all dense in HBM.
template <typename T>
__global__ void mfma_probe_kernel(const T* a, const T* b, float* d) {
  using MT = mfma_traits<T>;
  using frag = typename MT::frag;
  const int lane = threadIdx.x & 63;
  const int l31 = lane & 31, hi = lane >> 5;
  frag af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[l31 * 16 + 8 * hi + j];    // A[row=l31][k=8*hi+j]
    bf[j] = b[(8 * hi + j) * 32 + l31];  // B[k=8*hi+j][col=l31]
  }
  f32x16_t c = (f32x16_t)(0.f);
  c = MT::mma(af, bf, c);
#pragma unroll
  for (int r = 0; r < 16; ++r) d[ba_crow(r, hi) * 32 + l31] = c[r];
}


// ---- ds_read_tr16_b64 semantics probe (test/dev support).
// Fills LDS with element-index pattern, issues the transpose-read with a
// per-mode address pattern, dumps each lane's 4 returned u16s.
typedef __attribute__((ext_vector_type(4))) short s16x4_t;
__global__ void tr16_probe_kernel(int mode, int* out) {
  __shared__ unsigned short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  const int l = threadIdx.x & 63;
  int e;
  switch (mode) {
    case 0: e = 0; break;               // uniform base
    case 1: e = (l & 15) * 4; break;    // per-16-group column stride 4
    case 2: e = l * 4; break;           // per-lane stride 4
    case 3: e = (l >> 4) * 64; break;   // per-quarter base
    default: e = (l & 15) * 4 + (l >> 4) * 256; break;
  }
  auto p = (__attribute__((address_space(3))) s16x4_t*)&lds[e];
  s16x4_t v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = (int)(unsigned short)v[j];
}

}  // namespace

static thread_local char g_err[256] = "";
extern "C" const char* bahip_last_error(void) { return g_err; }

template <typename T, int D>
static int launch_fwd(const void* q, const void* k, const void* v, float* o,
                      float* lse, int64_t B, int64_t Sq, int64_t Sk, int64_t N,
                      const int64_t* qs, const int64_t* ks, const int64_t* vs,
                      float scale, int causal, void* stream) {
  // read per call (getenv is ~ns against ms-scale launches) so tests can
  // flip the variant paths in-process, like the backward plans do.
  // defaults: V^T image (tr16 measured -3%), per-subtile softmax (+3%)
  const char* e;
  const int vpath = (e = getenv("BA_FWD_VPATH")) ? atoi(e) : 0;
  const int subt = (e = getenv("BA_FWD_SUBT")) ? atoi(e) : 1;
  const int ntw = (e = getenv("BA_FWD_NT")) ? atoi(e) : 512;
  const int nbuf = (e = getenv("BA_FWD_NBUF")) ? atoi(e) : 2;
  const int kreg = (e = getenv("BA_FWD_KREG")) ? atoi(e) : 0;
  if (kreg == 1 || kreg == 2) {
    dim3 gk((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B);
    if (kreg == 1)
      attn_fwd_kernel<T, D, 64, 0, 0, 1, 512, 2, 1>
          <<<gk, dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, o, lse, (int)Sq,
              (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],
              vs[0], vs[1], vs[2], scale, causal, nullptr, nullptr, nullptr,
              0, 0, 0, 0, 0, 0);
    else
      attn_fwd_kernel<T, D, 64, 0, 0, 1, 512, 4, 2>
          <<<gk, dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, o, lse, (int)Sq,
              (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],
              vs[0], vs[1], vs[2], scale, causal, nullptr, nullptr, nullptr,
              0, 0, 0, 0, 0, 0);
    BA_CHECK_LAUNCH();
    return 0;
  }
  if (subt == 3) {
    attn_fwd_kernel<T, D, 64, 0, 0, 3, 256, 4>
        <<<dim3((unsigned)((Sq + 127) / 128), (unsigned)N, (unsigned)B),
           dim3(256), 0, (hipStream_t)stream>>>(
            (const T*)q, (const T*)k, (const T*)v, o, lse, (int)Sq, (int)Sk,
            (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2], vs[0], vs[1],
            vs[2], scale, causal, nullptr, nullptr, nullptr, 0, 0, 0, 0, 0,
            0);
    BA_CHECK_LAUNCH();
    return 0;
  }
  if (subt == 2) {
    attn_fwd_kernel<T, D, 64, 0, 0, 2, 512, 4>
        <<<dim3((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B),
           dim3(512), 0, (hipStream_t)stream>>>(
            (const T*)q, (const T*)k, (const T*)v, o, lse, (int)Sq, (int)Sk,
            (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2], vs[0], vs[1],
            vs[2], scale, causal, nullptr, nullptr, nullptr, 0, 0, 0, 0, 0,
            0);
    BA_CHECK_LAUNCH();
    return 0;
  }
  if (nbuf == 4 && ntw == 512) {
    if (vpath == 0)
      attn_fwd_kernel<T, D, 64, 0, 0, 1, 512, 4>
          <<<dim3((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B),
             dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, o, lse, (int)Sq,
              (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],
              vs[0], vs[1], vs[2], scale, causal, nullptr, nullptr, nullptr,
              0, 0, 0, 0, 0, 0);
    else
      attn_fwd_kernel<T, D, 64, 0, 1, 1, 512, 4>
          <<<dim3((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B),
             dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, o, lse, (int)Sq,
              (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2],
              vs[0], vs[1], vs[2], scale, causal, nullptr, nullptr, nullptr,
              0, 0, 0, 0, 0, 0);
    BA_CHECK_LAUNCH();
    return 0;
  }
#define FWD_LAUNCH(VP, ST, NTV)                                               \
  attn_fwd_kernel<T, D, 64, 0, VP, ST, NTV>                                   \
      <<<dim3((unsigned)((Sq + NTV / 2 - 1) / (NTV / 2)), (unsigned)N,        \
              (unsigned)B),                                                   \
         dim3(NTV), 0, (hipStream_t)stream>>>(                                \
          (const T*)q, (const T*)k, (const T*)v, o, lse, (int)Sq, (int)Sk,    \
          (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2], vs[0], vs[1],     \
          vs[2], scale, causal, nullptr, nullptr, nullptr, 0, 0, 0, 0, 0, 0)
  if (ntw == 256) {
    if (vpath == 0) FWD_LAUNCH(0, 1, 256);
    else FWD_LAUNCH(1, 1, 256);
  } else if (vpath == 0 && subt == 0) FWD_LAUNCH(0, 0, 512);
  else if (vpath == 0 && subt == 1) FWD_LAUNCH(0, 1, 512);
  else if (vpath == 1 && subt == 0) FWD_LAUNCH(1, 0, 512);
  else FWD_LAUNCH(1, 1, 512);
#undef FWD_LAUNCH
  BA_CHECK_LAUNCH();
  return 0;
}

template <typename T, int D>
static int launch_fwd_accum(const void* q, const void* k, const void* v,
                            int64_t B, int64_t Sq, int64_t Sk, int64_t N,
                            const int64_t* qs, const int64_t* ks,
                            const int64_t* vs, float scale, int causal,
                            float* acc, float* m, float* l,
                            const int64_t* as, const int64_t* mls,
                            int carry_in, void* stream) {
  static const int vpath = [] {
    const char* e = getenv("BA_FWD_VPATH");
    return e ? atoi(e) : 0;
  }();
  static const int subt = [] {
    const char* e = getenv("BA_FWD_SUBT");
    return e ? atoi(e) : 1;  // per-subtile softmax pipeline (+3% measured)
  }();
  static const int ntw = [] {
    const char* e = getenv("BA_FWD_NT");
    return e ? atoi(e) : 512;
  }();
  static const int nbuf = [] {
    const char* e = getenv("BA_FWD_NBUF");
    return e ? atoi(e) : 2;
  }();
  static const int kreg = [] {
    const char* e = getenv("BA_FWD_KREG");
    return e ? atoi(e) : 0;
  }();
  if (kreg == 1 || kreg == 2) {
    dim3 gridk((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B);
    if (kreg == 1)
      attn_fwd_kernel<T, D, 64, 1, 0, 1, 512, 2, 1>
          <<<gridk, dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, nullptr, nullptr,
              (int)Sq, (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1],
              ks[2], vs[0], vs[1], vs[2], scale, causal, acc, m, l, as[0],
              as[1], as[2], mls[0], mls[1], carry_in);
    else
      attn_fwd_kernel<T, D, 64, 1, 0, 1, 512, 4, 2>
          <<<gridk, dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, nullptr, nullptr,
              (int)Sq, (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1],
              ks[2], vs[0], vs[1], vs[2], scale, causal, acc, m, l, as[0],
              as[1], as[2], mls[0], mls[1], carry_in);
    BA_CHECK_LAUNCH();
    return 0;
  }
  if (subt == 3) {
    dim3 grid3((unsigned)((Sq + 127) / 128), (unsigned)N, (unsigned)B);
    attn_fwd_kernel<T, D, 64, 1, 0, 3, 256, 4>
        <<<grid3, dim3(256), 0, (hipStream_t)stream>>>(
            (const T*)q, (const T*)k, (const T*)v, nullptr, nullptr, (int)Sq,
            (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2], vs[0],
            vs[1], vs[2], scale, causal, acc, m, l, as[0], as[1], as[2],
            mls[0], mls[1], carry_in);
    BA_CHECK_LAUNCH();
    return 0;
  }
  if (subt == 2) {
    dim3 grid2((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B);
    attn_fwd_kernel<T, D, 64, 1, 0, 2, 512, 4>
        <<<grid2, dim3(512), 0, (hipStream_t)stream>>>(
            (const T*)q, (const T*)k, (const T*)v, nullptr, nullptr, (int)Sq,
            (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2], vs[0],
            vs[1], vs[2], scale, causal, acc, m, l, as[0], as[1], as[2],
            mls[0], mls[1], carry_in);
    BA_CHECK_LAUNCH();
    return 0;
  }
  if (nbuf == 4 && ntw == 512) {
    dim3 grid4((unsigned)((Sq + 255) / 256), (unsigned)N, (unsigned)B);
    if (vpath == 0)
      attn_fwd_kernel<T, D, 64, 1, 0, 1, 512, 4>
          <<<grid4, dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, nullptr, nullptr,
              (int)Sq, (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1],
              ks[2], vs[0], vs[1], vs[2], scale, causal, acc, m, l, as[0],
              as[1], as[2], mls[0], mls[1], carry_in);
    else
      attn_fwd_kernel<T, D, 64, 1, 1, 1, 512, 4>
          <<<grid4, dim3(512), 0, (hipStream_t)stream>>>(
              (const T*)q, (const T*)k, (const T*)v, nullptr, nullptr,
              (int)Sq, (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1],
              ks[2], vs[0], vs[1], vs[2], scale, causal, acc, m, l, as[0],
              as[1], as[2], mls[0], mls[1], carry_in);
    BA_CHECK_LAUNCH();
    return 0;
  }
#define FWD_ALAUNCH(VP, ST, NTV)                                              \
  attn_fwd_kernel<T, D, 64, 1, VP, ST, NTV>                                   \
      <<<dim3((unsigned)((Sq + NTV / 2 - 1) / (NTV / 2)), (unsigned)N,        \
              (unsigned)B),                                                   \
         dim3(NTV), 0, (hipStream_t)stream>>>(                                \
          (const T*)q, (const T*)k, (const T*)v, nullptr, nullptr, (int)Sq,   \
          (int)Sk, (int)N, qs[0], qs[1], qs[2], ks[0], ks[1], ks[2], vs[0],   \
          vs[1], vs[2], scale, causal, acc, m, l, as[0], as[1], as[2],        \
          mls[0], mls[1], carry_in)
  if (ntw == 256) {
    if (vpath == 0) FWD_ALAUNCH(0, 1, 256);
    else FWD_ALAUNCH(1, 1, 256);
  } else if (vpath == 0 && subt == 0) FWD_ALAUNCH(0, 0, 512);
  else if (vpath == 0 && subt == 1) FWD_ALAUNCH(0, 1, 512);
  else if (vpath == 1 && subt == 0) FWD_ALAUNCH(1, 0, 512);
  else FWD_ALAUNCH(1, 1, 512);
#undef FWD_ALAUNCH
  BA_CHECK_LAUNCH();
  return 0;
}

extern "C" int bahip_attn_fwd_accum(
    const void* q, const void* k, const void* v, int64_t B, int64_t Sq,
    int64_t Sk, int64_t N, int64_t D, const int64_t q_strides[3],
    const int64_t k_strides[3], const int64_t v_strides[3],
    float softmax_scale, int causal, int dtype, float* acc, float* m,
    float* l, const int64_t acc_strides[3], const int64_t ml_strides[2],
    int carry_in, void* stream) {
  if (causal && Sq != Sk) return 1001;
  if (D == 128 && dtype == BAHIP_BF16)
    return launch_fwd_accum<__bf16, 128>(q, k, v, B, Sq, Sk, N, q_strides,
                                         k_strides, v_strides, softmax_scale,
                                         causal, acc, m, l, acc_strides,
                                         ml_strides, carry_in, stream);
  if (D == 128 && dtype == BAHIP_F16)
    return launch_fwd_accum<_Float16, 128>(q, k, v, B, Sq, Sk, N, q_strides,
                                           k_strides, v_strides, softmax_scale,
                                           causal, acc, m, l, acc_strides,
                                           ml_strides, carry_in, stream);
  if (D == 64 && dtype == BAHIP_BF16)
    return launch_fwd_accum<__bf16, 64>(q, k, v, B, Sq, Sk, N, q_strides,
                                        k_strides, v_strides, softmax_scale,
                                        causal, acc, m, l, acc_strides,
                                        ml_strides, carry_in, stream);
  if (D == 64 && dtype == BAHIP_F16)
    return launch_fwd_accum<_Float16, 64>(q, k, v, B, Sq, Sk, N, q_strides,
                                          k_strides, v_strides, softmax_scale,
                                          causal, acc, m, l, acc_strides,
                                          ml_strides, carry_in, stream);
  return 1002;
}

extern "C" int bahip_attn_fwd_finalize(const float* acc, const float* m,
                                       const float* l, void* o, float* lse,
                                       int64_t B, int64_t S, int64_t N,
                                       int64_t D, int dtype, void* stream) {
  const int rows_per_block = (D == 64) ? 64 : 32;  // 4 waves x 64/(D/16)
  dim3 grid((unsigned)((S + rows_per_block - 1) / rows_per_block), (unsigned)N,
            (unsigned)B);
#define LAUNCH_FIN(T, DD)                                                   attn_fwd_finalize_kernel<T, DD><<<grid, 256, 0, (hipStream_t)stream>>>(       acc, m, l, (T*)o, lse, (int)S, (int)N)
  if (D == 128 && dtype == BAHIP_BF16) LAUNCH_FIN(__bf16, 128);
  else if (D == 128 && dtype == BAHIP_F16) LAUNCH_FIN(_Float16, 128);
  else if (D == 64 && dtype == BAHIP_BF16) LAUNCH_FIN(__bf16, 64);
  else if (D == 64 && dtype == BAHIP_F16) LAUNCH_FIN(_Float16, 64);
  else return 1002;
#undef LAUNCH_FIN
  BA_CHECK_LAUNCH();
  return 0;
}

extern "C" int bahip_attn_fwd(const void* q, const void* k, const void* v,
                              float* o, float* lse, int64_t B, int64_t Sq,
                              int64_t Sk, int64_t N, int64_t D,
                              const int64_t q_strides[3],
                              const int64_t k_strides[3],
                              const int64_t v_strides[3], float softmax_scale,
                              int causal, int dtype, void* stream) {
  if (causal && Sq != Sk) {
    snprintf(g_err, sizeof g_err, "causal tiles require Sq == Sk (%d vs %d)",
             (int)Sq, (int)Sk);
    return 1001;
  }
  if (D == 128) {
    if (dtype == BAHIP_BF16)
      return launch_fwd<__bf16, 128>(q, k, v, o, lse, B, Sq, Sk, N, q_strides,
                                     k_strides, v_strides, softmax_scale,
                                     causal, stream);
    if (dtype == BAHIP_F16)
      return launch_fwd<_Float16, 128>(q, k, v, o, lse, B, Sq, Sk, N,
                                       q_strides, k_strides, v_strides,
                                       softmax_scale, causal, stream);
  } else if (D == 64) {
    if (dtype == BAHIP_BF16)
      return launch_fwd<__bf16, 64>(q, k, v, o, lse, B, Sq, Sk, N, q_strides,
                                    k_strides, v_strides, softmax_scale,
                                    causal, stream);
    if (dtype == BAHIP_F16)
      return launch_fwd<_Float16, 64>(q, k, v, o, lse, B, Sq, Sk, N, q_strides,
                                      k_strides, v_strides, softmax_scale,
                                      causal, stream);
  }
  snprintf(g_err, sizeof g_err, "unsupported head_dim %d / dtype %d", (int)D,
           dtype);
  return 1002;
}

extern "C" int bahip_tr16_probe(int mode, int* out, void* stream) {
  tr16_probe_kernel<<<1, 64, 0, (hipStream_t)stream>>>(mode, out);
  BA_CHECK_LAUNCH();
  return 0;
}

extern "C" int bahip_mfma_probe(const void* a, const void* b, float* d,
                                int dtype, void* stream) {
  if (dtype == BAHIP_BF16)
    mfma_probe_kernel<__bf16><<<1, 64, 0, (hipStream_t)stream>>>(
        (const __bf16*)a, (const __bf16*)b, d);
  else
    mfma_probe_kernel<_Float16><<<1, 64, 0, (hipStream_t)stream>>>(
        (const _Float16*)a, (const _Float16*)b, d);
  BA_CHECK_LAUNCH();
  return 0;
}
