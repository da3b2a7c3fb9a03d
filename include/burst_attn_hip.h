/* C-ABI of the MI355X-native BurstAttention tile kernels (gfx950).
 *
 * These entry points replace, one for one, the flash-attn CUDA extension
 * calls the reference makes on its default path:
 *   bahip_attn_fwd            <- _flash_attn_forward(q,k,v,...) -> (o, lse)
 *                                as called at reference
 *                                burst_attn/burst_utils.py:150-160
 *   bahip_attn_bwd_preprocess <- flash bwd delta precompute
 *                                (sum(o*do), cf. reference lao.py:247-269 and
 *                                burst_attn_interface.py:272-278)
 *   bahip_attn_bwd            <- _flash_attn_backward(do,q,k,v,o,lse,dq,dk,dv,
 *                                ..., softmax_d) as called at
 *                                burst_attn/burst_utils.py:211-248
 *
 * Conventions:
 *   - q/k/v/do: fp16 or bf16, logical layout [B, S, N, D]; strides are in
 *     ELEMENTS as {batch, seq, head}; the innermost head_dim D is assumed
 *     contiguous.  Arbitrary seq slices (zigzag halves, striped shifts) are
 *     expressed through the strides + base pointer.
 *   - o (forward out) is fp32 [B, Sq, N, D] contiguous; lse fp32 [B, N, Sq]
 *     contiguous (flash-attn layout, burst_utils.py:150-163).
 *   - lse/delta inputs to the backward may be seq-sliced: strides are
 *     {batch, head} in elements, seq stride is 1.
 *   - dq/dk/dv outputs are fp32 ACCUMULATORS, strided {batch, seq, head}
 *     (D contiguous): every call ADDS its tile contribution in place, so
 *     the ring layer accumulates rounds without elementwise-add passes.
 *     Pass zero-filled buffers for plain (non-accumulating) semantics.
 *   - deterministic != 0 selects the atomic-free two-pass-dq kernel plan
 *     (bitwise run-to-run identical); deterministic == 0 (default)
 *     selects the fused dK+dQ plan with flash-attn's atomic fp32 dq
 *     accumulation (2 fewer tile GEMMs).
 *   - causal implies Sq == Sk (equal-length tiles; the ring's zigzag /
 *     striped bookkeeping reduces every other case to non-causal tiles).
 *   - stream is a hipStream_t.
 * Returns 0 on success; nonzero = error (message via bahip_last_error()).
 */
#pragma once
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define BAHIP_F16 0
#define BAHIP_BF16 1

const char* bahip_last_error(void);

int bahip_attn_fwd(
    const void* q, const void* k, const void* v,
    float* o, float* lse,
    int64_t B, int64_t Sq, int64_t Sk, int64_t N, int64_t D,
    const int64_t q_strides[3],
    const int64_t k_strides[3],
    const int64_t v_strides[3],
    float softmax_scale, int causal, int dtype,
    void* stream);

/* carry-in accumulator forward (in-kernel LSE merge; replaces the
 * per-round cuda_scale_out_lse_helper pass, burst_utils.py:20-33).
 * State: acc fp32 [B,S,N,D] (unnormalised O), m fp32 [B,N,S] (running max,
 * exp2 domain), l fp32 [B,N,S] (running sum); acc strided {b,s,h}, m/l
 * strided {b,h} with contiguous seq.  carry_in=0 initialises the state. */
int bahip_attn_fwd_accum(
    const void* q, const void* k, const void* v,
    int64_t B, int64_t Sq, int64_t Sk, int64_t N, int64_t D,
    const int64_t q_strides[3], const int64_t k_strides[3],
    const int64_t v_strides[3], float softmax_scale, int causal, int dtype,
    float* acc, float* m, float* l, const int64_t acc_strides[3],
    const int64_t ml_strides[2], int carry_in, void* stream);

/* o = acc / l cast to dtype; lse = ln(l) + m*ln2; contiguous full chunk */
int bahip_attn_fwd_finalize(
    const float* acc, const float* m, const float* l, void* o, float* lse,
    int64_t B, int64_t S, int64_t N, int64_t D, int dtype, void* stream);

int bahip_attn_bwd_preprocess(
    const void* o, const void* dout, float* delta,
    int64_t B, int64_t S, int64_t N, int64_t D,
    const int64_t o_strides[3], const int64_t do_strides[3],
    int o_dtype, int do_dtype, void* stream);

int bahip_attn_bwd(
    const void* dout, const void* q, const void* k, const void* v,
    const float* delta, const float* lse,
    float* dq, float* dk, float* dv,
    int64_t B, int64_t Sq, int64_t Sk, int64_t N, int64_t D,
    const int64_t do_strides[3], const int64_t q_strides[3],
    const int64_t k_strides[3], const int64_t v_strides[3],
    const int64_t delta_strides[2], const int64_t lse_strides[2],
    const int64_t dq_strides[3], const int64_t dk_strides[3],
    const int64_t dv_strides[3],
    float softmax_scale, int causal, int deterministic, int dtype,
    void* stream);

#ifdef __cplusplus
}
#endif
