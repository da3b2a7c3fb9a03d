"""Local flash-attention tile layer — the HIP/CDNA4 compute path.

This is the layer that replaces the reference's three interchangeable
backends (flash-attn CUDA extension / Triton ``lao.py`` / pure-torch math,
dispatched at ``burst_attn/burst_utils.py:103-249``) with ONE hand-written
gfx950 HIP kernel pair (north_star: no Triton, no flash-attn extension,
no multi-backend dispatch).

Provider contract (all tensors in flash layout [B, S, N, D]; lse/delta in
[B, N, S] fp32 — the layouts the flash-attn extension hands the reference,
``burst_utils.py:149-177``):

  fwd(q, k, v, scale, causal)            -> (o_i fp32 [B,Sq,N,D],
                                             lse_i fp32 [B,N,Sq])
  bwd_preprocess(o, do, out=None)        -> delta fp32 [B,N,S]
                                            (= rowsum(o*do); the flash bwd
                                            preprocess, cf. lao.py:247-269)
  bwd(do, q, k, v, delta, lse, scale,
      causal, deterministic)             -> (dq, dk, dv fp32 [.,.,N,D])
  bwd_accum(do, q, k, v, delta, lse,
      scale, causal, deterministic,
      dq, dk, dv)                        -> None; ADDS the tile's dq/dk/dv
                                            contribution into the given
                                            fp32 accumulators (strided
                                            views allowed) — the ring's
                                            round accumulation runs in the
                                            kernel epilogues, not python
                                            adds
  merge(o, lse, o_i, lse_i)              -> merged (o, lse); o [B,S,N,D]
                                            fp32, lse [B,S,N,1] fp32,
                                            per burst_utils.py:20-33.

The default provider is the gfx950 HIP extension and FAILS LOUDLY if the
extension is missing on a GPU machine — there is no CPU/eager fallback in
the product path.  Tests may inject an oracle-backed provider via
``_set_tile_provider_for_testing`` (test infrastructure only).
"""

import torch

__all__ = ["get_tile_provider", "_set_tile_provider_for_testing", "HipTileProvider"]


@torch.jit.script
def _merge_scale_out_lse(o, lse, o_i, lse_i):
    # restates cuda_scale_out_lse_helper (reference burst_utils.py:20-33)
    o_i = o_i.to(torch.float32)
    lse_i = lse_i.transpose(-2, -1).unsqueeze(-1).contiguous()
    new_lse = lse + torch.log1p(torch.exp(lse_i - lse))
    o = torch.exp(lse - new_lse) * o + torch.exp(lse_i - new_lse) * o_i
    return o, new_lse


class HipTileProvider:
    """gfx950 HIP kernels behind the C-ABI in include/burst_attn_hip.h."""

    def __init__(self):
        import json
        import os

        from . import _ext

        self._ext = _ext.load_extension()  # raises loudly if missing
        # BA_FWD_ASM: route fwd_accum through a .s-built hsaco.
        #   1 = the unmodified re-assembly (parity check of the module path)
        #   2 = the hand-scheduled variant (tools/s_patch.py transforms)
        self._asm_fwd = False
        asm = os.environ.get("BA_FWD_ASM", "0")
        if asm in ("1", "2"):
            here = os.path.dirname(os.path.abspath(__file__))
            name = "_asm_fwd.hsaco" if asm == "1" else "_asm_fwd_p.hsaco"
            with open(os.path.join(here, "_asm_fwd_syms.json")) as f:
                syms = json.load(f)
            self._ext.attn_fwd_asm_load(os.path.join(here, name),
                                        syms["f16_accum"],
                                        syms["bf16_accum"])
            self._asm_fwd = True

    def fwd(self, q, k, v, scale, causal):
        return self._ext.attn_fwd(q, k, v, float(scale), bool(causal))

    def bwd_preprocess(self, o, do, out=None):
        return self._ext.attn_bwd_preprocess(o, do, out)

    def bwd(self, do, q, k, v, delta, lse, scale, causal, deterministic):
        return self._ext.attn_bwd(
            do, q, k, v, delta, lse, float(scale), bool(causal), bool(deterministic)
        )

    def bwd_accum(self, do, q, k, v, delta, lse, scale, causal, deterministic,
                  dq, dk, dv):
        self._ext.attn_bwd_accum(
            do, q, k, v, delta, lse, float(scale), bool(causal),
            bool(deterministic), dq, dk, dv,
        )

    def merge(self, o, lse, o_i, lse_i):
        return _merge_scale_out_lse(o, lse, o_i, lse_i)

    # ---- carry-in accumulator path (in-kernel LSE merge; lao.py design) --
    # state = (acc fp32 [B,S,N,D] unnormalised O, m fp32 [B,N,S] exp2-domain
    # running max, l fp32 [B,N,S] running sum) over the rank's FULL chunk;
    # row_offset targets the zigzag-half / striped-shift row windows.
    def fwd_accum(self, state, q, k, v, scale, causal, row_offset=0):
        # the .s-built module carries the D=128 production variants only
        accum = (self._ext.attn_fwd_accum_asm
                 if self._asm_fwd and q.shape[3] == 128
                 else self._ext.attn_fwd_accum)
        if state is None:
            assert row_offset == 0, "state is created by a full-row round"
            B, S, N, D = q.shape
            acc = torch.empty(B, S, N, D, dtype=torch.float32, device=q.device)
            m = torch.empty(B, N, S, dtype=torch.float32, device=q.device)
            l = torch.empty(B, N, S, dtype=torch.float32, device=q.device)
            accum(q, k, v, float(scale), bool(causal), acc, m, l, False)
            return (acc, m, l)
        acc, m, l = state
        sq = q.shape[1]
        sl = slice(row_offset, row_offset + sq)
        accv, mv, lv = acc[:, sl], m[:, :, sl], l[:, :, sl]
        accum(q, k, v, float(scale), bool(causal), accv, mv, lv, True)
        return state

    def fwd_finalize(self, state, out_dtype):
        acc, m, l = state
        return self._ext.attn_fwd_finalize(acc, m, l, out_dtype)


_provider = None
_provider_override = None


def _set_tile_provider_for_testing(provider):
    """TEST INFRASTRUCTURE ONLY — inject a CPU (oracle) tile provider so the
    ring orchestration can be exercised without a GPU.  The product path
    never calls this."""
    global _provider_override
    _provider_override = provider


def get_tile_provider():
    global _provider
    if _provider_override is not None:
        return _provider_override
    if _provider is None:
        _provider = HipTileProvider()
    return _provider
