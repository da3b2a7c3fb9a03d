"""Oracle-backed tile provider — TEST INFRASTRUCTURE ONLY.

Lets the product's ring orchestration (burst_attn_amd.interface) run on CPU
(gloo) without the HIP extension, with the oracle supplying the local tile
math.  The oracle is the checker, never the shipped path (oracle/__init__.py
header); only tests inject this provider."""

import torch

import oracle


class OracleTileProvider:
    def fwd(self, q, k, v, scale, causal):
        return oracle.tile_fwd(q, k, v, scale, causal)

    def bwd_preprocess(self, o, do, out=None):
        # delta = rowsum(o * do) fp32, [B,S,N,D] -> [B,N,S]
        delta = (o.to(torch.float32) * do.to(torch.float32)).sum(-1).transpose(1, 2).contiguous()
        if out is not None:
            out.copy_(delta)
            return out
        return delta

    def bwd(self, do, q, k, v, delta, lse, scale, causal, deterministic):
        return oracle.tile_bwd(
            do, q, k, v, lse, scale, causal, softmax_d=delta
        )

    def bwd_accum(self, do, q, k, v, delta, lse, scale, causal, deterministic,
                  dq, dk, dv):
        dq_i, dk_i, dv_i = self.bwd(do, q, k, v, delta, lse, scale, causal,
                                    deterministic)
        dq += dq_i
        dk += dk_i
        dv += dv_i

    def merge(self, o, lse, o_i, lse_i):
        return oracle.scale_out_lse(o, lse, o_i, lse_i)

    # carry-in accumulator path: state = [o fp32 [B,S,N,D], lse [B,S,N,1]]
    # maintained with the reference merge (burst_utils.py:20-33)
    def fwd_accum(self, state, q, k, v, scale, causal, row_offset=0):
        o_i, lse_i = oracle.tile_fwd(q, k, v, scale, causal)
        if state is None:
            assert row_offset == 0
            return [o_i.to(torch.float32),
                    lse_i.transpose(-2, -1).unsqueeze(-1).contiguous()]
        o, lse = state
        sl = slice(row_offset, row_offset + q.shape[1])
        o[:, sl], lse[:, sl] = oracle.scale_out_lse(o[:, sl], lse[:, sl], o_i, lse_i)
        return state

    def fwd_finalize(self, state, out_dtype):
        o, lse = state
        return o.to(out_dtype), lse.squeeze(-1).transpose(1, 2).contiguous()
