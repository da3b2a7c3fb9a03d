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

    def bwd_preprocess(self, o, do):
        # delta = rowsum(o * do) fp32, [B,S,N,D] -> [B,N,S]
        return (o.to(torch.float32) * do.to(torch.float32)).sum(-1).transpose(1, 2).contiguous()

    def bwd(self, do, q, k, v, delta, lse, scale, causal, deterministic):
        return oracle.tile_bwd(
            do, q, k, v, lse, scale, causal, softmax_d=delta
        )

    def merge(self, o, lse, o_i, lse_i):
        return oracle.scale_out_lse(o, lse, o_i, lse_i)
