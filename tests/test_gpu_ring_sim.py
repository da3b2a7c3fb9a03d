"""Multi-round ring numerics on ONE GPU: the real HIP kernels driven
through W=4 virtual ranks (tests/ring_sim.py) vs full-sequence eager
attention.  Catches zigzag/striped slice+merge bookkeeping and
multi-round LSE-merge numerics that the W=1 end-to-end test cannot."""

import math

import pytest
import torch

import oracle
from .ring_sim import simulate_ring

pytestmark = pytest.mark.gpu


class _GpuProvider:
    def __init__(self):
        from burst_attn_amd.tile import HipTileProvider

        self._p = HipTileProvider()

    def fwd(self, q, k, v, scale, causal):
        return self._p.fwd(q, k, v, scale, causal)

    def bwd_preprocess(self, o, do):
        return self._p.bwd_preprocess(o, do)

    def bwd(self, do, q, k, v, delta, lse, scale, causal, det):
        return self._p.bwd(do, q, k, v, delta, lse, scale, causal, det)

    def merge(self, o, lse, o_i, lse_i):
        return self._p.merge(o, lse, o_i, lse_i)


@pytest.mark.parametrize("causal,striped", [
    (False, False), (True, False), (True, True), (False, True),
])
def test_ring_sim_w4_gpu(causal, striped):
    W = 4
    b, s, n, d = 1, 256 * W, 2, 128
    dtype = torch.float16
    g = torch.Generator().manual_seed(55)
    q = torch.randn(b, s, n, d, generator=g).to(dtype).cuda()
    k = torch.randn(b, s, n, d, generator=g).to(dtype).cuda()
    v = torch.randn(b, s, n, d, generator=g).to(dtype).cuda()
    do = torch.randn(b, s, n, d, generator=g).to(dtype).cuda()
    scale = 1.0 / math.sqrt(d)
    P = _GpuProvider()
    o, dq, dk, dv = simulate_ring(P, q, k, v, do, W, scale, causal, striped)
    o_ref, dq_r, dk_r, dv_r = oracle.ring_forward_backward_reference(
        q.cpu(), k.cpu(), v.cpu(), do.cpu(), scale, causal
    )
    tol = dict(rtol=2e-3, atol=1e-2)
    btol = dict(rtol=5e-3, atol=3e-2)
    torch.testing.assert_close(o.float().cpu(), o_ref, **tol)
    torch.testing.assert_close(dv.float().cpu(), dv_r, **btol)
    torch.testing.assert_close(dk.float().cpu(), dk_r, **btol)
    torch.testing.assert_close(dq.float().cpu(), dq_r, **btol)
