"""Property-based oracle coverage (hypothesis): the blockwise
online-softmax oracle (restating reference burst_utils.py:42-101) must
equal full eager attention for ARBITRARY small shapes, block splits and
scales — beyond the fixed golden-fixture shapes in test_oracle.py."""

import math

import torch
from hypothesis import given, settings, strategies as st

import oracle


@settings(max_examples=40, deadline=None)
@given(
    b=st.integers(1, 2),
    sq=st.integers(1, 96),
    sk=st.integers(1, 96),
    n=st.integers(1, 3),
    d=st.sampled_from([16, 32, 64]),
    q_block=st.integers(8, 64),
    k_block=st.integers(8, 64),
    scale_mul=st.floats(0.25, 4.0),
    seed=st.integers(0, 2**16),
)
def test_oracle_tile_fwd_matches_eager(b, sq, sk, n, d, q_block, k_block,
                                       scale_mul, seed):
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(b, sq, n, d, generator=g)
    k = torch.randn(b, sk, n, d, generator=g)
    v = torch.randn(b, sk, n, d, generator=g)
    scale = scale_mul / math.sqrt(d)
    o, lse = oracle.tile_fwd(q, k, v, scale, False, q_block=q_block,
                             k_block=k_block)
    # eager full attention in the same layout
    qe = q.transpose(1, 2).double()
    ke = k.transpose(1, 2).double()
    ve = v.transpose(1, 2).double()
    s = qe @ ke.transpose(-1, -2) * scale
    o_ref = (torch.softmax(s, -1) @ ve).transpose(1, 2).float()
    lse_ref = torch.logsumexp(s, -1).float()  # [b, n, sq]
    torch.testing.assert_close(o, o_ref, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(lse, lse_ref, rtol=1e-4, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(
    s=st.integers(2, 64),
    n=st.integers(1, 2),
    d=st.sampled_from([32, 64]),
    seed=st.integers(0, 2**16),
)
def test_oracle_tile_bwd_matches_autograd(s, n, d, seed):
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(1, s, n, d, generator=g, requires_grad=True)
    k = torch.randn(1, s, n, d, generator=g, requires_grad=True)
    v = torch.randn(1, s, n, d, generator=g, requires_grad=True)
    do = torch.randn(1, s, n, d, generator=g)
    scale = 1.0 / math.sqrt(d)
    qe = q.transpose(1, 2)
    ke = k.transpose(1, 2)
    ve = v.transpose(1, 2)
    o_ref = (torch.softmax(qe @ ke.transpose(-1, -2) * scale, -1) @ ve
             ).transpose(1, 2)
    dq_a, dk_a, dv_a = torch.autograd.grad(o_ref, (q, k, v), do)
    o, lse = oracle.tile_fwd(q.detach(), k.detach(), v.detach(), scale, False)
    dq, dk, dv = oracle.tile_bwd(do, q.detach(), k.detach(), v.detach(), lse,
                                 scale, False, o=o)
    torch.testing.assert_close(dq, dq_a, rtol=2e-4, atol=2e-5)
    torch.testing.assert_close(dk, dk_a, rtol=2e-4, atol=2e-5)
    torch.testing.assert_close(dv, dv_a, rtol=2e-4, atol=2e-5)
