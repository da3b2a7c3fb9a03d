"""Oracle vs the reference's own tile math (golden fixtures) and vs eager.

The golden fixtures under tests/golden/ were produced by
``oracle/gen_golden.py`` executing the reference's pure-torch tile
functions (burst_utils.py:20-33,42-101) in the build container — these
tests pin the CPU oracle to the reference without needing /root/reference
at run time."""

import glob
import math
import os

import numpy as np
import pytest
import torch

import oracle
from .conftest import GOLDEN_DIR

# reference test tolerance: rtol=1e-3, atol=1e-2 (test/checker.py:10) is for
# fp16; the oracle runs fp32 against fp32 golden vectors, so hold it tighter.
RTOL, ATOL = 1e-5, 1e-5


def _tile_fixtures():
    return sorted(glob.glob(os.path.join(GOLDEN_DIR, "tile_*.npz")))


def _merge_fixtures():
    return sorted(glob.glob(os.path.join(GOLDEN_DIR, "merge_*.npz")))


@pytest.mark.parametrize("path", _tile_fixtures())
def test_tile_fwd_matches_reference_golden(path):
    z = np.load(path)
    # golden layout [B,N,S,D] -> flash layout [B,S,N,D]
    to_flash = lambda a: torch.from_numpy(a).permute(0, 2, 1, 3).contiguous()
    q, k, v = to_flash(z["q"]), to_flash(z["k"]), to_flash(z["v"])
    scale = float(z["scale"])
    o, lse = oracle.tile_fwd(q, k, v, scale, causal=False)
    o_ref = to_flash(z["o"])
    torch.testing.assert_close(o, o_ref, rtol=RTOL, atol=ATOL)
    # reference math-path lse carries a +1e-5 regulariser
    # (burst_utils.py:71-73): log(l+1e-5)+m vs our exact log(l)+m
    lse_ref = torch.from_numpy(z["lse"])
    torch.testing.assert_close(lse, lse_ref, rtol=1e-5, atol=1e-4)


@pytest.mark.parametrize("path", _tile_fixtures())
def test_tile_bwd_matches_reference_golden(path):
    z = np.load(path)
    to_flash = lambda a: torch.from_numpy(a).permute(0, 2, 1, 3).contiguous()
    q, k, v, do, o = (
        to_flash(z["q"]), to_flash(z["k"]), to_flash(z["v"]),
        to_flash(z["do"]), to_flash(z["o"]),
    )
    lse = torch.from_numpy(z["lse"])  # [B,N,S]
    scale = float(z["scale"])
    dq, dk, dv = oracle.tile_bwd(do, q, k, v, lse, scale, causal=False, o=o)
    torch.testing.assert_close(dq, to_flash(z["dq"]), rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dk, to_flash(z["dk"]), rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dv, to_flash(z["dv"]), rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("path", _merge_fixtures())
def test_merge_matches_reference_golden(path):
    z = np.load(path)
    o = torch.from_numpy(z["o"]).clone()
    lse = torch.from_numpy(z["lse"]).clone()
    o_i = torch.from_numpy(z["o_i"])
    lse_i = torch.from_numpy(z["lse_i"])
    o_m, lse_m = oracle.scale_out_lse(o, lse, o_i, lse_i)
    torch.testing.assert_close(o_m, torch.from_numpy(z["o_merged"]), rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(lse_m, torch.from_numpy(z["lse_merged"]), rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("b,s,n,d", [(1, 128, 2, 64), (2, 96, 3, 128)])
def test_tile_fwd_vs_eager(b, s, n, d, causal):
    g = torch.Generator().manual_seed(s * d + causal)
    q = torch.randn(b, s, n, d, generator=g)
    k = torch.randn(b, s, n, d, generator=g)
    v = torch.randn(b, s, n, d, generator=g)
    o, lse = oracle.tile_fwd(q, k, v, None, causal, q_block=48, k_block=40)
    o_ref = oracle.eager_attention(q, k, v, None, causal)
    torch.testing.assert_close(o, o_ref, rtol=1e-5, atol=1e-5)
    # lse property: softmax denominator
    scale = 1.0 / math.sqrt(d)
    s_mat = torch.einsum("bsnd,btnd->bnst", q.float(), k.float()) * scale
    if causal:
        mask = torch.ones(s, s, dtype=torch.bool).triu(1)
        s_mat = s_mat.masked_fill(mask, float("-inf"))
    lse_ref = torch.logsumexp(s_mat, dim=-1)
    torch.testing.assert_close(lse, lse_ref, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("causal", [False, True])
def test_tile_bwd_vs_autograd(causal):
    b, s, n, d = 1, 96, 2, 64
    g = torch.Generator().manual_seed(7)
    q = torch.randn(b, s, n, d, generator=g, requires_grad=True)
    k = torch.randn(b, s, n, d, generator=g, requires_grad=True)
    v = torch.randn(b, s, n, d, generator=g, requires_grad=True)
    do = torch.randn(b, s, n, d, generator=g)
    o_ref = oracle.eager_attention(q, k, v, None, causal)
    dq_ref, dk_ref, dv_ref = torch.autograd.grad(o_ref, (q, k, v), do)
    o, lse = oracle.tile_fwd(q.detach(), k.detach(), v.detach(), None, causal)
    dq, dk, dv = oracle.tile_bwd(
        do, q.detach(), k.detach(), v.detach(), lse, None, causal, o=o,
        q_block=40, k_block=56,
    )
    torch.testing.assert_close(dq, dq_ref, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dk, dk_ref, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dv, dv_ref, rtol=1e-4, atol=1e-4)


def test_merge_equals_joint_softmax():
    """Splitting kv in two and LSE-merging the partial tiles must equal
    attention over the concatenated kv (the invariant the ring relies on,
    burst_attn_interface.py:214-242)."""
    b, s, n, d = 1, 64, 2, 32
    g = torch.Generator().manual_seed(3)
    q = torch.randn(b, s, n, d, generator=g)
    k = torch.randn(b, 2 * s, n, d, generator=g)
    v = torch.randn(b, 2 * s, n, d, generator=g)
    o1, lse1 = oracle.tile_fwd(q, k[:, :s], v[:, :s])
    o2, lse2 = oracle.tile_fwd(q, k[:, s:], v[:, s:])
    o = o1.to(torch.float32)
    lse = lse1.transpose(-2, -1).unsqueeze(-1).contiguous()
    o, lse = oracle.scale_out_lse(o, lse, o2, lse2)
    o_ref = oracle.eager_attention(q, k, v)
    torch.testing.assert_close(o, o_ref, rtol=1e-5, atol=1e-5)


def test_partition_roundtrip():
    from oracle.partition import get_chunk, unchunk

    t = torch.arange(4 * 64 * 3).reshape(4, 64, 3).float()
    for kw in ({"zigzag": False}, {"zigzag": True}, {"striped": True}):
        chunks = [get_chunk(t, 1, r, 4, **kw) for r in range(4)]
        torch.testing.assert_close(unchunk(chunks, 1, **kw), t)


@pytest.mark.parametrize("causal,striped", [
    (False, False), (True, False), (True, True),
])
def test_ring_sim_cpu_oracle_provider(causal, striped):
    """Sanity of the test-side ring simulator itself (oracle tiles, W=4)."""
    from .cpu_tile_provider import OracleTileProvider
    from .ring_sim import simulate_ring

    W = 4
    b, s, n, d = 1, 64 * W, 2, 32
    g = torch.Generator().manual_seed(66)
    q = torch.randn(b, s, n, d, generator=g)
    k = torch.randn(b, s, n, d, generator=g)
    v = torch.randn(b, s, n, d, generator=g)
    do = torch.randn(b, s, n, d, generator=g)
    P = OracleTileProvider()
    o, dq, dk, dv = simulate_ring(P, q, k, v, do, W, 1.0 / math.sqrt(d), causal, striped)
    o_ref, dq_r, dk_r, dv_r = oracle.ring_forward_backward_reference(q, k, v, do, None, causal)
    torch.testing.assert_close(o, o_ref, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dq, dq_r, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dk, dk_r, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dv, dv_r, rtol=1e-4, atol=1e-4)
