"""Two-level (double-ring) correctness on CPU: 4 gloo ranks arranged as
2 "nodes" x 2 ranks, group construction exactly like the reference's
``test/test_burst.py:120-156`` (``get_group``: intra = arange(W).reshape
(-1, local); inter = transpose; plus duplicated dq groups), checked
against full-sequence eager attention."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle
from oracle.partition import get_chunk

WORLD = 4
LOCAL = 2
RTOL, ATOL = 1e-4, 1e-4


def _make_groups(create_dq_group, world=WORLD, local=LOCAL):
    ranks = np.arange(world).reshape(-1, local)
    intra = [dist.new_group(list(r)) for r in ranks]
    inter = [dist.new_group(list(r)) for r in ranks.T]
    me = dist.get_rank()
    my_intra = intra[me // local]
    my_inter = inter[me % local]
    if not create_dq_group:
        return my_intra, my_inter
    intra2 = [dist.new_group(list(r)) for r in ranks]
    inter2 = [dist.new_group(list(r)) for r in ranks.T]
    return (my_intra, intra2[me // local]), (my_inter, inter2[me % local])


def _worker(rank, world, port, causal, striped, opt_bwd, dq_groups, fail_q,
            local=LOCAL):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from burst_attn_amd.tile import _set_tile_provider_for_testing
        from burst_attn_amd import burst_attn_func, burst_attn_func_striped
        from .cpu_tile_provider import OracleTileProvider

        _set_tile_provider_for_testing(OracleTileProvider())
        intra_g, inter_g = _make_groups(dq_groups, world, local)

        b, s_local, n, d = 1, 64, 2, 32
        s = s_local * world
        g = torch.Generator().manual_seed(424242)
        q = torch.randn(b, s, n, d, generator=g)
        k = torch.randn(b, s, n, d, generator=g)
        v = torch.randn(b, s, n, d, generator=g)
        do = torch.randn(b, s, n, d, generator=g)
        o_ref, dq_ref, dk_ref, dv_ref = oracle.ring_forward_backward_reference(
            q, k, v, do, None, causal
        )
        zig = causal and not striped
        ch = lambda t: get_chunk(t, 1, rank, world, zigzag=zig, striped=striped)
        qc = ch(q).requires_grad_()
        kc = ch(k).requires_grad_()
        vc = ch(v).requires_grad_()
        func = burst_attn_func_striped if striped else burst_attn_func
        o = func(qc, kc, vc, None, "cuda", causal, opt_bwd, False, None,
                 [intra_g, inter_g])
        dq, dk, dv = torch.autograd.grad(o, (qc, kc, vc), ch(do))
        torch.testing.assert_close(o, ch(o_ref), rtol=RTOL, atol=ATOL)
        torch.testing.assert_close(dv, ch(dv_ref), rtol=RTOL, atol=ATOL)
        torch.testing.assert_close(dk, ch(dk_ref), rtol=RTOL, atol=ATOL)
        torch.testing.assert_close(dq, ch(dq_ref), rtol=RTOL, atol=ATOL)
        dist.destroy_process_group()
    except Exception as e:
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


_PORT = [29650]


@pytest.mark.parametrize("causal,striped,opt_bwd,dq_groups", [
    (False, False, False, False),
    (False, False, False, True),   # separate dq ring groups
    (True, False, False, True),    # zigzag causal
    (True, True, False, True),     # striped causal
    (True, False, True, True),     # optimize_bwd_comm
])
def test_double_ring_matches_full_attention(causal, striped, opt_bwd, dq_groups):
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    fail_q = ctx.SimpleQueue()
    try:
        mp.spawn(
            _worker,
            args=(WORLD, _PORT[0], causal, striped, opt_bwd, dq_groups, fail_q),
            nprocs=WORLD,
            join=True,
        )
    except Exception:
        msgs = []
        while not fail_q.empty():
            msgs.append(fail_q.get())
        raise AssertionError("double-ring test failed:\n" + "\n".join(msgs))


@pytest.mark.parametrize("world,local", [(6, 3), (6, 2)])
def test_double_ring_rectangular(world, local):
    """Non-square hierarchies (3 "nodes" x 2 ranks and 2 x 3) — the
    intra/inter phase arithmetic must hold when intra_size != inter_size
    (reference comm.py:221-254 staging)."""
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    fail_q = ctx.SimpleQueue()
    try:
        mp.spawn(
            _worker,
            args=(world, _PORT[0], True, False, False, True, fail_q, local),
            nprocs=world,
            join=True,
        )
    except Exception:
        msgs = []
        while not fail_q.empty():
            msgs.append(fail_q.get())
        raise AssertionError("double-ring test failed:\n" + "\n".join(msgs))
