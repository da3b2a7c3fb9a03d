"""Multi-process (gloo, CPU) correctness of the ring orchestration.

Recreates the reference's integration test (test/test_burst.py:159-219) on
CPU: every rank builds the same seeded full sequence, runs single-process
eager attention as the oracle, chunks per rank (plain / zigzag / striped,
test_burst.py:44-58), runs burst_attn_func through the REAL orchestration
(interface + Ring over gloo) with the oracle tile provider injected, and
compares output and dq/dk/dv chunks.

This covers the N>1 product path (round structure, payload layouts, the
travelling-dq second ring, zigzag/striped bookkeeping) without a GPU —
the HIP tile itself is covered by tests/test_gpu_parity.py.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle
from oracle.partition import get_chunk

WORLD = 2
RTOL, ATOL = 1e-4, 1e-4


def _worker(rank, world, port, causal, striped, optimize_bwd_comm, deterministic, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from burst_attn_amd.tile import _set_tile_provider_for_testing
        from burst_attn_amd import burst_attn_func, burst_attn_func_striped
        from .cpu_tile_provider import OracleTileProvider

        _set_tile_provider_for_testing(OracleTileProvider())

        b, s_local, n, d = 2, 64, 2, 64
        s = s_local * world
        g = torch.Generator().manual_seed(20240915)
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
        o = func(qc, kc, vc, None, "cuda", causal, optimize_bwd_comm, deterministic)
        dq, dk, dv = torch.autograd.grad(o, (qc, kc, vc), ch(do))

        torch.testing.assert_close(o, ch(o_ref), rtol=RTOL, atol=ATOL)
        torch.testing.assert_close(dv, ch(dv_ref), rtol=RTOL, atol=ATOL)
        torch.testing.assert_close(dk, ch(dk_ref), rtol=RTOL, atol=ATOL)
        torch.testing.assert_close(dq, ch(dq_ref), rtol=RTOL, atol=ATOL)
        dist.destroy_process_group()
    except Exception as e:  # surface the real failure to the parent
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


_PORT = [29601]


@pytest.mark.parametrize("deterministic", [False])
@pytest.mark.parametrize("optimize_bwd_comm", [False, True])
@pytest.mark.parametrize("causal,striped", [
    (False, False),
    (True, False),   # zigzag
    (True, True),    # striped
    (False, True),   # striped layout, non-causal
])
def test_ring_matches_full_attention(causal, striped, optimize_bwd_comm, deterministic):
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    fail_q = ctx.SimpleQueue()
    procs = []
    try:
        mp.spawn(
            _worker,
            args=(WORLD, _PORT[0], causal, striped, optimize_bwd_comm,
                  deterministic, fail_q),
            nprocs=WORLD,
            join=True,
        )
    except Exception:
        msgs = []
        while not fail_q.empty():
            msgs.append(fail_q.get())
        raise AssertionError("ring test failed:\n" + "\n".join(msgs))


def test_world_size_one_no_comm():
    """W=1 degenerates to one local tile — must work without any P2P."""
    import subprocess, sys
    code = r"""
import os, sys, torch
import torch.distributed as dist
sys.path.insert(0, os.getcwd())
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29599")
dist.init_process_group("gloo", rank=0, world_size=1)
from burst_attn_amd.tile import _set_tile_provider_for_testing
from burst_attn_amd import burst_attn_func
from tests.cpu_tile_provider import OracleTileProvider
import oracle
_set_tile_provider_for_testing(OracleTileProvider())
g = torch.Generator().manual_seed(5)
q = torch.randn(1, 64, 2, 64, generator=g, requires_grad=True)
k = torch.randn(1, 64, 2, 64, generator=g, requires_grad=True)
v = torch.randn(1, 64, 2, 64, generator=g, requires_grad=True)
do = torch.randn(1, 64, 2, 64, generator=g)
o = burst_attn_func(q, k, v, None, "cuda", True)
dq, dk, dv = torch.autograd.grad(o, (q, k, v), do)
o_ref, dq_r, dk_r, dv_r = oracle.ring_forward_backward_reference(q, k, v, do, None, True)
torch.testing.assert_close(o, o_ref.to(o.dtype), rtol=1e-4, atol=1e-4)
torch.testing.assert_close(dq, dq_r, rtol=1e-4, atol=1e-4)
print("OK")
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0 and "OK" in r.stdout, r.stdout + r.stderr


@pytest.mark.parametrize("causal,striped", [(True, False), (True, True)])
def test_ring_odd_world_size(causal, striped):
    """W=3: odd ring size exercises the even/odd P2P ordering with two
    adjacent even ranks and non-power-of-two chunking."""
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    fail_q = ctx.SimpleQueue()
    try:
        mp.spawn(
            _worker,
            args=(3, _PORT[0], causal, striped, False, False, fail_q),
            nprocs=3,
            join=True,
        )
    except Exception:
        msgs = []
        while not fail_q.empty():
            msgs.append(fail_q.get())
        raise AssertionError("odd-ring test failed:\n" + "\n".join(msgs))
