"""gloo (CPU) parity of the RingQK/RingAV comparator
(benchmarks/ring_naive.py, restating reference benchmarks/ring_attn.py)
against full-sequence eager attention — the comparator must be CORRECT to
be a fair benchmark column (the upstream version cannot even run: its
`_ring` helper is broken)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _worker(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from benchmarks.ring_naive import ring_naive_attention

        bn, s_local, d = 4, 32, 64
        s = s_local * world
        g = torch.Generator().manual_seed(77)
        q = torch.randn(bn, s, d, generator=g)
        k = torch.randn(bn, s, d, generator=g)
        v = torch.randn(bn, s, d, generator=g)
        do = torch.randn(bn, s, d, generator=g)
        scale = 1.0 / (d ** 0.5)

        # eager full-sequence reference
        qf = q.clone().requires_grad_()
        kf = k.clone().requires_grad_()
        vf = v.clone().requires_grad_()
        o_ref = torch.softmax(qf @ kf.transpose(1, 2) * scale, -1) @ vf
        dq_r, dk_r, dv_r = torch.autograd.grad(o_ref, (qf, kf, vf), do)

        sl = slice(rank * s_local, (rank + 1) * s_local)
        qc = q[:, sl].contiguous().requires_grad_()
        kc = k[:, sl].contiguous().requires_grad_()
        vc = v[:, sl].contiguous().requires_grad_()
        o = ring_naive_attention(qc, kc, vc, scale)
        dq, dk, dv = torch.autograd.grad(o, (qc, kc, vc), do[:, sl].contiguous())

        tol = dict(rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(o, o_ref.detach()[:, sl], **tol)
        torch.testing.assert_close(dv, dv_r[:, sl], **tol)
        torch.testing.assert_close(dk, dk_r[:, sl], **tol)
        torch.testing.assert_close(dq, dq_r[:, sl], **tol)
        dist.destroy_process_group()
    except Exception as e:
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_ring_naive_matches_eager():
    ctx = mp.get_context("spawn")
    fail_q = ctx.SimpleQueue()
    try:
        mp.spawn(_worker, args=(WORLD, 29811, fail_q), nprocs=WORLD, join=True)
    except Exception:
        msgs = []
        while not fail_q.empty():
            msgs.append(fail_q.get())
        raise AssertionError("ring_naive test failed:\n" + "\n".join(msgs))
