#!/usr/bin/env python3
"""Real multi-GPU (RCCL) parity harness — the reference's test CLI
(test/test_burst.py:258-282) rebuilt for this package.  Run on an N-GPU
node:

    torchrun --nnodes 1 --nproc-per-node 8 --master-addr 127.0.0.1 \
        tests/run_multigpu_parity.py --all

Each rank builds the same seeded full sequence, computes the eager
reference with the CPU oracle (float32, full sequence), chunks per rank
(zigzag/striped per variant, test_burst.py:44-58), runs burst_attn_func
through the real RCCL ring, and compares o/dq/dk/dv chunks at the
reference tolerance (rtol=1e-3, atol=1e-2 fp16 — test/checker.py:10).

The logic is identical to tests/test_ring_cpu.py (gloo); this entry
exists so the same parity can be demonstrated over RCCL whenever a
multi-GPU box is available (the per-round driver boxes are 1-GPU).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import oracle
from oracle.partition import get_chunk


def run_case(causal, striped, opt_bwd, deterministic, dtype):
    from burst_attn_amd import burst_attn_func, burst_attn_func_striped
    from burst_attn_amd.comm import gather_obj, print_rank

    rank, world = dist.get_rank(), dist.get_world_size()
    b, s_local, n, d = 2, 256, 8, 128
    s = s_local * world
    g = torch.Generator().manual_seed(20240915)
    q = torch.randn(b, s, n, d, generator=g).to(dtype)
    k = torch.randn(b, s, n, d, generator=g).to(dtype)
    v = torch.randn(b, s, n, d, generator=g).to(dtype)
    do = torch.randn(b, s, n, d, generator=g).to(dtype)
    o_ref, dq_r, dk_r, dv_r = oracle.ring_forward_backward_reference(
        q, k, v, do, None, causal
    )
    zig = causal and not striped
    ch = lambda t: get_chunk(t, 1, rank, world, zigzag=zig, striped=striped)
    qc = ch(q).cuda().requires_grad_()
    kc = ch(k).cuda().requires_grad_()
    vc = ch(v).cuda().requires_grad_()
    func = burst_attn_func_striped if striped else burst_attn_func
    o = func(qc, kc, vc, None, "cuda", causal, opt_bwd, deterministic)
    dq, dk, dv = torch.autograd.grad(o, (qc, kc, vc), ch(do).cuda())
    ok = True
    msgs = []
    for name, got, ref in [("O", o, ch(o_ref)), ("dV", dv, ch(dv_r)),
                           ("dK", dk, ch(dk_r)), ("dQ", dq, ch(dq_r))]:
        try:
            torch.testing.assert_close(got.float().cpu(), ref,
                                       rtol=1e-3, atol=1e-2)
        except AssertionError as e:
            ok = False
            msgs.append(f"{name}: {str(e).splitlines()[0]}")
    res = gather_obj((rank, ok, msgs))
    if rank == 0:
        bad = [r for r in res if not r[1]]
        tag = (f"causal={causal} striped={striped} opt_bwd={opt_bwd} "
               f"det={deterministic}")
        if bad:
            print_rank(f"FAIL  {tag}: {bad}")
            return False
        print_rank(f"PASS  {tag}")
    return True


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--causal", action="store_true")
    p.add_argument("--striped", action="store_true")
    p.add_argument("--optimize_bwd_comm", action="store_true")
    p.add_argument("--deterministic", action="store_true")
    p.add_argument("--all", action="store_true")
    args = p.parse_args()
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    dist.init_process_group("nccl")
    ok = True
    if args.all:
        for causal in (False, True):
            for striped in (False, True):
                for opt in (False, True):
                    ok &= run_case(causal, striped, opt, False, torch.float16)
    else:
        ok = run_case(args.causal, args.striped, args.optimize_bwd_comm,
                      args.deterministic, torch.float16)
    dist.destroy_process_group()
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
