#!/usr/bin/env python3
"""Benchmark sweep — reproduces the reference's measurement methodology
(benchmarks/benchmark.py: warmup + timed mean, FLOPs model :17-24, per-GPU
TFLOPS = FLOPs/time/1e12/world_size :204-209, jsonlines rows :286-298).

Single node:
  python benchmarks/sweep.py                      # 1 GPU
  torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
      benchmarks/sweep.py --configs ring          # 8-GPU ring rows

Row fields mirror the reference's results_torch.jsonl semantics.
"""

import argparse
import json
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def flops_fwd(b, s, n, d, causal):
    f = 4.0 * b * s * s * n * d  # reference benchmark.py:17-24
    return f / 2 if causal else f


def timeit(fn, steps, warmup):
    for _ in range(warmup):
        fn()
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.barrier()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    t = torch.tensor([dt], dtype=torch.float64, device="cuda")
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def bench_burst(args, b, s_global, n, d, causal, striped, opt_bwd, steps, warmup):
    from burst_attn_amd import burst_attn_func, burst_attn_func_striped

    world = dist.get_world_size()
    rank = dist.get_rank()
    s_local = s_global // world
    dtype = torch.float16 if args.dtype == "fp16" else torch.bfloat16
    g = torch.Generator().manual_seed(1000 + rank)
    mk = lambda: torch.randn(b, s_local, n, d, generator=g).to(dtype).cuda()
    q, k, v, do = mk(), mk(), mk(), mk()
    func = burst_attn_func_striped if striped else burst_attn_func

    def fwd():
        with torch.no_grad():
            func(q, k, v, None, "cuda", causal, opt_bwd)

    def fwdbwd():
        qg, kg, vg = (t.detach().requires_grad_() for t in (q, k, v))
        o = func(qg, kg, vg, None, "cuda", causal, opt_bwd)
        torch.autograd.grad(o, (qg, kg, vg), do)

    t_f = timeit(fwd, steps, warmup)
    t_fb = timeit(fwdbwd, steps, warmup)
    f = flops_fwd(b, s_global, n, d, causal)
    return {
        "method": "burst_striped" if striped else "burst",
        "b": b, "s": s_global, "n": n, "d": d,
        "causal": causal, "opt_bwd": opt_bwd, "wsize": world,
        "dtype": args.dtype,
        "fwd_ms": round(t_f * 1e3, 2),
        "fwd_tflops_per_gpu": round(f / t_f / 1e12 / world, 2),
        "fwdbwd_ms": round(t_fb * 1e3, 2),
        "fwdbwd_tflops_per_gpu": round(3.5 * f / t_fb / 1e12 / world, 2),
    }


def bench_single_flash(args, b, s, n, d, causal, steps, warmup):
    """Single-GPU full-sequence tile — the reference README's
    'flash single GPU' comparator column, using our own fwd kernel."""
    from burst_attn_amd._ext import load_extension

    ext = load_extension()
    dtype = torch.float16 if args.dtype == "fp16" else torch.bfloat16
    g = torch.Generator().manual_seed(7)
    mk = lambda: torch.randn(b, s, n, d, generator=g).to(dtype).cuda()
    q, k, v = mk(), mk(), mk()
    scale = 1.0 / math.sqrt(d)
    fn = lambda: ext.attn_fwd(q, k, v, scale, causal)
    t_f = timeit(fn, steps, warmup)
    f = flops_fwd(b, s, n, d, causal)
    return {
        "method": "flash_single_gpu", "b": b, "s": s, "n": n, "d": d,
        "causal": causal, "wsize": 1, "dtype": args.dtype,
        "fwd_ms": round(t_f * 1e3, 2),
        "fwd_tflops_per_gpu": round(f / t_f / 1e12, 2),
    }


def bench_ring_naive(args, b, s_global, n, d, steps, warmup):
    """RingQK/RingAV comparator column (reference benchmarks/ring_attn.py;
    restated in benchmarks/ring_naive.py).  Materialises the full score
    slab — configs must fit O(S^2/W) per rank."""
    from benchmarks.ring_naive import ring_naive_attention, slab_bytes

    world = dist.get_world_size()
    rank = dist.get_rank()
    s_local = s_global // world
    dtype = torch.float16 if args.dtype == "fp16" else torch.bfloat16
    need = 4 * slab_bytes(b, n, s_local, s_global, dtype)  # slab+probs+grads
    free = torch.cuda.mem_get_info()[0]
    if need > free * 0.6:
        return None
    g = torch.Generator().manual_seed(2000 + rank)
    mk = lambda: torch.randn(b * n, s_local, d, generator=g).to(dtype).cuda()
    q, k, v, do = mk(), mk(), mk(), mk()
    scale = 1.0 / math.sqrt(d)

    def fwd():
        with torch.no_grad():
            ring_naive_attention(q, k, v, scale)

    def fwdbwd():
        qg, kg, vg = (t.detach().requires_grad_() for t in (q, k, v))
        o = ring_naive_attention(qg, kg, vg, scale)
        torch.autograd.grad(o, (qg, kg, vg), do)

    t_f = timeit(fwd, steps, warmup)
    t_fb = timeit(fwdbwd, steps, warmup)
    f = flops_fwd(b, s_global, n, d, False)
    return {
        "method": "ring_naive", "b": b, "s": s_global, "n": n, "d": d,
        "causal": False, "wsize": world, "dtype": args.dtype,
        "fwd_ms": round(t_f * 1e3, 2),
        "fwd_tflops_per_gpu": round(f / t_f / 1e12 / world, 2),
        "fwdbwd_ms": round(t_fb * 1e3, 2),
        "fwdbwd_tflops_per_gpu": round(3.5 * f / t_fb / 1e12 / world, 2),
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--configs",
                   choices=["quick", "ring", "batch", "single", "naive"],
                   default="quick")
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--dtype", choices=["fp16", "bf16"], default="fp16")
    p.add_argument("--out", default="results_mi355x.jsonl")
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        dist.init_process_group("nccl")
    else:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29744")
        dist.init_process_group("nccl", rank=0, world_size=1)
    rank = dist.get_rank()

    rows = []
    n, d = 32, 128
    if args.configs == "quick":  # 1-GPU-friendly sweep
        for s in (16384, 32768, 65536):
            for causal in (False, True):
                rows.append(bench_burst(args, 1, s, n, d, causal, False, causal,
                                        args.steps, args.warmup))
        if world == 1:
            rows.append(bench_single_flash(args, 1, 65536, n, d, False,
                                           args.steps, args.warmup))
    elif args.configs == "ring":  # the README seq sweep (per world size)
        for s in (65536, 131072, 262144, 524288):
            rows.append(bench_burst(args, 1, s, n, d, False, False, False,
                                    args.steps, args.warmup))
        rows.append(bench_burst(args, 1, 524288, n, d, True, False, True,
                                args.steps, args.warmup))  # config 4
    elif args.configs == "batch":  # README batch scaling at s=65536
        for b in (1, 2, 4, 8):
            rows.append(bench_burst(args, b, 65536, n, d, True, False, False,
                                    args.steps, args.warmup))
    elif args.configs == "single":
        for s in (65536, 131072, 262144):
            rows.append(bench_single_flash(args, 1, s, n, d, False,
                                           args.steps, args.warmup))
    elif args.configs == "naive":  # RingQK/RingAV comparator column
        for s in (8192, 16384, 32768):
            r = bench_ring_naive(args, 1, s, n, d, args.steps, args.warmup)
            if r is not None:
                rows.append(r)
            r = bench_burst(args, 1, s, n, d, False, False, False,
                            args.steps, args.warmup)
            rows.append(r)

    if rank == 0:
        with open(args.out, "a") as f:
            for r in rows:
                f.write(json.dumps(r) + "\n")
                print(json.dumps(r), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
