#!/usr/bin/env python3
"""BurstAttention benchmark — BASELINE.json's headline metric on MI355X.

Metric: attention TFLOP/s per GPU, fwd and fwd+bwd, on the reference's
headline configuration b=1, seq=262144, h=32, d=128 non-causal
(reference README.md:68-85).  FLOPs model is the reference's
(benchmarks/benchmark.py:17-24,204-209): fwd = 4*b*s^2*n*d (/2 causal),
bwd = 2.5x fwd, TFLOP/s/GPU = FLOPs / time / 1e12 / world_size.

Usage (driver contract):
  python bench.py --gpus N --steps K --warmup W
N=1 runs the whole sequence on one GPU (one flash-tile kernel per step);
N>1 is launched by the driver under torchrun (one rank per GPU, RCCL) and
runs the ring with seq/N tokens per rank — total work fixed => strong
scaling.  Rank 0 prints ONE JSON line.
"""

import argparse
import json
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.distributed as dist

PEAK_MFMA_DENSE = 2.5e15  # bf16/f16 dense MFMA peak, MI355X (spec)

# Reference's published per-GPU TFLOPS on ITS OWN hardware (8xA100, fp16,
# b=1 n=32 d=128 non-causal; BASELINE.md <- reference README.md:81-85).
# vs_baseline = our value / this number at the matching seq (context
# numbers — different hardware, same metric and config).
A100_FWD_TFLOPS = {65536: 147.0, 131072: 191.0, 262144: 203.0,
                   524288: 212.0, 1048576: 207.0}
A100_FWDBWD_TFLOPS = {65536: 170.0, 131072: 184.0, 262144: 191.0,
                      524288: 195.0, 1048576: 196.0}


def log(msg):
    if int(os.environ.get("RANK", 0)) == 0:
        print(msg, file=sys.stderr, flush=True)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--seq", type=int, default=262144)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--heads", type=int, default=32)
    p.add_argument("--dim", type=int, default=128)
    p.add_argument("--dtype", choices=["fp16", "bf16"], default="fp16")
    p.add_argument("--causal", action="store_true")
    p.add_argument("--optimize-bwd-comm", action="store_true",
                   help="ring the fp32 delta instead of o in backward")
    p.add_argument("--striped", action="store_true",
                   help="use the striped causal variant (OpBurstAttnStrip)")
    p.add_argument("--no-cpu-baseline", action="store_true")
    p.add_argument("--no-bwd", action="store_true",
                   help="skip the fwd+bwd timing leg")
    return p.parse_args()


def setup_dist(args):
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    if world > 1:
        local = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local)
        dist.init_process_group("nccl")
    else:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29733")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        dist.init_process_group("nccl", rank=0, world_size=1)
    return dist.get_rank(), dist.get_world_size()


def sync_all():
    if dist.get_world_size() > 1:
        dist.barrier()
    torch.cuda.synchronize()


def max_over_ranks(x):
    t = torch.tensor([x], dtype=torch.float64, device="cuda")
    if dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def time_loop(fn, steps, warmup):
    for _ in range(warmup):
        fn()
    sync_all()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    sync_all()
    return max_over_ranks((time.perf_counter() - t0) / steps)


def kernel_event_ms(ext, q, k, v, scale, causal, iters):
    """Average duration of the dominant kernel (the fused fwd flash tile),
    HIP events on the launching stream."""
    ev = [(torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True))
          for _ in range(iters)]
    ext.attn_fwd(q, k, v, scale, causal)  # warm
    torch.cuda.synchronize()
    for s, e in ev:
        s.record()
        ext.attn_fwd(q, k, v, scale, causal)
        e.record()
    torch.cuda.synchronize()
    return sum(s.elapsed_time(e) for s, e in ev) / iters


def cpu_baseline_leg(args, scale):
    """Reference math path (oracle CPU port, burst_utils.py:42-101) timed on
    the host cores: one bounded q-slab of the same workload."""
    import oracle  # checker/baseline only — never the measured GPU path

    cores = min(os.cpu_count() or 1, 16)
    torch.set_num_threads(cores)
    qs = 512 if args.seq >= 65536 else max(64, args.seq // 16)
    g = torch.Generator().manual_seed(1)
    qc = torch.randn(args.batch, qs, args.heads, args.dim, generator=g)
    kc = torch.randn(args.batch, args.seq, args.heads, args.dim, generator=g)
    vc = torch.randn(args.batch, args.seq, args.heads, args.dim, generator=g)
    t0 = time.perf_counter()
    oracle.tile_fwd(qc, kc, vc, scale, False, q_block=512, k_block=2048)
    dt = time.perf_counter() - t0
    flops = 4.0 * args.batch * qs * args.seq * args.heads * args.dim
    return {
        "value": round(flops / dt / 1e12, 4),
        "unit": "TFLOP/s",
        "cores": cores,
        "kind": "port",
        "sample": f"fwd q-slab {qs}x{args.seq} of the same workload "
                  f"(blockwise online-softmax, {dt:.1f}s)",
    }


def main():
    args = parse_args()
    rank, world = setup_dist(args)
    assert args.seq % world == 0
    s_local = args.seq // world
    dtype = torch.float16 if args.dtype == "fp16" else torch.bfloat16
    dev = torch.device("cuda")

    from burst_attn_amd import burst_attn_func, burst_attn_func_striped
    from burst_attn_amd._ext import load_extension

    ext = load_extension()
    scale = 1.0 / math.sqrt(args.dim)

    g = torch.Generator(device="cpu").manual_seed(1234 + rank)
    mk = lambda s: torch.randn(
        args.batch, s, args.heads, args.dim, generator=g
    ).to(dtype).to(dev)
    q, k, v = mk(s_local), mk(s_local), mk(s_local)
    do = mk(s_local)
    log(f"[bench] rank {rank}/{world} s_local={s_local} dtype={args.dtype}")

    causal = args.causal
    attn = burst_attn_func_striped if args.striped else burst_attn_func

    def fwd_step():
        with torch.no_grad():
            attn(q, k, v, None, "cuda", causal)

    def fwdbwd_step():
        qg = q.detach().requires_grad_()
        kg = k.detach().requires_grad_()
        vg = v.detach().requires_grad_()
        o = attn(qg, kg, vg, None, "cuda", causal, args.optimize_bwd_comm)
        torch.autograd.grad(o, (qg, kg, vg), do)

    t_fwd = time_loop(fwd_step, args.steps, args.warmup)
    t_fb = None if args.no_bwd else time_loop(fwdbwd_step, args.steps, args.warmup)

    flops_fwd = 4.0 * args.batch * args.seq**2 * args.heads * args.dim
    if causal:
        flops_fwd /= 2
    tflops_fwd = flops_fwd / t_fwd / 1e12 / world
    tflops_fb = (3.5 * flops_fwd) / t_fb / 1e12 / world if t_fb else None

    # roofline of the dominant kernel (local fwd tile), rank 0
    roofline = None
    if rank == 0:
        kms = kernel_event_ms(ext, q, k, v, scale, causal, iters=3)
        # algorithmic flops of ONE launch: this rank's q rows vs its
        # current kv (per ring step per GPU)
        launch_flops = 4.0 * args.batch * s_local * s_local * args.heads * args.dim
        if causal:
            launch_flops /= 2
        achieved = launch_flops / (kms / 1e3)
        # HBM bytes per launch: measured offline with rocprofv3 --pmc
        # FETCH_SIZE / WRITE_SIZE in separate passes, FETCH doubled per the
        # gfx950 wide-read undercount (MI355X_MICROARCH.md §HBM); see
        # profiles/hbm_traffic.json (regenerated per round)
        traffic = None
        try:
            with open(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                   "profiles", "hbm_traffic.json")) as f:
                tj = json.load(f)
            traffic = tj.get("fwd_launch_bytes", {}).get(
                f"b{args.batch}_s{s_local}_h{args.heads}_d{args.dim}_{args.dtype}")
        except OSError:
            pass
        roofline = {
            "bound": "mfma",
            "achieved": round(achieved / 1e12, 2),
            "peak": round(PEAK_MFMA_DENSE / 1e12, 2),
            "unit": "TFLOP/s",
            "frac": round(achieved / PEAK_MFMA_DENSE, 4),
            "traffic": traffic,
        }

    # ring comm/compute overlap evidence (N>1): compare the full step
    # against (a) compute-only (the W local tile calls, no ring) and
    # (b) comm-only (the W-1 k/v ring hops, no compute).  overlap_frac =
    # 1 - exposed/comm_only, where exposed = full - compute_only is the
    # comm time NOT hidden under the tile kernels (north_star: >=90%).
    ring_stats = None
    try:
      if world > 1:
        from burst_attn_amd.comm import Ring
        from burst_attn_amd.tile import get_tile_provider

        P = get_tile_provider()

        def compute_only_step():
            with torch.no_grad():
                state = None
                for _ in range(world):
                    state = P.fwd_accum(state, q, k, v, scale, False)
                P.fwd_finalize(state, dtype)

        bufs = [torch.empty_like(k), torch.empty_like(v)]

        def comm_only_step():
            ring = Ring(None, (None, None))
            for r in range(1, world):
                ring.double_ring_send_recv([k, v], bufs, r)
                ring.commit()
                ring.wait()

        t_comp = time_loop(compute_only_step, max(2, args.steps), args.warmup)
        t_comm = time_loop(comm_only_step, max(4, 2 * args.steps), args.warmup)
        exposed = max(0.0, t_fwd - t_comp)
        ring_stats = {
            "compute_only_ms": round(t_comp * 1e3, 2),
            "comm_only_ms": round(t_comm * 1e3, 2),
            "exposed_comm_ms": round(exposed * 1e3, 2),
            "overlap_frac": round(max(0.0, min(1.0, 1.0 - exposed / t_comm)), 4)
            if t_comm > 0 else None,
        }
    except Exception as e:  # never let the overlap probe kill the bench
        log(f"[bench] ring overlap probe failed: {e!r}")
        ring_stats = {"error": repr(e)}

    cpu_baseline = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline:
        log("[bench] timing CPU baseline (oracle port) ...")
        cpu_baseline = cpu_baseline_leg(args, scale)

    if rank == 0:
        # vs the reference's published number for this exact metric/config
        # (8xA100 fp16; context numbers — BASELINE.md)
        ref_fwd = (
            A100_FWD_TFLOPS.get(args.seq)
            if (args.batch == 1 and args.heads == 32 and args.dim == 128
                and not causal and not args.striped)
            else None
        )
        ref_fb = (
            A100_FWDBWD_TFLOPS.get(args.seq) if ref_fwd is not None else None
        )
        out = {
            "metric": "attention_fwd_TFLOPs_per_GPU",
            "value": round(tflops_fwd, 2),
            "unit": "TFLOP/s/GPU",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(t_fwd * 1e3, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": round(tflops_fwd / ref_fwd, 3) if ref_fwd else None,
            "vs_baseline_ref": (
                f"reference 8xA100 fwd {ref_fwd} TFLOP/s/GPU at seq={args.seq} "
                "(README.md:81-85)" if ref_fwd else None
            ),
            "fwd_bwd_vs_baseline": (
                round(tflops_fb / ref_fb, 3) if (tflops_fb and ref_fb) else None
            ),
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "workload": f"b={args.batch} seq={args.seq} h={args.heads} "
                            f"d={args.dim} {'causal' if causal else 'non-causal'}",
                "parallelism": f"ring sequence parallel, sp{world}"
                               + (" striped" if args.striped else ""),
            },
            "fwd_bwd_TFLOPs_per_GPU": round(tflops_fb, 2) if tflops_fb else None,
            "fwd_bwd_ms_per_step": round(t_fb * 1e3, 2) if t_fb else None,
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
            "ring": ring_stats,
        }
        print(json.dumps(out), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
