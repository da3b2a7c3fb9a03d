"""BurstAttention ring orchestration — public API and autograd layer.

Drop-in for the reference's ``burst_attn/burst_attn_interface.py``:
``burst_attn_func`` / ``burst_attn_func_striped`` keep the exact signature
and torch.autograd semantics (reference ``:135-158`` / ``:109-132``; grads
for q, k, v only, ``:398``).  Layout is the flash layout ``[B, S, N, H]``
(class docstring ``:162-168``); ``process_group`` is a torch.distributed
ProcessGroup (None → WORLD).

MI355X-first redesign decisions (vs the reference):
  * ONE local-tile backend — the gfx950 HIP kernel pair behind
    ``tile.get_tile_provider()`` — instead of the reference's
    cuda/triton/math dispatch (``:40-51``).  The ``flash`` argument is
    accepted for signature compatibility; the math-path layout
    (``flash`` not in {"cuda","triton"}; [B,N,S,D]) is not supported.
  * dq/dk/dv accumulate in fp32 across ring rounds (the reference
    accumulates in the input dtype, fp16) and are cast to the input dtype
    at the end.
  * ring payloads and round structure are otherwise identical —
    forward rings {k, v} (``:214-242``); backward rings
    {delta-or-o, grad_output, q, lse} plus a separate travelling-dq ring
    with one extra final hop (``:291-396``).

Causal load-balancing layouts (see oracle/partition.py):
  * ``OpBurstAttn``     — zigzag: rank r holds global chunks [r] and
    [2W-1-r]; per-round half-tile dispatch per ``:221-235`` (fwd) and
    ``:303-367`` (bwd).
  * ``OpBurstAttnStrip`` — striped: token t on rank t mod W; per-round
    one-token shift per ``:459-475`` (fwd) and ``:557-585`` (bwd).
"""

import math
import os

import torch

from .comm import Ring, get_rank, replicate
from .log_helper import get_logger
from .tile import get_tile_provider

_logger = get_logger(__name__, level="WARN")

__all__ = ["burst_attn_func", "burst_attn_func_striped", "OpBurstAttn", "OpBurstAttnStrip"]


def get_partition_id(double_group, r):
    """Source index of the K/V (fwd) or Q (bwd) chunk held at round r.

    Single ring: r-1 (equivalent to "source rank <= my rank" in the
    causal dispatch).  Double ring: the source chunk's global rank, from
    the intra/inter positions (reference burst_attn_interface.py:20-37).
    """
    if not double_group or double_group[0] is None:
        return r - 1
    import torch.distributed as dist

    intra = dist.get_world_size(double_group[0])
    inter = dist.get_world_size(double_group[1])
    ii = dist.get_rank(double_group[0])
    ir = dist.get_rank(double_group[1])
    return ((ir - (r - 1) // intra) % inter) * intra + (
        (ii - (r - 1) % intra) % intra
    )


def _record_stream(*tensors):
    """Stream hygiene for buffers swapped out while RCCL may still read
    them (reference ``burst_utils.py:36-39``)."""
    if torch.cuda.is_available():
        for t in tensors:
            if t.is_cuda:
                t.record_stream(torch.cuda.current_stream())
    return tensors


def _dq_wire_dtype(q, dq_ring):
    """Wire dtype for the travelling dq on the flat ring.

    The reference rings dq in the input dtype (fp16,
    ``burst_attn_interface.py:301,393``); we keep the in-round
    accumulation fp32 and cast at the hop, so the payload matches the
    reference at better numerics.  ``BA_DQ_WIRE=fp32`` restores the fp32
    wire.  The double ring merges/zeros the travelling buffer itself
    (``comm.py:187-218`` semantics), so it keeps the fp32 wire."""
    if dq_ring.world_size <= 1 or dq_ring.double_ring:
        return None
    if os.environ.get("BA_DQ_WIRE", "dtype") == "fp32":
        return None
    return q.dtype


def _check_flash_arg(flash):
    if flash not in ("cuda", "triton"):
        raise ValueError(
            "this MI355X-native BurstAttention has a single HIP tile backend "
            "operating in the flash layout [B,S,N,H]; the reference's math-"
            f"path layout (flash={flash!r}) is not supported"
        )


def burst_attn_func(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    softmax_scale: float = None,
    flash: str = "cuda",
    causal: bool = False,
    optimize_bwd_comm: bool = False,
    deterministic: bool = False,
    process_group=None,
    double_group=[None, None],
):
    return OpBurstAttn.apply(
        q, k, v, softmax_scale, flash, causal, optimize_bwd_comm,
        deterministic, process_group, double_group,
    )


def burst_attn_func_striped(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    softmax_scale: float = None,
    flash: str = "cuda",
    causal: bool = False,
    optimize_bwd_comm: bool = False,
    deterministic: bool = False,
    process_group=None,
    double_group=[None, None],
):
    return OpBurstAttnStrip.apply(
        q, k, v, softmax_scale, flash, causal, optimize_bwd_comm,
        deterministic, process_group, double_group,
    )


def _setup_ctx(ctx, q, softmax_scale, flash, causal, optimize_bwd_comm,
               deterministic, process_group, double_group):
    _check_flash_arg(flash)
    if isinstance(double_group[0], tuple):
        # [(group, dq_group), (group2, dq_group2)] — separate backward dq
        # ring groups (reference :188-194)
        ctx.dq_group = (double_group[0][1], double_group[1][1])
        double_group = (double_group[0][0], double_group[1][0])
    else:
        ctx.dq_group = None
    ctx.softmax_scale = (
        1.0 / math.sqrt(q.shape[-1]) if softmax_scale is None else softmax_scale
    )
    ctx.causal = causal
    ctx.optimize_bwd_comm = optimize_bwd_comm
    ctx.deterministic = deterministic
    ctx.process_group = process_group
    ctx.double_group = double_group
    return double_group


def _finalize_fwd(ctx, P, q, ori_k, ori_v, state):
    # o in q.dtype, lse [B,N,S] fp32 (reference :250-252 semantics)
    out, lse = P.fwd_finalize(state, q.dtype)
    ctx.save_for_backward(q, ori_k, ori_v, lse, replicate(out))
    return out


class OpBurstAttn(torch.autograd.Function):
    """Zigzag (default) ring attention; q,k,v: [B, S/W, N, H]."""

    @staticmethod
    def forward(ctx, q, k, v, softmax_scale=None, flash="cuda", causal=False,
                optimize_bwd_comm=False, deterministic=False,
                process_group=None, double_group=[None, None]):
        double_group = _setup_ctx(
            ctx, q, softmax_scale, flash, causal, optimize_bwd_comm,
            deterministic, process_group, double_group,
        )
        P = get_tile_provider()
        scale = ctx.softmax_scale
        ring = Ring(process_group, double_group)
        W, rank = ring.world_size, ring.rank
        ori_k, ori_v = replicate(k), replicate(v)
        comm_bufs = [torch.empty_like(k), torch.empty_like(v)]
        if causal:
            assert q.shape[1] % 2 == 0, (
                "zigzag causal needs an even per-rank seqlen (the rank holds "
                "two half-chunks)"
            )
        half = q.shape[1] // 2
        state = None  # provider-owned carry-in accumulator (in-kernel merge)
        record = []
        for r in range(1, W + 1):
            offset = get_partition_id(double_group, r)
            record.append(offset)
            split_kv = offset <= rank  # kv origin precedes this rank's chunks
            if r != W:
                ring.double_ring_send_recv([k, v], comm_bufs, r)
                ring.commit()
            if r == 1 or not causal:
                state = P.fwd_accum(state, q, k, v, scale, causal)
            elif split_kv:
                # kv chunks [src, 2W-1-src] with src < rank: only chunk
                # [src] (first half) is attended, by all local q (:225-231)
                state = P.fwd_accum(state, q, k[:, :half], v[:, :half], scale, False)
            else:
                # src > rank: only q's second half (chunk [2W-1-rank])
                # attends, to the full received kv (:232-235)
                state = P.fwd_accum(state, q[:, half:], k, v, scale, False,
                                    row_offset=half)
            if r != W:
                kv, comm_bufs = _record_stream(*comm_bufs), [k, v]
                k, v = kv
                ring.wait()
        _logger.info("fwd record of rank %d: %s", get_rank(), record)
        return _finalize_fwd(ctx, P, q, ori_k, ori_v, state)

    @staticmethod
    def backward(ctx, grad_output):
        q, k, v, lse, o = ctx.saved_tensors
        P = get_tile_provider()
        scale = ctx.softmax_scale
        grad_output = grad_output.contiguous()
        group, double_group = ctx.process_group, ctx.double_group
        ring = Ring(group, double_group)
        dq_ring = Ring(group, ctx.dq_group if ctx.dq_group is not None else double_group)
        W, rank = ring.world_size, ring.rank
        if ctx.causal:
            assert q.shape[1] % 2 == 0, (
                "zigzag causal needs an even per-rank seqlen (the rank holds "
                "two half-chunks)"
            )
        half = q.shape[1] // 2

        dq = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        dk = torch.zeros(k.shape, dtype=torch.float32, device=k.device)
        dv = torch.zeros(v.shape, dtype=torch.float32, device=v.device)
        # rank-local staging for rounds whose travelling dq is in flight:
        # kernels accumulate here during the hop, ONE add folds it in
        # after the wait (at W=1 the kernels hit dq directly)
        dq_local = torch.zeros_like(dq) if W > 1 else dq

        if ctx.optimize_bwd_comm:
            # ring the tiny fp32 delta instead of o (reference :269-278)
            dlt = P.bwd_preprocess(o, grad_output)  # [B,N,S] fp32
            delta_buf = None
        else:
            dlt = o  # o travels; delta recomputed per round
            delta_buf = torch.empty(
                q.shape[0], q.shape[2], q.shape[1], dtype=torch.float32,
                device=q.device,
            )
        wire_dtype = _dq_wire_dtype(q, dq_ring)
        if wire_dtype is not None:
            wire_s = torch.empty(q.shape, dtype=wire_dtype, device=q.device)
            wire_r = torch.empty_like(wire_s)

        read_bufs = [torch.empty_like(t) for t in (dlt, grad_output, q, lse)]
        # the fp32 swap buffer is only needed on the fp32-wire path
        dq_buf = [torch.empty_like(dq)] if wire_dtype is None else []
        for r in range(1, W + 1):
            offset = get_partition_id(double_group, r)
            split_q = offset <= rank  # q origin precedes this rank
            if r != W:
                ring.double_ring_send_recv([dlt, grad_output, q, lse], read_bufs, r)
                ring.commit()
            if r != 1:
                if wire_dtype is not None:
                    wire_s.copy_(dq)
                    dq_ring.send_recv([wire_s], [wire_r])
                else:
                    dq_ring.double_ring_send_recv_q([dq], dq_buf, r)
                dq_ring.commit()
            delta = (
                dlt if ctx.optimize_bwd_comm
                else P.bwd_preprocess(dlt, grad_output, out=delta_buf)
            )
            tgt = dq if r == 1 else dq_local
            if r == 1 or not ctx.causal:
                P.bwd_accum(
                    grad_output, q, k, v, delta, lse, scale, ctx.causal,
                    ctx.deterministic, tgt, dk, dv,
                )
                acc = "full"
            elif split_q:
                # travelling q's second half attends my full kv (:322-345)
                P.bwd_accum(
                    grad_output[:, half:], q[:, half:], k, v,
                    delta[:, :, half:], lse[:, :, half:], scale, False,
                    ctx.deterministic, tgt[:, half:], dk, dv,
                )
                acc = "q_half"
            else:
                # travelling q (all of it) attends only my kv first half
                # (:347-367)
                P.bwd_accum(
                    grad_output, q, k[:, :half], v[:, :half], delta, lse,
                    scale, False, ctx.deterministic,
                    tgt, dk[:, :half], dv[:, :half],
                )
                acc = "kv_half"
            if r != W:
                recv, read_bufs = (
                    _record_stream(*read_bufs),
                    [dlt, grad_output, q, lse],
                )
                dlt, grad_output, q, lse = recv
            ring.wait()
            if r != 1:
                dq_ring.wait()
                if wire_dtype is not None:
                    dq.copy_(wire_r)
                else:
                    recv, dq_buf = _record_stream(*dq_buf), [dq]
                    dq = recv[0]
                # fold this round's local contribution into the received dq
                if acc == "q_half":
                    dq[:, half:] += dq_local[:, half:]
                    dq_local[:, half:].zero_()
                else:
                    dq += dq_local
                    dq_local.zero_()
        # one extra hop returns the travelling dq to its owner (:393-396)
        if wire_dtype is not None:
            wire_s.copy_(dq)
            dq_ring.send_recv([wire_s], [wire_r])
            dq_ring.commit()
            dq_ring.wait()
            dq.copy_(wire_r)
        else:
            dq_ring.double_ring_send_recv_q([dq], dq_buf, W + 1)
            dq_ring.commit()
            dq_ring.wait()
            dq = _record_stream(*dq_buf)[0]
        return (
            dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype),
            None, None, None, None, None, None, None,
        )


class OpBurstAttnStrip(torch.autograd.Function):
    """Striped ring attention (token t → rank t mod W); q,k,v [B,S/W,N,H]."""

    @staticmethod
    def forward(ctx, q, k, v, softmax_scale=None, flash="cuda", causal=False,
                optimize_bwd_comm=False, deterministic=False,
                process_group=None, double_group=[None, None]):
        double_group = _setup_ctx(
            ctx, q, softmax_scale, flash, causal, optimize_bwd_comm,
            deterministic, process_group, double_group,
        )
        P = get_tile_provider()
        scale = ctx.softmax_scale
        ring = Ring(process_group, double_group)
        W, rank = ring.world_size, ring.rank
        ori_k, ori_v = replicate(k), replicate(v)
        comm_bufs = [torch.empty_like(k), torch.empty_like(v)]
        state = None
        for r in range(1, W + 1):
            offset = get_partition_id(double_group, r)
            causal_shift = offset > rank  # kv origin follows this rank
            if r != W:
                ring.double_ring_send_recv([k, v], comm_bufs, r)
                ring.commit()
            if not causal_shift or not causal:
                state = P.fwd_accum(state, q, k, v, scale, causal)
            else:
                # one-token shift: striped causal vs a later rank's kv
                # (:463-475)
                state = P.fwd_accum(state, q[:, 1:], k[:, :-1], v[:, :-1],
                                    scale, causal, row_offset=1)
            if r != W:
                kv, comm_bufs = _record_stream(*comm_bufs), [k, v]
                k, v = kv
                ring.wait()
        return _finalize_fwd(ctx, P, q, ori_k, ori_v, state)

    @staticmethod
    def backward(ctx, grad_output):
        q, k, v, lse, o = ctx.saved_tensors
        P = get_tile_provider()
        scale = ctx.softmax_scale
        grad_output = grad_output.contiguous()
        group, double_group = ctx.process_group, ctx.double_group
        ring = Ring(group, double_group)
        dq_ring = Ring(group, ctx.dq_group if ctx.dq_group is not None else double_group)
        W, rank = ring.world_size, ring.rank

        dq = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        dk = torch.zeros(k.shape, dtype=torch.float32, device=k.device)
        dv = torch.zeros(v.shape, dtype=torch.float32, device=v.device)
        dq_local = torch.zeros_like(dq) if W > 1 else dq

        if ctx.optimize_bwd_comm:
            dlt = P.bwd_preprocess(o, grad_output)
            delta_buf = None
        else:
            dlt = o
            delta_buf = torch.empty(
                q.shape[0], q.shape[2], q.shape[1], dtype=torch.float32,
                device=q.device,
            )
        wire_dtype = _dq_wire_dtype(q, dq_ring)
        if wire_dtype is not None:
            wire_s = torch.empty(q.shape, dtype=wire_dtype, device=q.device)
            wire_r = torch.empty_like(wire_s)

        read_bufs = [torch.empty_like(t) for t in (dlt, grad_output, q, lse)]
        # the fp32 swap buffer is only needed on the fp32-wire path
        dq_buf = [torch.empty_like(dq)] if wire_dtype is None else []
        for r in range(1, W + 1):
            offset = get_partition_id(double_group, r)
            causal_shift = offset <= rank and r != 1  # q origin precedes
            if r != W:
                ring.double_ring_send_recv([dlt, grad_output, q, lse], read_bufs, r)
                ring.commit()
            if r != 1:
                if wire_dtype is not None:
                    wire_s.copy_(dq)
                    dq_ring.send_recv([wire_s], [wire_r])
                else:
                    dq_ring.double_ring_send_recv_q([dq], dq_buf, r)
                dq_ring.commit()
            delta = (
                dlt if ctx.optimize_bwd_comm
                else P.bwd_preprocess(dlt, grad_output, out=delta_buf)
            )
            tgt = dq if r == 1 else dq_local
            if not causal_shift or not ctx.causal:
                P.bwd_accum(
                    grad_output, q, k, v, delta, lse, scale, ctx.causal,
                    ctx.deterministic, tgt, dk, dv,
                )
                shifted = False
            else:
                # shifted tile: q[1:] vs k[:-1] (reference :557-585)
                P.bwd_accum(
                    grad_output[:, 1:], q[:, 1:], k[:, :-1], v[:, :-1],
                    delta[:, :, 1:], lse[:, :, 1:], scale, ctx.causal,
                    ctx.deterministic, tgt[:, 1:], dk[:, :-1], dv[:, :-1],
                )
                shifted = True
            if r != W:
                recv, read_bufs = (
                    _record_stream(*read_bufs),
                    [dlt, grad_output, q, lse],
                )
                dlt, grad_output, q, lse = recv
            ring.wait()
            if r != 1:
                dq_ring.wait()
                if wire_dtype is not None:
                    dq.copy_(wire_r)
                else:
                    recv, dq_buf = _record_stream(*dq_buf), [dq]
                    dq = recv[0]
                if shifted:
                    dq[:, 1:] += dq_local[:, 1:]
                    dq_local[:, 1:].zero_()
                else:
                    dq += dq_local
                    dq_local.zero_()
        if wire_dtype is not None:
            wire_s.copy_(dq)
            dq_ring.send_recv([wire_s], [wire_r])
            dq_ring.commit()
            dq_ring.wait()
            dq.copy_(wire_r)
        else:
            dq_ring.double_ring_send_recv_q([dq], dq_buf, W + 1)
            dq_ring.commit()
            dq_ring.wait()
            dq = _record_stream(*dq_buf)[0]
        return (
            dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype),
            None, None, None, None, None, None, None,
        )
