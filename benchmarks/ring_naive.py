"""ColossalAI-style non-flash ring attention — the reference's benchmark
COMPARATOR (restatement of /root/reference/benchmarks/ring_attn.py:16-130;
SURVEY.md §2 row 12).  Not part of the package: it exists so the sweep can
print the same context column the reference's README tables carry.

Algorithm (as the reference's RingQK/RingAV pair):
  * RingQK gathers the FULL score slab [B*N, S/W, S] by ringing K around
    (one matmul per incoming block, ring_attn.py:36-42);
  * softmax over the materialised slab;
  * RingAV rings V around, accumulating out += probs[:, :, blk] @ V_blk
    (ring_attn.py:93-105);
  * backward: dK/dV computed full-width from the local operand and
    all-reduced, own block sliced out (ring_attn.py:48-53, 110-115); dQ /
    dprobs accumulate over a second ring of K / V (ring_attn.py:60-68,
    118-129).

Deviations from the reference, by design (documented, not copied):
  * torch.distributed only (the reference drives bmtrain; its `_ring`
    helper is also broken — comm.py:40-45 passes 3 args to the 2-arg
    ring_send_recv, so the comparator cannot run as-is upstream);
  * the reference omits softmax_scale in backward and divides dQ/dK/dV by
    world_size (ring_attn.py:53-54, 69, 114-115) — mathematically wrong;
    this restatement applies the scale and drops the division so the
    comparator can be parity-tested against eager attention.

Memory: the score slab is O(S^2/W) PER RANK — the reason BurstAttention
exists.  Callers must size configs to fit (see slab_bytes()).
"""

import torch
import torch.distributed as dist

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from burst_attn_amd.comm import Ring  # noqa: E402


def _hop(tensor):
    """One ring hop: send to rank+1, receive from rank-1 (batched P2P)."""
    ring = Ring(None, (None, None))
    buf = torch.empty_like(tensor)
    ring.send_recv([tensor], [buf])
    ring.commit()
    ring.wait()
    return buf


def _block_owner(i, rank, world):
    """Whose K/V block arrives after i hops (ring_attn.py:5-9)."""
    return (rank - i - 1) % world


def slab_bytes(b, n, s_local, s_global, dtype=torch.float16):
    return b * n * s_local * s_global * dtype.itemsize


class RingQK(torch.autograd.Function):
    """scores[:, :, blk] = q @ K_blk^T * scale over a K ring
    (ring_attn.py:16-72, corrected per module docstring)."""

    @staticmethod
    def forward(ctx, sub_q, sub_k, scale):
        world = dist.get_world_size()
        rank = dist.get_rank()
        bn, s_local, d = sub_q.shape
        ctx.save_for_backward(sub_q, sub_k)
        ctx.scale = scale
        scores = torch.empty(bn, s_local, s_local * world, dtype=sub_q.dtype,
                             device=sub_q.device)
        blk = lambda r: slice(r * s_local, (r + 1) * s_local)
        scores[:, :, blk(rank)] = torch.matmul(
            sub_q, sub_k.transpose(2, 1)) * scale
        k_cur = sub_k
        for i in range(world - 1):
            k_cur = _hop(k_cur)
            owner = _block_owner(i, rank, world)
            scores[:, :, blk(owner)] = torch.matmul(
                sub_q, k_cur.transpose(2, 1)) * scale
        return scores

    @staticmethod
    def backward(ctx, grad_scores):
        sub_q, sub_k = ctx.saved_tensors
        world = dist.get_world_size()
        rank = dist.get_rank()
        s_local = sub_q.shape[1]
        blk = lambda r: slice(r * s_local, (r + 1) * s_local)
        # dK: full-width from the local q, summed across ranks, own slice
        grad_k = torch.matmul(grad_scores.transpose(2, 1), sub_q) * ctx.scale
        dist.all_reduce(grad_k)
        grad_k = grad_k[:, blk(rank)].contiguous()
        # dQ: accumulate over a second K ring
        grad_q = torch.matmul(grad_scores[:, :, blk(rank)], sub_k) * ctx.scale
        k_cur = sub_k
        for i in range(world - 1):
            k_cur = _hop(k_cur)
            owner = _block_owner(i, rank, world)
            grad_q += torch.matmul(grad_scores[:, :, blk(owner)], k_cur) * ctx.scale
        return grad_q, grad_k, None


class RingAV(torch.autograd.Function):
    """out += probs[:, :, blk] @ V_blk over a V ring
    (ring_attn.py:75-130, corrected per module docstring)."""

    @staticmethod
    def forward(ctx, probs, sub_v):
        world = dist.get_world_size()
        rank = dist.get_rank()
        s_local = sub_v.shape[1]
        ctx.save_for_backward(probs, sub_v)
        blk = lambda r: slice(r * s_local, (r + 1) * s_local)
        out = torch.matmul(probs[:, :, blk(rank)], sub_v)
        v_cur = sub_v
        for i in range(world - 1):
            v_cur = _hop(v_cur)
            owner = _block_owner(i, rank, world)
            out += torch.matmul(probs[:, :, blk(owner)], v_cur)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        probs, sub_v = ctx.saved_tensors
        world = dist.get_world_size()
        rank = dist.get_rank()
        s_local = sub_v.shape[1]
        blk = lambda r: slice(r * s_local, (r + 1) * s_local)
        grad_v = torch.matmul(probs.transpose(2, 1), grad_out)
        dist.all_reduce(grad_v)
        grad_v = grad_v[:, blk(rank)].contiguous()
        grad_probs = torch.empty_like(probs)
        grad_probs[:, :, blk(rank)] = torch.matmul(
            grad_out, sub_v.transpose(2, 1))
        v_cur = sub_v
        for i in range(world - 1):
            v_cur = _hop(v_cur)
            owner = _block_owner(i, rank, world)
            grad_probs[:, :, blk(owner)] = torch.matmul(
                grad_out, v_cur.transpose(2, 1))
        return grad_probs, grad_v


def ring_naive_attention(sub_q, sub_k, sub_v, scale):
    """Full comparator pass: RingQK -> softmax -> RingAV.

    Layout [B*N, S/W, D] (the comparator's own layout, not the flash
    layout — reference benchmark code reshapes the same way)."""
    scores = RingQK.apply(sub_q, sub_k, scale)
    probs = torch.softmax(scores, dim=-1)
    return RingAV.apply(probs, sub_v)
