"""CPU restatement of the reference's attention tile math (fp32, blockwise).

TEST INFRASTRUCTURE ONLY (see oracle/__init__.py header).

Every function documents the reference code it restates
(MayDomine/Burst-Attention @ 2024-10-08, paths relative to the reference
repo root).  All tensors here use the *flash* layout ``[B, S, N, D]``
(batch, sequence, heads, head_dim) that the public ``burst_attn_func``
contract uses (reference ``burst_attn/burst_attn_interface.py:162-168``);
``lse`` is ``[B, N, S]`` fp32, matching what the flash-attn CUDA extension
returns to the reference (``burst_attn/burst_utils.py:150-163``).
"""

import math

import torch

__all__ = [
    "eager_attention",
    "tile_fwd",
    "tile_bwd",
    "scale_out_lse",
    "merge_tile_output",
    "ring_forward_reference",
    "ring_forward_backward_reference",
]


def _as_f32_bnsd(t):
    # [B,S,N,D] -> [B,N,S,D] fp32
    return t.to(torch.float32).permute(0, 2, 1, 3)


def eager_attention(q, k, v, softmax_scale=None, causal=False):
    """Full-sequence eager attention — the ground truth.

    Restates the reference's eager comparator ``benchmarks/utils.py:49-54``
    (softmax(q k^T * scale) v) plus the plain causal mask that flash-attn
    applies on the reference's default path (``test/test_burst.py:175``).

    q, k, v: [B, S, N, D] (any float dtype).  Returns o [B, S, N, D] fp32.
    """
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    qf, kf, vf = _as_f32_bnsd(q), _as_f32_bnsd(k), _as_f32_bnsd(v)
    scores = torch.matmul(qf, kf.transpose(-2, -1)) * softmax_scale
    if causal:
        sq, sk = scores.shape[-2], scores.shape[-1]
        # equal-length tiles only in this codebase; mask k_pos > q_pos
        mask = torch.ones(sq, sk, dtype=torch.bool).triu(1 + (sk - sq))
        scores = scores.masked_fill(mask, float("-inf"))
    p = torch.softmax(scores, dim=-1)
    o = torch.matmul(p, vf)
    return o.permute(0, 2, 1, 3).contiguous()


def tile_fwd(q, k, v, softmax_scale=None, causal=False, q_block=512, k_block=512):
    """One local flash tile: o = softmax(q k^T * scale) v, plus its LSE.

    Blockwise online-softmax restatement of the reference's math path
    ``burst_attn/burst_utils.py:42-74`` (``inter_normal_attn``), with the
    flash-attn ``"cuda"``-path semantics the product replicates:
      * lse is the exact log-sum-exp of the *scaled* scores, natural log,
        fp32, layout [B, N, S] (``burst_utils.py:150-163``) — without the
        ``+1e-5`` regulariser the math path adds at ``burst_utils.py:71-73``
        (that epsilon exists only on the math path);
      * o is returned normalised, in fp32 (caller casts).

    Returns (o [B,S,N,D] fp32, lse [B,N,S] fp32).
    """
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    qf, kf, vf = _as_f32_bnsd(q), _as_f32_bnsd(k), _as_f32_bnsd(v)
    B, N, Sq, D = qf.shape
    Sk = kf.shape[2]
    if causal:
        assert Sq == Sk, "causal tiles are equal-length in this codebase"
    o = torch.zeros(B, N, Sq, D, dtype=torch.float32)
    lse = torch.empty(B, N, Sq, dtype=torch.float32)
    for q0 in range(0, Sq, q_block):
        q1 = min(q0 + q_block, Sq)
        qb = qf[:, :, q0:q1]
        m = torch.full((B, N, q1 - q0, 1), float("-inf"))
        l = torch.zeros(B, N, q1 - q0, 1)
        acc = torch.zeros(B, N, q1 - q0, D)
        k_hi = Sk if not causal else q1  # causal: keys beyond q1-1 are masked
        for k0 in range(0, k_hi, k_block):
            k1 = min(k0 + k_block, k_hi)
            s = torch.matmul(qb, kf[:, :, k0:k1].transpose(-2, -1)) * softmax_scale
            if causal and k1 > q0:
                qi = torch.arange(q0, q1).unsqueeze(1)
                kj = torch.arange(k0, k1).unsqueeze(0)
                s = s.masked_fill(kj > qi, float("-inf"))
            m_new = torch.maximum(m, s.amax(dim=-1, keepdim=True))
            # rows still at -inf (no keys seen yet) keep alpha = 1
            alpha = torch.where(
                torch.isinf(m) & (m < 0), torch.ones_like(m), torch.exp(m - m_new)
            )
            # rows fully masked in this block: exp(s - (-inf)) would be NaN;
            # subtract 0 there instead, giving p = exp(-inf) = 0
            m_sub = torch.where(
                torch.isinf(m_new) & (m_new < 0), torch.zeros_like(m_new), m_new
            )
            p = torch.exp(s - m_sub)
            l = l * alpha + p.sum(dim=-1, keepdim=True)
            acc = acc * alpha + torch.matmul(p, vf[:, :, k0:k1])
            m = m_new
        o[:, :, q0:q1] = acc / l
        lse[:, :, q0:q1] = (torch.log(l) + m).squeeze(-1)
    return o.permute(0, 2, 1, 3).contiguous(), lse


def tile_bwd(
    do,
    q,
    k,
    v,
    lse,
    softmax_scale=None,
    causal=False,
    softmax_d=None,
    o=None,
    q_block=512,
    k_block=512,
):
    """Backward of one local flash tile.

    Restates the reference's math-path backward
    ``burst_attn/burst_utils.py:77-101`` (``inter_normal_attn_backward``):
        p   = exp(q k^T * scale - lse)
        dv += p^T do
        dp  = do v^T
        ds  = p * (dp - delta) * scale
        dq  = ds k;  dk += ds^T q
    with delta either supplied externally (``softmax_d``; the
    ``optimize_bwd_comm`` path, ``burst_utils.py:195-229``) or computed as
    rowsum(o * do) fp32 (``burst_attn_interface.py:272-278``).

    do,q,k,v,(o): [B,S,N,D]; lse/(softmax_d): [B,N,S] fp32.
    Returns (dq, dk, dv) fp32 in [B,S,N,D].
    """
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    dof, qf, kf, vf = (
        _as_f32_bnsd(do),
        _as_f32_bnsd(q),
        _as_f32_bnsd(k),
        _as_f32_bnsd(v),
    )
    B, N, Sq, D = qf.shape
    Sk = kf.shape[2]
    if causal:
        assert Sq == Sk
    if softmax_d is None:
        assert o is not None, "need o to compute delta = rowsum(o*do)"
        delta = (_as_f32_bnsd(o) * dof).sum(-1)  # [B,N,Sq]
    else:
        delta = softmax_d.to(torch.float32)
    dq = torch.zeros_like(qf)
    dk = torch.zeros_like(kf)
    dv = torch.zeros_like(vf)
    for q0 in range(0, Sq, q_block):
        q1 = min(q0 + q_block, Sq)
        qb, dob = qf[:, :, q0:q1], dof[:, :, q0:q1]
        lseb = lse[:, :, q0:q1].to(torch.float32).unsqueeze(-1)
        deltab = delta[:, :, q0:q1].unsqueeze(-1)
        k_hi = Sk if not causal else q1
        for k0 in range(0, k_hi, k_block):
            k1 = min(k0 + k_block, k_hi)
            kb, vb = kf[:, :, k0:k1], vf[:, :, k0:k1]
            s = torch.matmul(qb, kb.transpose(-2, -1)) * softmax_scale
            if causal and k1 > q0:
                qi = torch.arange(q0, q1).unsqueeze(1)
                kj = torch.arange(k0, k1).unsqueeze(0)
                s = s.masked_fill(kj > qi, float("-inf"))
            p = torch.exp(s - lseb)  # burst_utils.py:88
            dv[:, :, k0:k1] += torch.matmul(p.transpose(-2, -1), dob)
            dp = torch.matmul(dob, vb.transpose(-2, -1))
            ds = p * (dp - deltab) * softmax_scale
            dq[:, :, q0:q1] += torch.matmul(ds, kb)
            dk[:, :, k0:k1] += torch.matmul(ds.transpose(-2, -1), qb)
    perm = lambda t: t.permute(0, 2, 1, 3).contiguous()
    return perm(dq), perm(dk), perm(dv)


def scale_out_lse(o, lse, o_i, lse_i):
    """LSE merge of a new partial tile into the accumulator.

    Restates ``burst_attn/burst_utils.py:20-33``
    (``cuda_scale_out_lse_helper``):
        new_lse = lse + log(1 + exp(lse_i - lse))
        o       = exp(lse - new_lse) * o + exp(lse_i - new_lse) * o_i

    o:   [B,S,N,D] fp32 accumulator; lse: [B,S,N,1] fp32 accumulator
    o_i: [B,S,N,D] new tile output;  lse_i: [B,N,S] fp32 new tile lse.
    Returns (o, lse) merged.
    """
    o_i = o_i.to(torch.float32)
    lse_i = lse_i.transpose(-2, -1).unsqueeze(-1).contiguous()
    new_lse = lse + torch.log1p(torch.exp(lse_i - lse))
    o = torch.exp(lse - new_lse) * o + torch.exp(lse_i - new_lse) * o_i
    return o, new_lse


def merge_tile_output(o, lse, o_i, lse_i, q_len, k_len):
    """Merge dispatch per ``burst_attn/burst_utils.py:161-176``.

    * first round (o is None): adopt o_i (fp32) and lse_i as [B,S,N,1];
    * zigzag half round (q_len == k_len // 2): merge into o[:, half:];
    * striped shift round (lse rows == lse_i rows + 1): merge into o[:, 1:];
    * otherwise merge the whole tile.
    Returns (o [B,S,N,D] fp32, lse [B,S,N,1] fp32).
    """
    if o is None:
        return o_i.to(torch.float32), lse_i.transpose(-2, -1).unsqueeze(-1).contiguous()
    if q_len == k_len // 2:
        half = o.shape[1] // 2
        o[:, half:], lse[:, half:] = scale_out_lse(o[:, half:], lse[:, half:], o_i, lse_i)
    elif lse.shape[1] == lse_i.shape[2] + 1:
        o[:, 1:], lse[:, 1:] = scale_out_lse(o[:, 1:], lse[:, 1:], o_i, lse_i)
    else:
        o, lse = scale_out_lse(o, lse, o_i, lse_i)
    return o, lse


def ring_forward_reference(q_full, k_full, v_full, softmax_scale=None, causal=False):
    """Expected full-sequence forward output for ring tests (eager)."""
    return eager_attention(q_full, k_full, v_full, softmax_scale, causal)


def ring_forward_backward_reference(
    q_full, k_full, v_full, do_full, softmax_scale=None, causal=False
):
    """Expected (o, dq, dk, dv) for the full sequence, via autograd on the
    eager reference — what the distributed ring must reproduce chunk-wise
    (reference test oracle structure: ``test/test_burst.py:175,184,215-218``).
    """
    q = q_full.detach().to(torch.float32).requires_grad_()
    k = k_full.detach().to(torch.float32).requires_grad_()
    v = v_full.detach().to(torch.float32).requires_grad_()
    o = eager_attention(q, k, v, softmax_scale, causal)
    dq, dk, dv = torch.autograd.grad(o, (q, k, v), do_full.to(torch.float32))
    return o.detach(), dq, dk, dv
