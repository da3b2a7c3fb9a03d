"""Single-process ring simulator — TEST INFRASTRUCTURE ONLY.

Replays the reference's ring round structure
(burst_attn_interface.py:214-242 fwd, :291-396 bwd) over W virtual ranks
in one process, with a pluggable tile provider (HIP kernels on GPU, the
oracle on CPU).  Used to exercise the multi-round zigzag/striped
bookkeeping against full-sequence eager attention on ONE GPU — the real
multi-process path is covered by tests/test_ring_cpu.py (gloo) and the
driver's 8-GPU round-end runs.

Independent of burst_attn_amd.interface by design: a second
implementation of the round bookkeeping that must agree with both the
eager oracle and (via the gloo tests) the product orchestration.
"""

import torch

from oracle.partition import get_chunk, unchunk


def _merge_full(P, o, lse, o_i, lse_i):
    if o is None:
        return o_i.to(torch.float32), lse_i.transpose(-2, -1).unsqueeze(-1).contiguous()
    return P.merge(o, lse, o_i, lse_i)


def simulate_ring(P, q_full, k_full, v_full, do_full, W, scale, causal,
                  striped=False, optimize_bwd_comm=False):
    """Returns (o_full, dq_full, dk_full, dv_full) reassembled."""
    zig = causal and not striped
    dim = 1
    ch = lambda t, r: get_chunk(t, dim, r, W, zigzag=zig, striped=striped)
    qs = [ch(q_full, r) for r in range(W)]
    ks = [ch(k_full, r) for r in range(W)]
    vs = [ch(v_full, r) for r in range(W)]
    dos = [ch(do_full, r) for r in range(W)]
    s_local = qs[0].shape[1]
    half = s_local // 2

    # ---- forward ----
    os_, lses = [], []
    for rank in range(W):
        o = lse = None
        for r in range(1, W + 1):
            src = (rank - (r - 1)) % W
            k, v = ks[src], vs[src]
            if striped:
                causal_shift = (r - 1) > rank
                if not causal_shift or not causal:
                    o_i, lse_i = P.fwd(qs[rank], k, v, scale, causal)
                    o, lse = _merge_full(P, o, lse, o_i, lse_i)
                else:
                    o_i, lse_i = P.fwd(qs[rank][:, 1:], k[:, :-1], v[:, :-1], scale, causal)
                    o[:, 1:], lse[:, 1:] = P.merge(o[:, 1:], lse[:, 1:], o_i, lse_i)
            else:
                split_kv = (r - 1) <= rank
                if r == 1 or not causal:
                    o_i, lse_i = P.fwd(qs[rank], k, v, scale, causal)
                    o, lse = _merge_full(P, o, lse, o_i, lse_i)
                elif split_kv:
                    o_i, lse_i = P.fwd(qs[rank], k[:, :half], v[:, :half], scale, False)
                    o, lse = P.merge(o, lse, o_i, lse_i)
                else:
                    o_i, lse_i = P.fwd(qs[rank][:, half:], k, v, scale, False)
                    o[:, half:], lse[:, half:] = P.merge(o[:, half:], lse[:, half:], o_i, lse_i)
        os_.append(o)
        lses.append(lse.squeeze(-1).transpose(1, 2).contiguous())

    # ---- backward (kv-resident view: rank holds kv, q travels) ----
    dqs = [torch.zeros_like(qs[r], dtype=torch.float32) for r in range(W)]
    dks = [torch.zeros_like(ks[r], dtype=torch.float32) for r in range(W)]
    dvs = [torch.zeros_like(vs[r], dtype=torch.float32) for r in range(W)]
    for rank in range(W):
        k, v = ks[rank], vs[rank]
        for r in range(1, W + 1):
            src = (rank - (r - 1)) % W  # owner of the q-set at round r
            q, do, lse = qs[src], dos[src], lses[src]
            delta = P.bwd_preprocess(os_[src].to(q.dtype), do)
            if striped:
                # striped bwd: q from an EARLIER rank is shifted
                causal_shift = ((r - 1) <= rank) and r != 1
                if not causal_shift or not causal:
                    dq_i, dk_i, dv_i = P.bwd(do, q, k, v, delta, lse, scale, causal, True)
                    dqs[src] += dq_i
                    dks[rank] += dk_i
                    dvs[rank] += dv_i
                else:
                    dq_i, dk_i, dv_i = P.bwd(
                        do[:, 1:], q[:, 1:], k[:, :-1], v[:, :-1],
                        delta[:, :, 1:], lse[:, :, 1:], scale, causal, True)
                    dqs[src][:, 1:] += dq_i
                    dks[rank][:, :-1] += dk_i
                    dvs[rank][:, :-1] += dv_i
            else:
                split_q = (r - 1) <= rank
                if r == 1 or not causal:
                    dq_i, dk_i, dv_i = P.bwd(do, q, k, v, delta, lse, scale, causal, True)
                    dqs[src] += dq_i
                    dks[rank] += dk_i
                    dvs[rank] += dv_i
                elif split_q:
                    dq_i, dk_i, dv_i = P.bwd(
                        do[:, half:], q[:, half:], k, v,
                        delta[:, :, half:], lse[:, :, half:], scale, False, True)
                    dqs[src][:, half:] += dq_i
                    dks[rank] += dk_i
                    dvs[rank] += dv_i
                else:
                    dq_i, dk_i, dv_i = P.bwd(
                        do, q, k[:, :half], v[:, :half], delta, lse, scale, False, True)
                    dqs[src] += dq_i
                    dks[rank][:, :half] += dk_i
                    dvs[rank][:, :half] += dv_i

    un = lambda lst: unchunk([t for t in lst], dim, zigzag=zig, striped=striped)
    return un(os_), un(dqs), un(dks), un(dvs)
