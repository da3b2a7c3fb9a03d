"""Sequence partitioning used by the ring (test-infrastructure restatement).

Restates the reference's data partitioning ``test/test_burst.py:44-58``
(``get_chunk``): how a full sequence is split across the W ring ranks.

* plain:   rank r gets contiguous chunk r of W.
* zigzag (``half_reputation``): split into 2W chunks; rank r gets
  cat(chunk[r], chunk[2W-1-r]) — the causal load-balancing layout of
  ``OpBurstAttn`` (``burst_attn_interface.py:221-235``).
* striped: token t goes to rank t mod W, keeping local order — the
  ``OpBurstAttnStrip`` layout.
"""

import torch

__all__ = ["get_chunk", "unchunk"]


def get_chunk(t, dim, rank, world_size, zigzag=False, striped=False):
    """Per-rank chunk of ``t`` along ``dim`` (cf. test/test_burst.py:44-58)."""
    if striped:
        s = t.shape[dim]
        assert s % world_size == 0
        idx = torch.arange(rank, s, world_size, device=t.device)
        return t.index_select(dim, idx).contiguous()
    if zigzag:
        splits = t.chunk(world_size * 2, dim=dim)
        return torch.cat(
            [splits[rank], splits[world_size * 2 - rank - 1]], dim=dim
        ).contiguous()
    return t.chunk(world_size, dim=dim)[rank].contiguous()


def unchunk(chunks, dim, zigzag=False, striped=False):
    """Reassemble the full sequence from per-rank chunks (inverse of
    ``get_chunk``); used by tests to compare against the full-sequence
    oracle."""
    world_size = len(chunks)
    if striped:
        s_local = chunks[0].shape[dim]
        parts = []
        for i in range(s_local):
            for r in range(world_size):
                parts.append(chunks[r].narrow(dim, i, 1))
        return torch.cat(parts, dim=dim)
    if zigzag:
        first, second = [], []
        for r in range(world_size):
            c = chunks[r]
            half = c.shape[dim] // 2
            first.append(c.narrow(dim, 0, half))
            second.append(c.narrow(dim, half, half))
        return torch.cat(first + second[::-1], dim=dim)
    return torch.cat(list(chunks), dim=dim)
