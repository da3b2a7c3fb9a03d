"""MI355X-native BurstAttention.

A from-scratch gfx950 (CDNA4) implementation of the reference
MayDomine/Burst-Attention hot path: ring-distributed exact attention with
hand-written HIP flash-tile kernels and RCCL P2P over xGMI.

Public API (drop-in for the reference ``burst_attn`` package):
    burst_attn_func, burst_attn_func_striped
"""

from .interface import (  # noqa: F401
    burst_attn_func,
    burst_attn_func_striped,
    OpBurstAttn,
    OpBurstAttnStrip,
)

__version__ = "0.1.0"
