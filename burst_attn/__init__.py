"""Drop-in alias package: existing reference call sites do
``from burst_attn import burst_attn_func`` (reference
burst_attn/__init__.py:1).  This shim re-exports the MI355X-native
implementation under that import path."""

from burst_attn_amd import (  # noqa: F401
    burst_attn_func,
    burst_attn_func_striped,
    OpBurstAttn,
    OpBurstAttnStrip,
)
from burst_attn_amd import comm  # noqa: F401
