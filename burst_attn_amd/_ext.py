"""Loader for the in-tree gfx950 HIP extension (burst_attn_amd/_C.so).

The extension is built IN-TREE by ``__graft_entry__.build()`` (hipcc
--offload-arch=gfx950) so the .so travels with the repo snapshot.  There
is deliberately NO fallback: if the extension is missing, the product
path fails loudly (north_star: the HIP kernels ARE the compute path)."""

import importlib
import os

_cached = None


def load_extension():
    global _cached
    if _cached is not None:
        return _cached
    try:
        _cached = importlib.import_module("burst_attn_amd._C")
    except ImportError as e:
        here = os.path.dirname(__file__)
        raise RuntimeError(
            "burst_attn_amd HIP extension (_C.so) is not built. Run "
            "`python -c \"import __graft_entry__ as g; g.build()\"` from the "
            f"repo root (expected {here}/_C*.so). There is no CPU/eager "
            "fallback in the product path."
        ) from e
    return _cached
