"""CPU oracle for the BurstAttention hot path.

TEST INFRASTRUCTURE ONLY.  Only ``tests/``, ``__graft_entry__.smoke()`` and
``bench.py``'s ``cpu_baseline`` leg may import this package, and only as the
checker / reported CPU baseline — never as the thing measured or shipped.
The product path (``burst_attn_amd``) must never route through this code.

This package is a CPU restatement of the reference algorithm
(MayDomine/Burst-Attention @ 2024-10-08):

- tile math:   reference ``burst_attn/burst_utils.py:42-101``
  (``inter_normal_attn`` / ``inter_normal_attn_backward``)
- LSE merge:   reference ``burst_attn/burst_utils.py:20-33``
  (``cuda_scale_out_lse_helper``) and the merge dispatch at
  ``burst_utils.py:149-177``
- ring rounds: reference ``burst_attn/burst_attn_interface.py:214-242``
  (forward) and ``:291-390`` (backward), zigzag/striped partitioning per
  ``test/test_burst.py:44-58``

Pinned against the reference itself: ``oracle/gen_golden.py`` (run in the
build container where ``/root/reference`` exists) executes the reference's
own pure-torch tile functions and stores golden vectors under
``tests/golden/``; ``tests/test_oracle.py`` checks this restatement against
those fixtures.  The GPU box never reads ``/root/reference``.
"""

from .attn import (  # noqa: F401
    eager_attention,
    tile_fwd,
    tile_bwd,
    scale_out_lse,
    ring_forward_reference,
    ring_forward_backward_reference,
)
from .partition import get_chunk, unchunk  # noqa: F401
