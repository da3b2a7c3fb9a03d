"""Generate golden fixtures pinning the oracle to the reference's own math.

Run IN THE BUILD CONTAINER ONLY (where /root/reference exists):

    python -m oracle.gen_golden

The reference package cannot be imported whole (it imports ``bmtrain``,
``flash_attn`` and Triton ``lao`` at module import; none are installed
here — SURVEY.md §8c).  But its *tile math* is pure torch:

  * ``inter_normal_attn``          burst_attn/burst_utils.py:42-74
  * ``inter_normal_attn_backward`` burst_attn/burst_utils.py:77-101
  * ``cuda_scale_out_lse_helper``  burst_attn/burst_utils.py:20-33

This script loads exactly those three function definitions out of the
reference source via ``ast`` (no reference code is copied into the repo;
the functions are executed from the read-only reference checkout), runs
them on seeded inputs, and stores inputs+outputs under ``tests/golden/``.
``tests/test_oracle.py`` then checks the repo's CPU oracle against these
fixtures — on any machine, without the reference present.
"""

import ast
import os
import sys

import numpy as np
import torch

REFERENCE = os.environ.get("BURST_REFERENCE", "/root/reference")
GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "..", "tests", "golden")

_ALLOW = {
    "inter_normal_attn",
    "inter_normal_attn_backward",
    "cuda_scale_out_lse_helper",
}


def load_reference_tile_math():
    """Exec only the three pure-torch tile functions from the reference
    source (decorators stripped — @torch.jit.script is irrelevant on CPU)."""
    src_path = os.path.join(REFERENCE, "burst_attn", "burst_utils.py")
    with open(src_path) as f:
        tree = ast.parse(f.read())
    keep = []
    for node in tree.body:
        if isinstance(node, ast.FunctionDef) and node.name in _ALLOW:
            node.decorator_list = []
            keep.append(node)
    assert len(keep) == len(_ALLOW), f"found only {[n.name for n in keep]}"
    mod = ast.Module(body=keep, type_ignores=[])
    ns = {"torch": torch}
    exec(compile(mod, src_path, "exec"), ns)
    return ns


def gen_tile_fixture(ns, name, b, sq, sk, n, d, seed, rounds=2):
    """Golden vectors for one tile shape.

    Replays the reference math path over ``rounds`` kv chunks with carry-in
    (m_i, lse_i, acc_o) exactly as ``OpBurstAttn.forward`` does on the math
    path (burst_attn_interface.py:214-248: attn_forward per round, final
    o = acc_o * exp(m - lse)), then the backward on the concatenated kv.
    Reference layout is [B,N,S,D] (class docstring :162-168 'Normal').
    """
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(b, n, sq, d, generator=g, dtype=torch.float32)
    ks = [torch.randn(b, n, sk, d, generator=g, dtype=torch.float32) for _ in range(rounds)]
    vs = [torch.randn(b, n, sk, d, generator=g, dtype=torch.float32) for _ in range(rounds)]
    do = torch.randn(b, n, sq, d, generator=g, dtype=torch.float32)
    scale = d ** -0.5

    m_i = lse_i = acc_o = None
    for r in range(rounds):
        acc_o, m_i, lse_i = ns["inter_normal_attn"](
            q, ks[r], vs[r], m_i, lse_i, acc_o, scale, None
        )
    o = acc_o * torch.exp(m_i - lse_i)  # burst_attn_interface.py:246-248

    k_cat = torch.cat(ks, dim=2)
    v_cat = torch.cat(vs, dim=2)
    dq = torch.zeros_like(q)
    dk = torch.zeros_like(k_cat)
    dv = torch.zeros_like(v_cat)
    delta = (o * do).sum(-1, keepdim=True)  # burst_attn_interface.py:272-278
    ns["inter_normal_attn_backward"](
        do, q, k_cat, v_cat, delta, lse_i, dq, dk, dv, scale, None
    )

    out = dict(
        q=q, k=k_cat, v=v_cat, do=do, o=o,
        m=m_i.squeeze(-1), lse=lse_i.squeeze(-1),
        dq=dq, dk=dk, dv=dv,
        scale=np.float32(scale), rounds=np.int64(rounds),
    )
    path = os.path.join(GOLDEN_DIR, f"{name}.npz")
    np.savez_compressed(path, **{k: (t.numpy() if torch.is_tensor(t) else t) for k, t in out.items()})
    print(f"wrote {path}: q{tuple(q.shape)} k{tuple(k_cat.shape)}")


def gen_merge_fixture(ns, name, b, s, n, d, seed):
    """Golden vectors for the LSE merge (cuda_scale_out_lse_helper),
    layouts exactly as the reference calls it at burst_utils.py:161-176:
    o [B,S,N,D] fp32, lse [B,S,N,1] fp32, o_i [B,S,N,D], lse_i [B,N,S]."""
    g = torch.Generator().manual_seed(seed)
    o = torch.randn(b, s, n, d, generator=g, dtype=torch.float32)
    lse = torch.randn(b, s, n, 1, generator=g, dtype=torch.float32) * 2
    o_i = torch.randn(b, s, n, d, generator=g, dtype=torch.float32)
    lse_i = torch.randn(b, n, s, generator=g, dtype=torch.float32) * 2
    o_m, lse_m = ns["cuda_scale_out_lse_helper"](o.clone(), lse.clone(), o_i, lse_i)
    path = os.path.join(GOLDEN_DIR, f"{name}.npz")
    np.savez_compressed(
        path,
        o=o.numpy(), lse=lse.numpy(), o_i=o_i.numpy(), lse_i=lse_i.numpy(),
        o_merged=o_m.numpy(), lse_merged=lse_m.numpy(),
    )
    print(f"wrote {path}")


def main():
    os.makedirs(GOLDEN_DIR, exist_ok=True)
    torch.set_num_threads(max(1, os.cpu_count() or 1))
    ns = load_reference_tile_math()
    # d=64 and d=128 tiles; square and half-kv (zigzag) shapes
    gen_tile_fixture(ns, "tile_b1_s256_n4_d64", 1, 256, 256, 4, 64, seed=1234)
    gen_tile_fixture(ns, "tile_b2_s192_n2_d128", 2, 192, 192, 2, 128, seed=4321)
    gen_tile_fixture(ns, "tile_halfkv_b1_s256_n2_d128", 1, 256, 128, 2, 128, seed=7)
    gen_merge_fixture(ns, "merge_b1_s128_n4_d64", 1, 128, 4, 64, seed=99)
    gen_merge_fixture(ns, "merge_b2_s96_n2_d128", 2, 96, 2, 128, seed=100)


if __name__ == "__main__":
    sys.exit(main())
