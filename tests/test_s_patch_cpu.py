"""CPU guard for the assembly patch flow (tools/s_patch.py).

The round-3 on-ramp hand-edits the compiler's .s between `hipcc -S` and
the assemble step; its anchors must fail LOUDLY if the compiler output
drifts.  This test regenerates the .s (or uses the build tree's copy),
applies the qk_split transform, and checks the structural invariants —
no GPU needed (clang assembles for amdgcn without one).
"""

import json
import os
import re
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BUILD = os.path.join(REPO, "burst_attn_amd", "csrc", "_build")
S_PATH = os.path.join(BUILD, "attn_fwd_scaffold.s")
SYMS = os.path.join(REPO, "burst_attn_amd", "_asm_fwd_syms.json")

sys.path.insert(0, os.path.join(REPO, "tools"))
import s_patch  # noqa: E402


@pytest.fixture(scope="module")
def scaffold():
    if not (os.path.exists(S_PATH) and os.path.exists(SYMS)):
        from burst_attn_amd import build_ext

        build_ext._build_asm_hsaco(verbose=False, force=True)
    return open(S_PATH).read(), json.load(open(SYMS))


def test_symbols_extracted(scaffold):
    _, syms = scaffold
    assert set(syms) == {"f16_accum", "bf16_accum"}
    for s in syms.values():
        assert s.startswith("_ZN12_GLOBAL__N_115attn_fwd_kernel")


def test_qk_split_structure(scaffold):
    text, syms = scaffold
    out = s_patch.qk_split(text, list(syms.values()))
    for sym in syms.values():
        # the spare range derives from the kernel's own allocation so it can
        # never collide with live registers as the kernel evolves
        lo, hi, new_vgpr = s_patch._spare_range(text, sym)
        body = out[out.index(sym + ":"):]
        body = body[:body.index(".Lfunc_end")]
        # odd chain members renamed: 4 MFMAs per 8-deep chain target the
        # spare range; chain count follows the hand-unrolled tile copies
        spare = len(re.findall(
            r"v_mfma_f32_32x32x16_\w+ v\[%d:%d\]" % (lo, hi), body))
        assert spare >= 8 and spare % 4 == 0, \
            f"{sym}: expected 4 spare-accumulator MFMAs per chain, got {spare}"
        n_chains = spare // 4
        # merge epilogues: 8 packed adds per chain, sourcing the spare
        # range (the kernel's own rowsum also uses v_pk_add_f32 — those
        # source softmax registers, not the spare accumulator)
        srcs = [int(s) for s in re.findall(
            r"v_pk_add_f32 v\[\d+:\d+\], v\[\d+:\d+\], v\[(\d+):\d+\]", body)]
        adds_all = sum(1 for s in srcs if lo <= s <= hi)
        assert adds_all == 8 * n_chains, \
            f"{sym}: expected {8 * n_chains} merge adds, got {adds_all}"
        # register allocation raised past the spare range, 4-aligned
        desc = out[out.index(".amdhsa_kernel " + sym):]
        desc = desc[:desc.index(".end_amdhsa_kernel")]
        assert f".amdhsa_next_free_vgpr {new_vgpr}" in desc
        assert f".amdhsa_accum_offset {new_vgpr}" in desc
        assert new_vgpr % 4 == 0 and new_vgpr <= 256


def test_qk_split_asserts_on_drift(scaffold):
    text, syms = scaffold
    # a scaffold without the expected chains must be rejected, not patched
    broken = text.replace("v_mfma_f32_32x32x16_f16 v[66:81]",
                          "v_mfma_f32_32x32x16_f16 v[40:55]")
    with pytest.raises(AssertionError):
        s_patch.qk_split(broken, list(syms.values()))


def test_patched_s_assembles(scaffold, tmp_path):
    text, syms = scaffold
    out = s_patch.qk_split(text, list(syms.values()))
    sp = tmp_path / "patched.s"
    sp.write_text(out)
    clang = "/opt/rocm/lib/llvm/bin/clang"
    if not os.path.exists(clang):
        pytest.skip("rocm llvm not present")
    r = subprocess.run(
        [clang, "-x", "assembler", "-target", "amdgcn-amd-amdhsa",
         "-mcpu=gfx950", "-c", str(sp), "-o", str(tmp_path / "patched.o")],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-2000:]


def test_fwd_kernels_no_outlining_no_spills(scaffold):
    """Regression net for the inliner: every fwd kernel instantiation must
    be call-free (an outlined helper lambda turns into an s_swappc whose
    ABI spills hundreds of scratch bytes — this crushed the SUBT=2/3
    variants once) and spill-free (the production and scaffold configs
    are tuned to their register budgets)."""
    text, _ = scaffold
    checked = 0
    for m in re.finditer(r"\.amdhsa_kernel\s+(\S+)", text):
        sym = m.group(1)
        if "attn_fwd_kernel" not in sym:
            continue
        desc = text[m.start():text.index(".end_amdhsa_kernel", m.start())]
        spill = int(re.search(
            r"\.amdhsa_private_segment_fixed_size\s+(\d+)", desc).group(1))
        body_start = re.search("^" + re.escape(sym) + r":.*$", text, re.M)
        body = text[body_start.start():text.index(".Lfunc_end",
                                                  body_start.start())]
        assert "s_swappc" not in body, f"{sym}: outlined call in kernel body"
        # spill-free is required for the maintained configs (SUBT=1
        # production, SUBT=2/3 pipelined scaffolds); the legacy SUBT=0
        # joint-softmax variant carries a pre-existing few-byte spill
        subt = re.search(r"Li128ELi64ELi\dELi\dELi(\d)E", sym)
        if subt and subt.group(1) != "0":
            assert spill == 0, f"{sym}: {spill} bytes of scratch spill"
        checked += 1
    assert checked >= 20, f"only {checked} fwd kernels found in the scaffold"
