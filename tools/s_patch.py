"""Hand-scheduling transforms on the compiler-generated forward assembly.

Applied between `hipcc -S` and the `clang -x assembler` step by
build_ext._build_asm_hsaco (the tools/asm_probe flow).  Each transform is
anchored on hard patterns of the production kernel's steady loop and
raises AssertionError if the compiler output shifts — a failed patch
fails the build loudly rather than assembling something subtly wrong.

Transform `qk_split` — break the serial QK accumulation chain.

The steady loop computes each 32x64 S subtile as EIGHT serially
dependent v_mfma_f32_32x32x16 ops into v[66:81] (the D=128 reduction,
one accumulate per 16-wide K chunk).  A dependent same-accumulator MFMA
cannot start until the previous one completes, so the MFMA pipe idles
roughly half of every QK phase (measured: the mfma_pipe worksheet's
serial-chain row).  The transform renames the ODD chain members to the
spare registers v[224:239] (the kernel uses 222 of the 256 two-wave
arch VGPRs — the rename keeps 2 waves/SIMD) which splits each chain
into two interleaved 4-deep chains that issue back-to-back, then adds
the two partial accumulators with 8 packed fp32 adds before the first
consumer.  No instruction moves; lgkmcnt/vmcnt bookkeeping is
untouched.  Changes the in-D summation order (a+b+c+d -> (a+c)+(b+d)),
which is within the parity tolerances and still run-to-run
deterministic.
"""

import re

ACC = (66, 81)          # compiler's QK accumulator range


def _spare_range(text, sym):
    """First even register at/above the kernel's next_free_vgpr — the odd
    chain's accumulator lives there so it can never collide with live
    registers regardless of how the allocation shifts between builds."""
    m = re.search(r"\.amdhsa_kernel\s+" + re.escape(sym) +
                  r".*?\.amdhsa_next_free_vgpr\s+(\d+)", text, re.S)
    assert m, f"next_free_vgpr not found for {sym}"
    nv = int(m.group(1))
    lo = (nv + 3) & ~3  # 4-aligned so the patched accum_offset stays legal
    assert lo + 16 <= 256, f"{sym}: no 16-register spare below the 2-wave ceiling (nv={nv})"
    return lo, lo + 15, lo + 16


def _parse_mfma(line):
    m = re.match(
        r"(\s*)(v_mfma_f32_32x32x16_\w+)\s+v\[(\d+):(\d+)\],\s*(\S+(?:\[\d+:\d+\])?),"
        r"\s*(\S+(?:\[\d+:\d+\])?),\s*(.+?)\s*$",
        line.split(";")[0].rstrip(),
    )
    return m


def _qk_split_kernel(lines, start, end, mnem, spare_lo):
    """Apply the chain split inside one kernel's [start,end) line range."""
    acc = f"v[{ACC[0]}:{ACC[1]}]"
    spare = f"v[{spare_lo}:{spare_lo + 15}]"
    # find the accumulator-chain MFMAs, in order; the steady loop holds
    # one 8-deep chain per subtile (2 per tile copy — the tile loop is
    # hand-unrolled by the buffer period, so 2 copies at NBUF=2)
    chain_idx = []
    for n in range(start, end):
        m = _parse_mfma(lines[n])
        if m and m.group(2).startswith(mnem) and f"v[{m.group(3)}:{m.group(4)}]" == acc:
            chain_idx.append(n)
    assert chain_idx and len(chain_idx) % 8 == 0, \
        f"expected a multiple of 8 QK mfma into {acc}, got {len(chain_idx)}"
    chains = [chain_idx[i:i + 8] for i in range(0, len(chain_idx), 8)]
    for ch in chains:
        m0 = _parse_mfma(lines[ch[0]])
        assert m0.group(7) == "0", f"chain start must accumulate from 0: {lines[ch[0]]}"
        for pos, n in enumerate(ch):
            m = _parse_mfma(lines[n])
            if pos == 0:
                continue
            assert m.group(7) == acc, f"chain member src2 mismatch: {lines[n]}"
            if pos % 2 == 1:  # odd members -> the spare accumulator
                src2 = "0" if pos == 1 else spare
                lines[n] = (f"{m.group(1)}{m.group(2)} {spare}, {m.group(5)}, "
                            f"{m.group(6)}, {src2}")
        # combine the two partials right after the last chain member, before
        # any consumer.  s_nop 15 + s_nop 1 = 18 wait states covers the
        # 32x32 MFMA -> VALU read-after-write hazard window.
        adds = ["\ts_nop 15", "\ts_nop 1"]
        for r in range(0, 16, 2):
            adds.append(
                f"\tv_pk_add_f32 v[{ACC[0]+r}:{ACC[0]+r+1}], "
                f"v[{ACC[0]+r}:{ACC[0]+r+1}], v[{spare_lo+r}:{spare_lo+r+1}]"
            )
        lines[ch[7]] = lines[ch[7]] + "\n" + "\n".join(adds)
    return len(chains)


def qk_split(text, symbols):
    """Apply to every kernel named in `symbols`; returns patched text."""
    lines = text.split("\n")
    spares = {sym: _spare_range(text, sym) for sym in symbols}
    for sym in symbols:
        # kernel body range
        hdr = None
        for n, l in enumerate(lines):
            if l.startswith(sym + ":"):
                hdr = n
                break
        assert hdr is not None, f"kernel label not found: {sym}"
        endn = next(n for n in range(hdr, len(lines))
                    if lines[n].lstrip().startswith(".Lfunc_end")
                    or lines[n].startswith(".Lfunc_end"))
        mnem = "v_mfma_f32_32x32x16_"
        _qk_split_kernel(lines, hdr, endn, mnem, spares[sym][0])
    text = "\n".join(lines)
    # raise the register allocation for the patched kernels
    for sym in symbols:
        new_vgpr = spares[sym][2]
        # the .amdhsa descriptor block for this kernel
        m = re.search(r"\.amdhsa_kernel\s+" + re.escape(sym), text)
        assert m, f"descriptor not found: {sym}"
        blk_end = text.index(".end_amdhsa_kernel", m.start())
        blk = text[m.start():blk_end]
        for field in ("next_free_vgpr", "accum_offset"):
            blk2 = re.sub(r"(\.amdhsa_" + field + r"\s+)\d+",
                          lambda mm: mm.group(1) + str(new_vgpr), blk)
            assert blk2 != blk or f".amdhsa_{field} {new_vgpr}" in blk, \
                f"descriptor field {field} not patched for {sym}"
            blk = blk2
        text = text[:m.start()] + blk + text[blk_end:]
        # metadata symbol feeding .vgpr_count
        pat = re.compile(r"(\.set\s+" + re.escape(sym) + r"\.num_vgpr,\s*)(\S+)")
        m2 = pat.search(text)
        assert m2, f".num_vgpr set-line not found: {sym}"
        text = text[:m2.start()] + m2.group(1) + str(new_vgpr) + text[m2.end():]
    return text


TRANSFORMS = {"qk_split": qk_split}


def apply(transform, s_in, s_out, symbols):
    text = open(s_in).read()
    text = TRANSFORMS[transform](text, symbols)
    with open(s_out, "w") as f:
        f.write(text)


if __name__ == "__main__":
    import json
    import sys

    tname, s_in, s_out, syms_json = sys.argv[1:5]
    syms = json.load(open(syms_json))
    apply(tname, s_in, s_out, list(syms.values()))
    print(f"applied {tname}: {s_in} -> {s_out}")
