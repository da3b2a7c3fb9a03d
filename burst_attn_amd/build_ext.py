"""In-tree build of the gfx950 HIP extension (burst_attn_amd/_C.so).

Driven by hipcc directly (no JIT cache — the .so must live in-tree so the
gpurun snapshot carries it).  Called by __graft_entry__.build().
"""

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")
BUILD = os.path.join(PKG_DIR, "csrc", "_build")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")

DEVICE_SOURCES = ["attn_fwd.hip", "attn_bwd.hip"]
HOST_SOURCES = ["ext_torch.cpp"]


def _run(cmd, verbose):
    if verbose:
        print("+ " + " ".join(cmd), flush=True)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout + r.stderr)
        raise RuntimeError(f"build command failed: {' '.join(cmd[:3])} ...")
    return r


def _newer(src, dst):
    return not os.path.exists(dst) or os.path.getmtime(src) > os.path.getmtime(dst)


def build(verbose=True, force=False):
    import torch
    from torch.utils import cpp_extension

    os.makedirs(BUILD, exist_ok=True)
    out_so = os.path.join(PKG_DIR, "_C.so")

    torch_includes = cpp_extension.include_paths()
    py_include = sysconfig.get_paths()["include"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)

    common = [
        f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-DNDEBUG",
    ]
    objs = []
    hdr = os.path.join(CSRC, "attn_common.h")
    abihdr = os.path.join(PKG_DIR, "..", "include", "burst_attn_hip.h")
    for src in DEVICE_SOURCES:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD, src.replace(".hip", ".o"))
        objs.append(op)
        if force or _newer(sp, op) or _newer(hdr, op) or _newer(abihdr, op):
            _run([HIPCC, *common, "-c", sp, "-o", op], verbose)
    for src in HOST_SOURCES:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD, src.replace(".cpp", ".o"))
        objs.append(op)
        if force or _newer(sp, op) or _newer(abihdr, op):
            cmd = [
                HIPCC, *common,
                f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
                "-DTORCH_EXTENSION_NAME=_C",
                "-DTORCH_API_INCLUDE_EXTENSION_H",
                "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
            ]
            for inc in torch_includes + [py_include]:
                cmd += ["-I", inc]
            cmd += ["-c", sp, "-o", op]
            _run(cmd, verbose)
    if force or any(_newer(o, out_so) for o in objs):
        lib_dirs = cpp_extension.library_paths()
        cmd = [HIPCC, "-shared", "-fPIC", *objs, "-o", out_so]
        for ld in lib_dirs:
            cmd += [f"-L{ld}", f"-Wl,-rpath,{ld}"]
        cmd += ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
                "-ltorch_hip", "-lc10_hip", "-lamdhip64"]
        _run(cmd, verbose)
    _build_asm_hsaco(verbose, force)
    if verbose:
        print(f"built {out_so}")
    return out_so


# ---- .s -> .hsaco side-build (the round-3 asm on-ramp) ----------------
# The forward is ALSO shipped as a hipModule-loadable code object built
# from its own compiler-generated assembly (tools/asm_probe recipe).
# BA_FWD_ASM=1 routes fwd_accum through it at runtime — today that
# re-assembles the unmodified .s (a parity/latency check of the flow);
# round 3 patches the .s steady loop with the hand schedule before the
# assemble step.
LLVM = os.path.join(os.path.dirname(os.path.dirname(HIPCC)), "lib", "llvm",
                    "bin") if "rocm" in HIPCC else "/opt/rocm/lib/llvm/bin"


def _build_asm_hsaco(verbose, force):
    import json
    import re

    src = os.path.join(CSRC, "attn_fwd.hip")
    s_path = os.path.join(BUILD, "attn_fwd_scaffold.s")
    o_path = os.path.join(BUILD, "attn_fwd_scaffold.o")
    hsaco = os.path.join(PKG_DIR, "_asm_fwd.hsaco")
    syms = os.path.join(PKG_DIR, "_asm_fwd_syms.json")
    patch_mod = os.path.join(PKG_DIR, "..", "tools", "s_patch.py")
    if not (force or _newer(src, hsaco) or _newer(patch_mod, hsaco)):
        return hsaco
    _run([HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-DNDEBUG",
          "-S", "--cuda-device-only", src, "-o", s_path], verbose)
    clang = os.path.join(LLVM, "clang")
    lld = os.path.join(LLVM, "ld.lld")
    _run([clang, "-x", "assembler", "-target", "amdgcn-amd-amdhsa",
          f"-mcpu={ARCH}", "-c", s_path, "-o", o_path], verbose)
    _run([lld, "-shared", o_path, "-o", hsaco], verbose)
    # record the mangled names of the dispatchable variants (the SUBT=1
    # NBUF=2 production instantiations, accum form, fp16 and bf16)
    text = open(s_path).read()
    names = {}
    for key, pat in (
        ("f16_accum", r"_ZN12_GLOBAL__N_115attn_fwd_kernelIDF16_Li128ELi64ELi1ELi0ELi1ELi512ELi2ELi0E\S*"),
        ("bf16_accum", r"_ZN12_GLOBAL__N_115attn_fwd_kernelIDF16bLi128ELi64ELi1ELi0ELi1ELi512ELi2ELi0E\S*"),
    ):
        m = re.search(r"\.globl\t(" + pat + ")", text)
        if m:
            names[key] = m.group(1)
    with open(syms, "w") as f:
        json.dump(names, f, indent=1)
    # hand-scheduled variant (BA_FWD_ASM=2): apply tools/s_patch.py
    # transforms to the .s and assemble a second module
    import importlib.util

    spec = importlib.util.spec_from_file_location("s_patch", patch_mod)
    s_patch = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(s_patch)
    sp_path = os.path.join(BUILD, "attn_fwd_patched.s")
    spo_path = os.path.join(BUILD, "attn_fwd_patched.o")
    hsaco_p = os.path.join(PKG_DIR, "_asm_fwd_p.hsaco")
    s_patch.apply("qk_split", s_path, sp_path, list(names.values()))
    _run([clang, "-x", "assembler", "-target", "amdgcn-amd-amdhsa",
          f"-mcpu={ARCH}", "-c", sp_path, "-o", spo_path], verbose)
    _run([lld, "-shared", spo_path, "-o", hsaco_p], verbose)
    if verbose:
        print(f"built {hsaco} and {hsaco_p} (qk_split)")
    return hsaco


if __name__ == "__main__":
    build(force="--force" in sys.argv)
