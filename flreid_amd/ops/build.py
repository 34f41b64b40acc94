"""In-tree build of the MI355X HIP extension.

`python -m flreid_amd.ops.build` (or __graft_entry__.build()) compiles
ops/csrc/*.{hip,cpp} with hipcc for gfx950 into flreid_amd/ops/_flreid_hip.so.
The built .so travels to GPU boxes with the repo snapshot.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
OUT = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_flreid_hip.so")

SOURCES = ["module.cpp", "elementwise.hip", "distance.hip", "window_attn.hip", "triplet.hip", "adaptive_gemm.hip", "conv3x3.hip", "conv3x3_img.hip", "conv3x3_img_ldsw.hip", "bn_train.hip", "drift.hip", "kd.hip", "patch_merge.hip"]


def _pybind11_includes():
    import pybind11
    return [pybind11.get_include()]


def needs_rebuild() -> bool:
    if not os.path.exists(OUT):
        return True
    out_mtime = os.path.getmtime(OUT)
    for f in os.listdir(CSRC):
        if os.path.getmtime(os.path.join(CSRC, f)) > out_mtime:
            return True
    return False


def build(force: bool = False, arch: str = "gfx950", verbose: bool = True,
          debug: bool = False, asan: bool = False) -> str:
    """debug=True (or --debug / FLREID_BUILD_DEBUG=1): -g -O1 — the
    kernel-debug build for rocgdb / serialized-fault runs.
    asan=True (or --asan / FLREID_BUILD_ASAN=1): additionally instruments
    the HOST side with AddressSanitizer; run python with
    LD_PRELOAD=$(/opt/rocm/lib/llvm/bin/clang
    -print-file-name=libclang_rt.asan-x86_64.so) ASAN_OPTIONS=detect_leaks=0.
    Release build is the default."""
    if not force and not needs_rebuild():
        if verbose:
            print(f"[flreid build] up to date: {OUT}")
        return OUT
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    includes = [sysconfig.get_paths()["include"]] + _pybind11_includes()
    debug = debug or os.environ.get("FLREID_BUILD_DEBUG", "0") == "1"
    asan = asan or os.environ.get("FLREID_BUILD_ASAN", "0") == "1"
    opt = ["-g", "-O1"] if (debug or asan) else ["-O3"]
    extra = []
    if asan:
        # host-side ASan only: device code stays uninstrumented (xnack-free)
        extra = ["-fsanitize=address", "-shared-libsan",
                 "-Xarch_device", "-fno-sanitize=all"]
    cmd = [
        hipcc, f"--offload-arch={arch}", *opt, *extra, "-std=c++17", "-fPIC",
        "-shared", "-fvisibility=hidden",
        *[f"-I{p}" for p in includes],
        *[os.path.join(CSRC, s) for s in SOURCES],
        "-o", OUT,
    ]
    if verbose:
        print("[flreid build]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build(force=any(f in sys.argv for f in ("--force", "--debug", "--asan")),
          debug="--debug" in sys.argv, asan="--asan" in sys.argv)
