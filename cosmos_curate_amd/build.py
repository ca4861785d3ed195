"""Build libcchot.so (the C-ABI hot-path library) with hipcc for gfx950.

Built IN-TREE (cosmos_curate_amd/lib/libcchot.so) so the .so travels with
the gpurun snapshot; .gitignore keeps it out of history.  Idempotent: object
files rebuild only when sources are newer.

Per-file flags:
- cc_pixel.hip: -ffp-contract=off — the pixel kernels promise bit-exact f32
  arithmetic vs the numpy oracle (oracle/color.py); FMA contraction would
  change rounding.
- cc_gemm.hip: default contraction (MFMA does the math anyway).
"""

from __future__ import annotations

import pathlib
import subprocess

PKG = pathlib.Path(__file__).resolve().parent
CSRC = PKG / "csrc"
LIBDIR = PKG / "lib"
LIB = LIBDIR / "libcchot.so"

HIPCC = "hipcc"
ARCH = "--offload-arch=gfx950"
COMMON = ["-O3", "-std=c++17", "-fPIC", f"-I{PKG.parent}"]

SOURCES: list[tuple[str, list[str]]] = [
    ("cc_demux.cpp", []),
    ("cc_decode.cpp", []),
    ("cc_pixel.hip", [ARCH, "-ffp-contract=off"]),
    ("cc_gemm.hip", [ARCH]),
    ("cc_dedup.hip", [ARCH]),
    ("cc_ln.hip", [ARCH]),
    ("cc_attn.hip", [ARCH]),
]


def _needs_build(src: pathlib.Path, obj: pathlib.Path) -> bool:
    if not obj.exists():
        return True
    dep = [src, CSRC / "cc_common.hpp", PKG.parent / "include" / "cc_hotpath.h"]
    return any(d.stat().st_mtime > obj.stat().st_mtime for d in dep if d.exists())


def build(verbose: bool = True) -> pathlib.Path:
    LIBDIR.mkdir(exist_ok=True)
    objs = []
    rebuilt = False
    for name, extra in SOURCES:
        src = CSRC / name
        obj = LIBDIR / (src.stem + ".o")
        objs.append(obj)
        if _needs_build(src, obj):
            cmd = [HIPCC, *COMMON, *extra, "-c", str(src), "-o", str(obj)]
            if verbose:
                print("[cc build]", " ".join(cmd))
            subprocess.run(cmd, check=True)
            rebuilt = True
    if rebuilt or not LIB.exists():
        cmd = [HIPCC, ARCH, "-shared", "-fPIC", *map(str, objs), "-o", str(LIB)]
        if verbose:
            print("[cc build]", " ".join(cmd))
        subprocess.run(cmd, check=True)
    return LIB


if __name__ == "__main__":
    build()
