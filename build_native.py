#!/usr/bin/env python3
"""Build the native TFRecord codec extension (host C++ + gfx950 HIP kernels).

Drives hipcc directly — one extension, compiled in-tree so the .so ships to
the GPU box with the repo snapshot:
    spark_tfrecord_amd/_native.so
No hipify, no CUDA-compat layers: csrc/ is written as HIP/CDNA4 source.
"""

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
OUT = ROOT / "spark_tfrecord_amd" / "_native.so"
SOURCES = [ROOT / "csrc" / "ext.cpp", ROOT / "csrc" / "hip" / "kernels.hip",
           ROOT / "csrc" / "hip" / "inflate.hip"]
HEADERS = list((ROOT / "csrc").rglob("*.h"))


def _include_flags():
    import pybind11

    incs = [sysconfig.get_paths()["include"], pybind11.get_include()]
    return [f"-I{p}" for p in incs]


def needs_rebuild() -> bool:
    if not OUT.exists():
        return True
    out_mtime = OUT.stat().st_mtime
    return any(p.stat().st_mtime > out_mtime for p in SOURCES + HEADERS)


def build(force: bool = False, verbose: bool = True,
          sanitize: bool = False) -> Path:
    if not force and not needs_rebuild():
        return OUT
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    objdir = ROOT / "build"
    objdir.mkdir(exist_ok=True)
    # host-side ASan/UBSan build for the C++ codec + IO pool (SURVEY.md §5
    # race-detection/sanitizer row); device code is unaffected
    san_flags = (["-Xarch_host", "-fsanitize=address,undefined",
                  "-fno-omit-frame-pointer"] if sanitize else [])
    objs = []
    for src in SOURCES:
        obj = objdir / (src.stem + (".san.o" if sanitize else ".o"))
        cmd = [
            "hipcc",
            "--offload-arch=gfx950",
            "-O3",
            "-std=c++17",
            "-fPIC",
            "-DTFREC_WITH_HIP",
            "-Xarch_host",
            "-msse4.2",
            *san_flags,
            "-x",
            "hip",
            *_include_flags(),
            "-c",
            str(src),
            "-o",
            str(obj),
        ]
        if verbose:
            print("[build_native]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True, cwd=ROOT)
        objs.append(obj)
    link = ["hipcc", "-shared", "-fPIC", *(
        ["-fsanitize=address,undefined", "-shared-libasan"] if sanitize else []),
        *map(str, objs), "-o", str(OUT)]
    if verbose:
        print("[build_native]", " ".join(link), flush=True)
    subprocess.run(link, check=True, cwd=ROOT)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv, sanitize="--sanitize" in sys.argv)
    print(f"built {OUT}")
