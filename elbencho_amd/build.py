"""In-tree build of the elbencho_amd native engine (_core.so).

Compiles the C++/HIP sources with hipcc for gfx950 (MI355X). The .so is
placed inside the package directory so it travels with repo snapshots
(gpurun) while staying out of git history (.gitignore: *.so).

Usage:
    python -m elbencho_amd.build          # build if stale
    python -m elbencho_amd.build --force  # rebuild
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")
OUT_SO = os.path.join(PKG_DIR, "_core.so")
BUILD_DIR = os.path.join(PKG_DIR, "csrc", ".build")

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
GPU_ARCH = os.environ.get("EB_GPU_ARCH", "gfx950")

SOURCES = ["module.cpp", "engine.cpp", "gpu_kernels.hip"]
HEADERS = [
    "common.h", "engine.h", "gpu.h", "histogram.h", "offsetgen.h",
    "rand.h", "rate.h", "uring.h",
]


def _include_flags() -> list[str]:
    import pybind11

    return [
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
    ]


def _needs_build() -> bool:
    if not os.path.exists(OUT_SO):
        return True
    so_mtime = os.path.getmtime(OUT_SO)
    for f in SOURCES + HEADERS + ["../build.py"]:
        p = os.path.join(CSRC, f)
        if os.path.exists(p) and os.path.getmtime(p) > so_mtime:
            return True
    return False


def build(force: bool = False, verbose: bool = True) -> str:
    """Compile _core.so for gfx950. Returns the path to the .so."""
    if not force and not _needs_build():
        return OUT_SO

    os.makedirs(BUILD_DIR, exist_ok=True)

    common = [
        f"--offload-arch={GPU_ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-Wall",
        "-Wno-unused-function",
    ] + _include_flags()

    objs = []
    procs = []
    for src in SOURCES:
        obj = os.path.join(BUILD_DIR, src.replace("/", "_") + ".o")
        objs.append(obj)
        cmd = [HIPCC, *common, "-c", os.path.join(CSRC, src), "-o", obj]
        if verbose:
            print("[build]", " ".join(cmd), file=sys.stderr)
        procs.append(subprocess.Popen(cmd, stderr=subprocess.PIPE, text=True))

    errors = []
    for src, p in zip(SOURCES, procs):
        _, err = p.communicate()
        if p.returncode != 0:
            errors.append(f"--- {src} ---\n{err}")
        elif err.strip() and verbose:
            print(err, file=sys.stderr)
    if errors:
        raise RuntimeError("hipcc compilation failed:\n" + "\n".join(errors))

    link = [HIPCC, "-shared", "-fPIC", *objs, "-o", OUT_SO]
    if verbose:
        print("[build]", " ".join(link), file=sys.stderr)
    res = subprocess.run(link, stderr=subprocess.PIPE, text=True)
    if res.returncode != 0:
        raise RuntimeError("link failed:\n" + res.stderr)

    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT_SO)
