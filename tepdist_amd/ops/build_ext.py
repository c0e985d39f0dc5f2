"""In-tree hipcc build of the gfx950 kernel extension.

Builds tepdist_amd/ops/_tepdist_hip.so directly with hipcc (cross-compiles
fine on a GPU-less box; the .so travels to the GPU box with the repo
snapshot). No torch headers, no hipify: pure HIP + pybind11.
"""

from __future__ import annotations

import concurrent.futures as cf
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
OUT_SO = OPS_DIR / "_tepdist_hip.so"
BUILD = CSRC / "build"

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("TEPDIST_GFX_ARCH", "gfx950")

SOURCES = [
    "gemm.hip",
    "gemm256.hip",
    "attention.hip",
    "transpose.hip",
    "layernorm.hip",
    "softmax.hip",
    "embedding.hip",
    "cross_entropy.hip",
    "dropout.hip",
    "elementwise.hip",
    "adamw.hip",
    "llama_ops.hip",
    "debug.hip",
    "bindings.cpp",
]


def _includes():
    import pybind11
    return [
        "-I", pybind11.get_include(),
        "-I", sysconfig.get_paths()["include"],
        "-I", str(CSRC),
    ]


def _needs_build() -> bool:
    if not OUT_SO.exists():
        return True
    so_mtime = OUT_SO.stat().st_mtime
    deps = [CSRC / s for s in SOURCES] + [CSRC / "common.h", CSRC / "kernels.h",
                                          Path(__file__)]
    return any(d.stat().st_mtime > so_mtime for d in deps)


RT_CSRC = OPS_DIR.parent / "runtime" / "csrc"
RT_SO = OPS_DIR.parent / "runtime" / "_tepdist_rt.so"
CXX = os.environ.get("CXX", "g++")


def build_rt(force: bool = False, verbose: bool = True) -> Path:
    """Builds the pure-C++ native runtime core (scheduler simulator,
    lifetime/GC, dominance tree) — no HIP dependency, imports on CPU."""
    src = RT_CSRC / "rt_core.cpp"
    if not force and RT_SO.exists() and \
            RT_SO.stat().st_mtime > src.stat().st_mtime:
        return RT_SO
    cmd = [CXX, "-O3", "-std=c++17", "-shared", "-fPIC",
           *_includes(), str(src), "-o", str(RT_SO)]
    if verbose:
        print("[build_ext]", " ".join(cmd), flush=True)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            f"rt_core build failed:\n{r.stdout}\n{r.stderr}")
    return RT_SO


BLT_SO = OPS_DIR / "_tepdist_blt.so"


def build_blt(force: bool = False, verbose: bool = True) -> Path:
    """Builds the hipBLASLt fused-epilogue driver (host-only C++, links
    libhipblaslt)."""
    src = CSRC / "blaslt.cpp"
    if not force and BLT_SO.exists() and \
            BLT_SO.stat().st_mtime > src.stat().st_mtime:
        return BLT_SO
    cmd = [HIPCC, "-O3", "-std=c++17", "-shared", "-fPIC",
           *_includes(), str(src), "-L", "/opt/rocm/lib", "-lhipblaslt",
           "-o", str(BLT_SO)]
    if verbose:
        print("[build_ext]", " ".join(cmd), flush=True)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"blaslt build failed:\n{r.stdout}\n{r.stderr}")
    return BLT_SO


def build(force: bool = False, verbose: bool = True) -> Path:
    build_rt(force, verbose)
    build_blt(force, verbose)
    if not force and not _needs_build():
        return OUT_SO
    BUILD.mkdir(exist_ok=True)
    inc = _includes()
    base = [HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
            "-Wno-unused-result"] + inc

    def compile_one(src: str) -> Path:
        obj = BUILD / (src.replace(".", "_") + ".o")
        lang = ["-x", "hip"] if src.endswith(".hip") else []
        cmd = base + lang + ["-c", str(CSRC / src), "-o", str(obj)]
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(
                f"hipcc failed for {src}:\n{r.stdout}\n{r.stderr}")
        return obj

    with cf.ThreadPoolExecutor(max_workers=min(8, len(SOURCES))) as ex:
        objs = list(ex.map(compile_one, SOURCES))

    link = [HIPCC, f"--offload-arch={ARCH}", "-shared", "-fPIC",
            *(str(o) for o in objs), "-o", str(OUT_SO)]
    if verbose:
        print("[build_ext]", " ".join(link), flush=True)
    r = subprocess.run(link, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"hipcc link failed:\n{r.stdout}\n{r.stderr}")
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {OUT_SO}")
