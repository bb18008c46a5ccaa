"""In-tree native build for torchft_amd.

Builds two extensions directly (no JIT cache — the .so files live in-tree so
they travel to GPU boxes with the repo snapshot):

* ``torchft_amd/_ftcore``  — C++ coordination core (lighthouse/manager), CPU-only,
  compiled with g++; no torch dependency so it builds in seconds anywhere.
* ``torchft_amd/_hip_kernels`` — CDNA4 HIP kernels (gfx950), compiled with hipcc
  against libtorch; cross-compiles on CPU-only boxes.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
CSRC = REPO / "csrc"
PKG = REPO / "torchft_amd"

EXT_SUFFIX = sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def _newer(target: Path, sources: list[Path]) -> bool:
    if not target.exists():
        return False
    t = target.stat().st_mtime
    return all(s.stat().st_mtime < t for s in sources)


def _run(cmd: list[str]) -> None:
    print("[torchft_amd build]", " ".join(str(c) for c in cmd), flush=True)
    subprocess.check_call([str(c) for c in cmd])


def build_ftcore(force: bool = False) -> Path:
    """Compile the C++ coordination core with g++."""
    import pybind11

    out = PKG / f"_ftcore{EXT_SUFFIX}"
    sources = [
        CSRC / "coord" / "lighthouse.cpp",
        CSRC / "coord" / "manager.cpp",
        CSRC / "coord" / "bindings.cpp",
    ]
    headers = [CSRC / "coord" / "wire.h", CSRC / "coord" / "coord.h"]
    if not force and _newer(out, sources + headers):
        return out
    py_inc = sysconfig.get_paths()["include"]
    cmd = [
        "g++", "-O2", "-g", "-std=c++17", "-shared", "-fPIC", "-pthread",
        f"-I{pybind11.get_include()}", f"-I{py_inc}",
        *[str(s) for s in sources],
        "-o", str(out),
    ]
    _run(cmd)
    return out


def build_hip_kernels(force: bool = False) -> Path | None:
    """Compile the gfx950 HIP kernel extension via torch.utils.cpp_extension.

    Cross-compiles on machines with no GPU (hipcc targets gfx950 regardless).
    The built .so is copied in-tree (torchft_amd/_hip_kernels.so) so the repo
    snapshot shipped to GPU boxes carries it.
    """
    src_dir = CSRC / "kernels"
    sources = sorted(src_dir.glob("*.hip")) + [src_dir / "bindings.cpp"]
    headers = [src_dir / "kernels.h"]
    out = PKG / "_hip_kernels.so"
    if not force and _newer(out, sources + headers):
        return out

    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    build_dir = REPO / "build" / "hip_kernels"
    build_dir.mkdir(parents=True, exist_ok=True)
    load(
        name="_hip_kernels",
        sources=[str(s) for s in sources],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        build_directory=str(build_dir),
        verbose=True,
    )
    built = build_dir / "_hip_kernels.so"
    import shutil

    shutil.copy2(built, out)
    return out


def build_all(force: bool = False) -> None:
    build_ftcore(force=force)
    build_hip_kernels(force=force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
