"""In-tree build of the HIP kernel library (gfx950 only, no JIT cache).

The library is pure HIP with a C ABI (ctypes-loaded) — no torch/pybind
link, so there is nothing to version-match.  ``hipcc --offload-arch=gfx950``
cross-compiles fine on a GPU-less box; the produced ``.so`` travels to the
GPU box with the repo snapshot.
"""

from __future__ import annotations

import os
import subprocess
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
LIB_PATH = PKG_DIR / "_vamd_hip.so"
SOURCES = ["scheduler_kernels.hip", "cycle_runner.hip"]
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("VAMD_GPU_ARCH", "gfx950")


def _mtime(p: Path) -> float:
    try:
        return p.stat().st_mtime
    except FileNotFoundError:
        return 0.0


def needs_build() -> bool:
    if not LIB_PATH.exists():
        return True
    lib_t = _mtime(LIB_PATH)
    deps = [CSRC / s for s in SOURCES] + [CSRC / "vamd_api.h"]
    return any(_mtime(d) > lib_t for d in deps)


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_build():
        return LIB_PATH
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-fPIC", "-shared", "-o", str(LIB_PATH),
    ] + [str(CSRC / s) for s in SOURCES]
    if verbose:
        print("[vamd build]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True, cwd=str(CSRC))
    return LIB_PATH


if __name__ == "__main__":
    build(force="--force" in os.sys.argv)
