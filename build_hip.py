"""In-tree build of the CDNA4 HIP extension.

``python build_hip.py`` (or setup.py build_ext --inplace, or
__graft_entry__.build()) compiles csrc/*.hip + bindings with
``hipcc --offload-arch=gfx950`` into
asyncframework_amd/_hip_core.<abi>.so. The .so is git-ignored but travels
with the gpurun snapshot, so GPU boxes never JIT."""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "hipcc")

SOURCES = [ROOT / "csrc" / "kernels.hip", ROOT / "csrc" / "bindings.cpp",
           ROOT / "csrc" / "libsvm_parser.cpp",
           ROOT / "csrc" / "engine_native.cpp"]
HEADERS = [ROOT / "csrc" / "philox.h"]


def so_path() -> Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return ROOT / "asyncframework_amd" / f"_hip_core{suffix}"


def needs_build(out: Path) -> bool:
    if not out.exists():
        return True
    mtime = out.stat().st_mtime
    return any(s.stat().st_mtime > mtime for s in SOURCES + HEADERS)


def build(force: bool = False, verbose: bool = True) -> Path:
    out = so_path()
    if not force and not needs_build(out):
        if verbose:
            print(f"[build_hip] up to date: {out}")
        return out
    import pybind11
    inc_py = sysconfig.get_paths()["include"]
    inc_pb = pybind11.get_include()
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-shared",
        "-fPIC", "-DNDEBUG", f"-I{inc_py}", f"-I{inc_pb}",
        *[str(s) for s in SOURCES], "-o", str(out),
    ]
    if verbose:
        print("[build_hip]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
