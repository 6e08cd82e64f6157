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
    if force or needs_build(out):
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
    elif verbose:
        print(f"[build_hip] up to date: {out}")
    build_dist_core(verbose=verbose)
    return out


DIST_SRC = ROOT / "csrc" / "server_dist.cpp"
DIST_DIR = ROOT / "asyncframework_amd" / "_dist_build"


def build_dist_core(verbose: bool = True) -> Path:
    """The C++ dist server is a torch extension (needs c10d headers) built
    into its own in-tree .so so _hip_core stays torch-free."""
    out = DIST_DIR / "_dist_core.so"
    if out.exists() and out.stat().st_mtime >= DIST_SRC.stat().st_mtime:
        if verbose:
            print(f"[build_hip] up to date: {out}")
        return out
    from torch.utils.cpp_extension import load
    DIST_DIR.mkdir(parents=True, exist_ok=True)
    load(name="_dist_core", sources=[str(DIST_SRC)],
         build_directory=str(DIST_DIR), verbose=verbose)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
