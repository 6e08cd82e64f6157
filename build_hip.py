"""In-tree build of the CDNA4 HIP extension, with source-hash provenance.

``python build_hip.py`` (or setup.py build_ext --inplace, or
__graft_entry__.build()) compiles csrc/*.hip + bindings with
``hipcc --offload-arch=gfx950`` into
asyncframework_amd/_hip_core.<abi>.so. The .so is git-ignored but travels
with the gpurun snapshot, so GPU boxes never JIT.

Provenance (round-1 postmortem): a stale prebuilt .so once shipped a
deadlocked engine because rebuilds were mtime-gated and snapshot copies
reset mtimes. Now every build embeds a sha256 of the sources
(``-DASYNCAMD_SRC_HASH``) and writes a sidecar json; ``ensure_fresh()``
(called from the package __init__) recomputes the source hash and
force-rebuilds on any mismatch, so a source-edited tree can never load a
stale binary silently."""

from __future__ import annotations

import hashlib
import json
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
           ROOT / "csrc" / "engine_native.cpp",
           ROOT / "csrc" / "engine_resident.hip"]
HEADERS = [ROOT / "csrc" / "philox.h", ROOT / "csrc" / "multi_update.h",
           ROOT / "csrc" / "grad_wave.h"]
DIST_SRC = ROOT / "csrc" / "server_dist.cpp"
DIST_HIP = ROOT / "csrc" / "dist_update.hip"
DIST_DIR = ROOT / "asyncframework_amd" / "_dist_build"
PROV = ROOT / "asyncframework_amd" / "_hip_core.provenance.json"


def so_path() -> Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return ROOT / "asyncframework_amd" / f"_hip_core{suffix}"


def src_hash() -> str:
    """sha256 over the _hip_core sources + target arch (hex, 16 chars)."""
    h = hashlib.sha256()
    h.update(ARCH.encode())
    for s in sorted(SOURCES + HEADERS):
        h.update(s.name.encode())
        h.update(s.read_bytes())
    return h.hexdigest()[:16]


def dist_src_hash() -> str:
    h = hashlib.sha256()
    h.update(ARCH.encode())
    h.update(DIST_SRC.read_bytes())
    h.update(DIST_HIP.read_bytes())
    return h.hexdigest()[:16]


def _read_prov() -> dict:
    try:
        return json.loads(PROV.read_text())
    except (OSError, ValueError):
        return {}


def _write_prov(**kv) -> None:
    prov = _read_prov()
    prov.update(kv)
    PROV.write_text(json.dumps(prov, indent=1) + "\n")


def needs_build(out: Path) -> bool:
    if not out.exists():
        return True
    return _read_prov().get("src_hash") != src_hash()


def build(force: bool = False, verbose: bool = True) -> Path:
    out = so_path()
    cur = src_hash()
    if force or needs_build(out):
        import pybind11
        inc_py = sysconfig.get_paths()["include"]
        inc_pb = pybind11.get_include()
        cmd = [
            HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-shared",
            "-fPIC", "-DNDEBUG", f'-DASYNCAMD_SRC_HASH="{cur}"',
            f"-I{inc_py}", f"-I{inc_pb}",
            *[str(s) for s in SOURCES], "-o", str(out),
        ]
        if verbose:
            print("[build_hip]", " ".join(cmd))
        subprocess.run(cmd, check=True)
        _write_prov(src_hash=cur, arch=ARCH)
    elif verbose:
        print(f"[build_hip] up to date: {out} ({cur})")
    build_dist_core(verbose=verbose)
    return out


def build_dist_core(force: bool = False, verbose: bool = True) -> Path:
    """The C++ dist server is a torch extension (needs c10d headers) built
    into its own in-tree .so so _hip_core stays torch-free."""
    out = DIST_DIR / "_dist_core.so"
    cur = dist_src_hash()
    if not force and out.exists() and _read_prov().get("dist_src_hash") == cur:
        if verbose:
            print(f"[build_hip] up to date: {out} ({cur})")
        return out
    from torch.utils.cpp_extension import load
    DIST_DIR.mkdir(parents=True, exist_ok=True)
    if force and out.exists():
        out.unlink()  # torch load() otherwise reuses by its own mtime check
    load(name="_dist_core", sources=[str(DIST_SRC), str(DIST_HIP)],
         build_directory=str(DIST_DIR), verbose=verbose)
    _write_prov(dist_src_hash=cur)
    return out


def ensure_fresh(verbose: bool = False) -> None:
    """Import-time guard: rebuild any extension whose recorded source hash
    differs from the sources on disk. No-op (cheap hashing only) when
    everything matches."""
    if needs_build(so_path()):
        print("[build_hip] provenance mismatch or missing _hip_core.so — "
              "rebuilding for", ARCH, file=sys.stderr)
        build(force=True, verbose=True)
        return  # build() also refreshes _dist_core
    out = DIST_DIR / "_dist_core.so"
    if not out.exists() or _read_prov().get("dist_src_hash") != dist_src_hash():
        print("[build_hip] provenance mismatch or missing _dist_core.so — "
              "rebuilding", file=sys.stderr)
        build_dist_core(force=True, verbose=True)


if __name__ == "__main__":
    build(force="--force" in sys.argv)
