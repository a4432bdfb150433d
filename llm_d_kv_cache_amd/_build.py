"""In-tree native extension builder.

Compiles the two native modules next to their sources so the built .so files
travel with the repo snapshot to GPU boxes:

  - ``_kvcore``    (g++): control plane — index, hashing, events, ZMTP.
  - ``_kvoffload`` (hipcc, --offload-arch=gfx950): data plane — storage
    offload engine with CDNA4 gather/scatter kernels.

Builds are mtime-cached and protected by a file lock so parallel test
workers don't race. hipcc cross-compiles gfx950 on machines without a GPU.
"""
from __future__ import annotations

import fcntl
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
ROCM = Path(os.environ.get("ROCM_PATH", "/opt/rocm"))
HIPCC = str(ROCM / "bin" / "hipcc")
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950").split(";")[0]


def _pybind_includes() -> list[str]:
    import pybind11

    return [pybind11.get_include(), sysconfig.get_paths()["include"]]


def _needs_build(target: Path, sources: list[Path]) -> bool:
    if not target.exists():
        return True
    t = target.stat().st_mtime
    deps: list[Path] = []
    for src in sources:
        deps.append(src)
    # any header under csrc/ is a potential dependency
    deps.extend(CSRC.rglob("*.h"))
    return any(d.stat().st_mtime > t for d in deps if d.exists())


def _run(cmd: list[str]) -> None:
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(
            f"build failed: {' '.join(cmd)}\n--- stdout ---\n{proc.stdout}"
            f"\n--- stderr ---\n{proc.stderr}"
        )


def _locked(fn):
    def wrapper(*args, **kwargs):
        lock_path = PKG_DIR / ".build.lock"
        with open(lock_path, "w") as lock:
            fcntl.flock(lock, fcntl.LOCK_EX)
            try:
                return fn(*args, **kwargs)
            finally:
                fcntl.flock(lock, fcntl.LOCK_UN)

    return wrapper


@_locked
def build_kvcore(force: bool = False) -> Path:
    """Build the CPU control-plane module with g++."""
    target = PKG_DIR / "_kvcore.so"
    sources = [CSRC / "bindings" / "kvcore_module.cc"]
    if not force and not _needs_build(target, sources):
        return target
    cmd = [
        "g++", "-O3", "-std=c++17", "-Werror=return-type", "-shared", "-fPIC", "-fvisibility=hidden",
        "-pthread", "-Wall",
    ]
    for inc in _pybind_includes():
        cmd += ["-I", inc]
    cmd += [str(s) for s in sources]
    cmd += ["-o", str(target)]
    _run(cmd)
    return target


@_locked
def build_kvoffload(force: bool = False) -> Path:
    """Build the GPU data-plane module with hipcc for gfx950."""
    target = PKG_DIR / "_kvoffload.so"
    sources = [CSRC / "bindings" / "kvoffload_module.cc"]
    hip_sources = sorted((CSRC / "offload").glob("*.hip"))
    all_sources = sources + hip_sources
    if not all(s.exists() for s in sources):
        raise FileNotFoundError("offload sources missing")
    if not force and not _needs_build(target, all_sources):
        return target
    cmd = [
        HIPCC, f"--offload-arch={GFX_ARCH}", "-O3", "-std=c++17", "-Werror=return-type", "-shared",
        "-fPIC", "-fvisibility=hidden", "-pthread", "-Wall",
        "-Wno-unused-result",
    ]
    for inc in _pybind_includes():
        cmd += ["-I", inc]
    cmd += [str(s) for s in all_sources]
    cmd += ["-lnuma", "-o", str(target)]
    _run(cmd)
    return target


def build_all(force: bool = False) -> None:
    build_kvcore(force=force)
    if (CSRC / "bindings" / "kvoffload_module.cc").exists():
        build_kvoffload(force=force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
    print("built:", [p.name for p in PKG_DIR.glob("_*.so")])
