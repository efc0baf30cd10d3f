"""In-tree build of the gfx950 HIP kernels (plain hipcc, no hipify, no JIT
cache): produces ps_pytorch_amd/ops/libps_hip.so which travels with the repo
snapshot to GPU boxes. hipcc cross-compiles without a GPU present."""
from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
KERNEL_DIR = OPS_DIR / "kernels"
SO_PATH = OPS_DIR / "libps_hip.so"
ARCH = os.environ.get("PS_ROCM_ARCH", "gfx950")

SOURCES = sorted(KERNEL_DIR.glob("*.hip"))


def _hipcc() -> str:
    for cand in (os.environ.get("HIPCC"), "/opt/rocm/bin/hipcc", "hipcc"):
        if cand and (os.path.sep not in cand or os.path.exists(cand)):
            return cand
    return "hipcc"


def needs_build() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    deps = list(SOURCES) + list(KERNEL_DIR.glob("*.h")) + [Path(__file__)]
    return any(p.stat().st_mtime > so_mtime for p in deps)


def build(verbose: bool = True, force: bool = False) -> Path:
    if not force and not needs_build():
        return SO_PATH
    cmd = [
        _hipcc(), f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC", "-o", str(SO_PATH),
    ] + [str(s) for s in SOURCES]
    if verbose:
        print("[ps_pytorch_amd.ops.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
