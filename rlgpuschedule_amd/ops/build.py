"""In-tree hipcc build of the CDNA4 extensions (gfx950 only, no JIT cache).

Built .so files live next to this file so they travel with the repo snapshot
to GPU boxes (they are git-ignored but NOT gpurun-ignored).
"""
from __future__ import annotations

import os
import subprocess
from pathlib import Path

PKG = Path(__file__).resolve().parent.parent
CSRC = PKG / "csrc"
OUT = PKG / "ops"

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("RLS_OFFLOAD_ARCH", "gfx950")

# name -> sources
_LIBS = {
    "librlsched_env.so": ["env_step.hip"],
    "librlsched_nn.so": ["nn_kernels.hip"],
    "librlsched_train.so": ["train_kernels.hip"],
    "librlsched_gru.so": ["gru_train.hip"],
}


def _needs_build(so: Path, srcs: list[Path]) -> bool:
    if not so.exists():
        return True
    mt = so.stat().st_mtime
    hdrs = list(CSRC.glob("*.h"))
    return any(p.stat().st_mtime > mt for p in srcs + hdrs)


def build_lib(name: str, verbose: bool = False) -> Path:
    srcs = [CSRC / s for s in _LIBS[name]]
    missing = [s for s in srcs if not s.exists()]
    if missing:
        raise FileNotFoundError(f"missing sources for {name}: {missing}")
    so = OUT / name
    if not _needs_build(so, srcs):
        return so
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-fPIC", "-shared", "-o", str(so),
    ] + [str(s) for s in srcs]
    if verbose:
        print("+", " ".join(cmd))
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"hipcc failed for {name}:\n{r.stdout}\n{r.stderr}")
    return so


def build_all(verbose: bool = False) -> list[Path]:
    out = []
    for name, srcs in _LIBS.items():
        if all((CSRC / s).exists() for s in srcs):
            out.append(build_lib(name, verbose))
    return out


if __name__ == "__main__":
    for p in build_all(verbose=True):
        print("built", p)
