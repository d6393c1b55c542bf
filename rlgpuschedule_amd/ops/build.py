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
    "librlsched_mega.so": ["rollout_mega.hip"],
}


def source_hash(name: str) -> str:
    """sha256-prefix over this lib's sources + all shared headers.
    Embedded at compile time (-DRLS_SOURCE_HASH) and checked at load:
    a committed .so whose sources changed can never silently test old
    kernels."""
    import hashlib
    h = hashlib.sha256()
    srcs = [CSRC / s for s in _LIBS[name]] + sorted(CSRC.glob("*.h"))
    for p in sorted(srcs):
        h.update(p.name.encode())
        h.update(p.read_bytes())
    return h.hexdigest()[:16]


def embedded_hash(so: Path) -> str | None:
    """Read the RLSHASH marker from the binary without dlopen (dlopen
    caches handles per path, so a load-check-rebuild-reload cycle would
    see the stale mapping)."""
    data = so.read_bytes()
    i = data.find(b"RLSHASH:")
    if i < 0:
        return None
    return data[i + 8:i + 24].decode(errors="replace")


def _needs_build(so: Path, srcs: list[Path], name: str) -> bool:
    if not so.exists():
        return True
    # content-addressed: rebuild IFF the embedded source hash disagrees
    # (mtimes are unreliable across clones/snapshots)
    return embedded_hash(so) != source_hash(name)


def build_lib(name: str, verbose: bool = False) -> Path:
    srcs = [CSRC / s for s in _LIBS[name]]
    missing = [s for s in srcs if not s.exists()]
    if missing:
        raise FileNotFoundError(f"missing sources for {name}: {missing}")
    so = OUT / name
    if not _needs_build(so, srcs, name):
        return so
    h = source_hash(name)
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-fPIC", "-shared", f'-DRLS_SOURCE_HASH="{h}"', "-o", str(so),
    ] + [str(s) for s in srcs]
    if verbose:
        print("+", " ".join(cmd))
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"hipcc failed for {name}:\n{r.stdout}\n{r.stderr}")
    got = embedded_hash(so)
    if got != h:
        raise RuntimeError(f"{name}: embedded hash {got!r} != expected {h!r} "
                           "after build")
    return so


def verify_loaded(lib, name: str) -> None:
    """Assert a dlopen'ed lib carries the hash of the current sources.
    Called by every ops loader right after CDLL."""
    import ctypes
    try:
        fn = lib.rlsched_source_hash
    except AttributeError:
        raise RuntimeError(
            f"{name} has no rlsched_source_hash export — stale pre-round-2 "
            "binary; delete it and rebuild (python -m "
            "rlgpuschedule_amd.ops.build)") from None
    fn.restype = ctypes.c_char_p
    got = fn().decode(errors="replace").split("RLSHASH:")[-1]
    want = source_hash(name)
    if got != want:
        raise RuntimeError(
            f"{name} is stale: embedded source hash {got} != computed "
            f"{want}; the mapped binary does not match csrc/ — rebuild")


def build_all(verbose: bool = False) -> list[Path]:
    out = []
    for name, srcs in _LIBS.items():
        if all((CSRC / s).exists() for s in srcs):
            out.append(build_lib(name, verbose))
    return out


if __name__ == "__main__":
    for p in build_all(verbose=True):
        print("built", p)
