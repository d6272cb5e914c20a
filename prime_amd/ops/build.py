"""In-tree build of the gfx950 HIP kernel library.

The kernels export a plain C API (no libtorch linkage), so the build is a
single `hipcc -shared` producing `prime_amd/ops/libprime_hip.so`. The .so is
git-ignored but ships to GPU boxes with the source snapshot; `build()` skips
recompilation when the .so is newer than every source file.
"""
from __future__ import annotations

import os
import subprocess
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
LIB_PATH = OPS_DIR / "libprime_hip.so"
ARCH = os.environ.get("PRIME_AMD_ARCH", "gfx950")


def sources() -> list[Path]:
    return sorted(CSRC.glob("*.hip"))


def source_hash() -> str:
    """Content hash of every source + the arch — mtimes are useless after
    a fresh checkout (everything gets the checkout time, so a stale
    prebuilt .so would win silently)."""
    import hashlib

    h = hashlib.sha256()
    h.update(ARCH.encode())
    for s in sources() + sorted(CSRC.glob("*.h")):
        h.update(s.name.encode())
        h.update(s.read_bytes())
    return h.hexdigest()


def needs_build() -> bool:
    if not LIB_PATH.exists():
        return True
    stamp = LIB_PATH.with_suffix(".so.hash")
    if not stamp.exists():
        return True
    return stamp.read_text().strip() != source_hash()


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_build():
        return LIB_PATH
    srcs = sources()
    if not srcs:
        raise RuntimeError(f"no .hip sources under {CSRC}")
    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-shared",
        "-fPIC",
        "-fvisibility=hidden",
        *[str(s) for s in srcs],
        "-o",
        str(LIB_PATH),
    ]
    if verbose:
        print(f"[prime_amd] building {LIB_PATH.name}: {' '.join(cmd)}", flush=True)
    subprocess.run(cmd, check=True, cwd=str(CSRC))
    LIB_PATH.with_suffix(".so.hash").write_text(source_hash() + "\n")
    return LIB_PATH


if __name__ == "__main__":
    build(force="--force" in os.sys.argv)
