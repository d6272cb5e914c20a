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


def needs_build() -> bool:
    if not LIB_PATH.exists():
        return True
    lib_mtime = LIB_PATH.stat().st_mtime
    deps = sources() + list(CSRC.glob("*.h"))
    return any(s.stat().st_mtime > lib_mtime for s in deps)


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
    return LIB_PATH


if __name__ == "__main__":
    build(force="--force" in os.sys.argv)
