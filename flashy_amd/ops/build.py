# Copyright (c) Flashy-AMD authors.
"""In-tree build of the gfx950 native extension.

``python -m flashy_amd.ops.build`` compiles csrc/*.hip + ext.cpp with hipcc
(--offload-arch=gfx950) into ``flashy_amd/ops/_hip_ops.<abi>.so``.  The .so
lives in the source tree (it is .gitignored but ships to the GPU box with the
gpurun snapshot).  hipcc cross-compiles without a GPU, so this runs on the
CPU-only build box too.
"""
from __future__ import annotations

import subprocess
import sys
import sysconfig
from pathlib import Path

HERE = Path(__file__).resolve().parent
CSRC = HERE / "csrc"
ARCH = "gfx950"


def so_path() -> Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return HERE / f"_hip_ops{suffix}"


def sources() -> list:
    return sorted(CSRC.glob("*.hip")) + sorted(CSRC.glob("*.cpp"))


def needs_build() -> bool:
    out = so_path()
    if not out.exists():
        return True
    newest = max(p.stat().st_mtime for p in sources() + [CSRC / "common.h"])
    return out.stat().st_mtime < newest


def build(verbose: bool = True, force: bool = False) -> Path:
    out = so_path()
    if not force and not needs_build():
        if verbose:
            print(f"[flashy_amd.ops.build] up to date: {out}")
        return out
    import pybind11
    py_include = sysconfig.get_paths()["include"]
    cmd = [
        "hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-fPIC", "-shared",
        "-DNDEBUG",
        f"-I{py_include}", f"-I{pybind11.get_include()}", f"-I{CSRC}",
        *[str(s) for s in sources()],
        "-o", str(out),
    ]
    if verbose:
        print("[flashy_amd.ops.build]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
