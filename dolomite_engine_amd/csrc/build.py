"""Build libdolomite_hip.so in-tree for gfx950.

Usage: python -m dolomite_engine_amd.csrc.build
Called by __graft_entry__.build(). hipcc cross-compiles without a GPU.
The .so lands next to the package (dolomite_engine_amd/libdolomite_hip.so)
so it travels with the repo snapshot to the GPU box.
"""

import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).resolve().parent
PKG = CSRC.parent
REPO = PKG.parent
SO_PATH = PKG / "libdolomite_hip.so"

SOURCES = [CSRC / "elementwise.hip", CSRC / "attention.hip", CSRC / "moe_gemm.hip", CSRC / "data_index.cpp"]


def _needs_rebuild() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    deps = SOURCES + [CSRC / "common.h", REPO / "include" / "dolomite_hip.h"]
    return any(p.stat().st_mtime > so_mtime for p in deps)


def build(force: bool = False) -> Path:
    if not force and not _needs_rebuild():
        return SO_PATH
    cmd = [
        "hipcc",
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        *[str(s) for s in SOURCES],
        "-o",
        str(SO_PATH),
    ]
    print("+", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
