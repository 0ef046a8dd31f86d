"""Build the gfx950 HIP probe extension in-tree.

Usage: ``python -m cro_amd.hip.build``  (also driven by __graft_entry__.build)

hipcc cross-compiles for gfx950 without a GPU present; the resulting .so
lives next to the source so it travels with repo snapshots.
"""

from __future__ import annotations

import os
import subprocess
import sys

HIP_DIR = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(HIP_DIR, "probe.hip")
OUT = os.path.join(HIP_DIR, "libcroprobe.so")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def build(force: bool = False) -> str:
    if (
        not force
        and os.path.exists(OUT)
        and os.path.getmtime(OUT) >= os.path.getmtime(SRC)
    ):
        return OUT
    cmd = [
        HIPCC,
        "--offload-arch=gfx950",
        "-O3",
        "-fPIC",
        "-shared",
        SRC,
        "-o",
        OUT,
    ]
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    path = build(force="--force" in sys.argv)
    print(path)
