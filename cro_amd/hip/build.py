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
AGENT_DIR = os.path.join(os.path.dirname(HIP_DIR), "agent")
SRC = os.path.join(HIP_DIR, "probe.hip")
OUT = os.path.join(HIP_DIR, "libcroprobe.so")
AGENT_SRC = os.path.join(AGENT_DIR, "croagent.cpp")
AGENT_OUT = os.path.join(AGENT_DIR, "croagent")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def build(force: bool = False) -> str:
    if (
        force
        or not os.path.exists(OUT)
        or os.path.getmtime(OUT) < os.path.getmtime(SRC)
    ):
        subprocess.run(
            [HIPCC, "--offload-arch=gfx950", "-O3", "-fPIC", "-shared", SRC, "-o", OUT],
            check=True,
        )
    build_agent(force=force)
    return OUT


def build_agent(force: bool = False) -> str:
    """croagent: the native node-agent CLI (links libcroprobe)."""
    if (
        not force
        and os.path.exists(AGENT_OUT)
        and os.path.getmtime(AGENT_OUT) >= os.path.getmtime(AGENT_SRC)
    ):
        return AGENT_OUT
    subprocess.run(
        [
            HIPCC,
            "--offload-arch=gfx950",
            "-O2",
            AGENT_SRC,
            f"-L{HIP_DIR}",
            "-lcroprobe",
            f"-Wl,-rpath,{HIP_DIR}",
            "-Wl,-rpath,$ORIGIN/../hip",
            "-o",
            AGENT_OUT,
        ],
        check=True,
    )
    return AGENT_OUT


if __name__ == "__main__":
    path = build(force="--force" in sys.argv)
    print(path)
    print(AGENT_OUT)
