#!/usr/bin/env python3
"""Build libveomni_hip.so in-tree (gfx950 only).

Run: python veomni_amd/csrc/build.py  (or via __graft_entry__.build()).
The .so lands at veomni_amd/libveomni_hip.so — git-ignored but shipped to the
GPU box by gpurun (built artefacts travel with the snapshot).
"""

import os
import subprocess
import sys

CSRC = os.path.dirname(os.path.abspath(__file__))
PKG = os.path.dirname(CSRC)
OUT = os.path.join(PKG, "libveomni_hip.so")

SOURCES = [
    "vh_abi.cpp",
    "vh_moe_ops.hip",
    "vh_group_gemm.hip",
    "vh_group_gemm8.hip",
    "vh_norms.hip",
    "vh_attention.hip",
    "vh_ce.hip",
    "vh_adamw.hip",
]


def build(verbose: bool = True) -> str:
    srcs = [os.path.join(CSRC, s) for s in SOURCES]
    newest_src = max(os.path.getmtime(s) for s in srcs + [os.path.join(CSRC, "vh_common.h")])
    if os.path.exists(OUT) and os.path.getmtime(OUT) > newest_src:
        if verbose:
            print(f"[vh build] up to date: {OUT}")
        return OUT
    cmd = [
        "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-shared", "-x", "hip",
        *srcs,
        "-o", OUT,
    ]
    if verbose:
        print("[vh build]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build()
    sys.exit(0)
