#!/usr/bin/env python3
"""Build libbydb_gpu.so in-tree for gfx950.

hipcc cross-compiles without a GPU; the built .so travels to the GPU box
with the repo snapshot.  Usage: python banyandb_amd/build.py
"""
import os
import subprocess
import sys

PKG = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG, "csrc")
OUT = os.path.join(PKG, "libbydb_gpu.so")


def build(verbose=True):
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    srcs = [os.path.join(CSRC, "kernels.hip"), os.path.join(CSRC, "encode.cpp"),
            os.path.join(CSRC, "frame.cpp"),
            os.path.join(CSRC, "part_io.cpp")]
    cmd = [
        hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-shared", "-o", OUT, *srcs, "-ldl",
    ]
    newest_src = max(os.path.getmtime(s) for s in srcs + [
        os.path.join(PKG, "..", "include", "bydb_gpu.h")])
    if os.path.exists(OUT) and os.path.getmtime(OUT) > newest_src:
        if verbose:
            print(f"up to date: {OUT}")
        return OUT
    if verbose:
        print(" ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build()
    sys.exit(0)
