"""Build driver: compiles every native piece in-tree (the .so files travel
to the GPU box with the repo snapshot; nothing is JIT-cached outside).

  python -m serenedb_amd.build [--force]
"""

import os
import subprocess
import sys

PKG = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(PKG)

HOST_SRC = os.path.join(PKG, "csrc", "host", "sdb_host.cpp")
GPU_SRC = os.path.join(PKG, "csrc", "hip", "sdb_gpu.hip")
SCAN_SRC = os.path.join(PKG, "csrc", "hip", "sdb_scan.hip")
HOST_SO = os.path.join(PKG, "libsdb_host.so")
GPU_SO = os.path.join(PKG, "libsdb_gpu.so")

# fp32 BM25 bit-parity across gcc/hipcc/oracle requires no FMA contraction
HOST_CMD = ["g++", "-O2", "-std=c++17", "-ffp-contract=off", "-fPIC",
            "-shared", "-Wall", HOST_SRC, "-o", HOST_SO]


def _newer(src, out):
    return not os.path.exists(out) or os.path.getmtime(src) > os.path.getmtime(out)


def build(force=False, verbose=True):
    def run(cmd):
        if verbose:
            print("+", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)

    if force or _newer(HOST_SRC, HOST_SO):
        run(HOST_CMD)
    gpu_srcs = [GPU_SRC] + ([SCAN_SRC] if os.path.exists(SCAN_SRC) else [])
    if force or any(_newer(s, GPU_SO) for s in gpu_srcs):
        run(["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
             "-ffp-contract=off", "-fPIC", "-shared", "-Wno-unused-result"]
            + gpu_srcs + ["-o", GPU_SO])
    # oracle (test infrastructure) builds via its own Makefile
    run(["make", "-C", os.path.join(REPO, "oracle"), "-s"])


if __name__ == "__main__":
    build(force="--force" in sys.argv)
