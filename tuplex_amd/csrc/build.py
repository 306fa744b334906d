"""Build libtpx_gpu.so in-tree (the built .so travels with the repo snapshot)."""
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
PKG = os.path.dirname(HERE)
OUT = os.path.join(PKG, "libtpx_gpu.so")


def build(verbose=True):
    src = os.path.join(HERE, "tpx_abi.cpp")
    cmd = [
        "hipcc", "-O2", "-std=c++17", "-fPIC", "-shared", src,
        "-o", OUT, "-lhiprtc",
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)
    return OUT


if __name__ == "__main__":
    build()
    sys.exit(0)
