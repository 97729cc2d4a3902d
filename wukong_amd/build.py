"""Build the native engine in-tree: hipcc --offload-arch=gfx950.

Usage: python -m wukong_amd.build [--force]
The .so is committed-adjacent (gitignored) and travels to the GPU box
with the gpurun snapshot.
"""
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = [os.path.join(HERE, "csrc", f)
       for f in ("lubm_gen.cpp", "watdiv_gen.cpp", "store.cpp", "gpu_engine.hip")]
OUT = os.path.join(HERE, "libwukong_hip.so")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def build(force=False, verbose=True):
    if not force and os.path.exists(OUT):
        newest = max(os.path.getmtime(s) for s in SRC + [
            os.path.join(HERE, "csrc", "wk_types.h"),
            os.path.join(HERE, "csrc", "wk_store.h")])
        if os.path.getmtime(OUT) > newest:
            if verbose:
                print(f"wukong_amd.build: {OUT} up to date")
            return OUT
    cmd = [HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
           "-fopenmp", "-shared", *SRC, "-o", OUT]
    if verbose:
        print("wukong_amd.build:", " ".join(cmd))
    subprocess.check_call(cmd)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
