"""Build the HIP kernel library for gfx950 (in-tree, travels with the
repo snapshot).  Invoked by __graft_entry__.build() and on demand."""
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(HERE, "cpp")
OUT = os.path.join(HERE, "libgbt_hip.so")

SOURCES = ["hist.hip", "partition.hip", "evaluate.hip", "compress.hip",
           "predict.hip", "shap.hip", "shap_paths.hip", "shap_ix.hip",
           "driver.hip", "csr.hip", "gpair.hip", "mt_evaluate.hip",
           "cpu_hist.cpp"]


def build(force: bool = False) -> str:
    sources = [os.path.join(SRC, s) for s in SOURCES
               if os.path.exists(os.path.join(SRC, s))]
    if not force and os.path.exists(OUT):
        newest = max(os.path.getmtime(s) for s in sources +
                     [os.path.join(SRC, "gbt_kernels.h")])
        if os.path.getmtime(OUT) >= newest:
            return OUT
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17",
           "-ffp-contract=off",  # bit-match the numpy fp64 oracle
           "-fopenmp",           # native CPU hist/partition (cpu_hist.cpp)
           "-shared", "-fPIC", "-o", OUT] + sources
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
