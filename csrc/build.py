"""Build the CDNA4 kernel library in-tree.

hipcc cross-compiles gfx950 without a GPU, so this runs in CPU-only CI.
The resulting ``_libllmops.so`` lives inside the package (travels with
the repo snapshot to GPU boxes; it is git-ignored so history stays
source-only).
"""

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRCS = [
    os.path.join(ROOT, "csrc", "llm_ops.hip"),
    os.path.join(ROOT, "csrc", "xgmi_comm.hip"),
]
DEPS = [os.path.join(ROOT, "csrc", "common.h")]
OUT = os.path.join(ROOT, "llm_np_cp_amd", "_libllmops.so")


def build(verbose: bool = True) -> str:
    cmd = [
        "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
        "-fPIC", "-shared", *SRCS, "-o", OUT,
    ]
    if verbose:
        print("+", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


def needs_rebuild() -> bool:
    if not os.path.exists(OUT):
        return True
    out_m = os.path.getmtime(OUT)
    return any(os.path.getmtime(s) > out_m for s in SRCS + DEPS)


def ensure_built(verbose: bool = False) -> str:
    if needs_rebuild():
        return build(verbose=verbose)
    return OUT


if __name__ == "__main__":
    build()
    print("built", OUT)
