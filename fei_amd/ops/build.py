"""Build the gfx950 HIP kernel library in-tree.

The .so lives at fei_amd/ops/libfei_kernels.so so it travels with the repo
snapshot to GPU boxes (JIT caches under ~/.cache do not). hipcc
cross-compiles gfx950 without a GPU, so this runs on CPU-only machines too.
"""

from __future__ import annotations

import os
import subprocess
import sys

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
OUT = os.path.join(os.path.dirname(os.path.abspath(__file__)), "libfei_kernels.so")
SOURCES = ["fei_kernels.hip", "attn_prefill.hip", "gemv.hip", "attn_decode_fused.hip", "gemv_fp8.hip", "stream_layer.hip"]
HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("FEI_AMD_ARCH", "gfx950")


def needs_build() -> bool:
    if not os.path.exists(OUT):
        return True
    out_mtime = os.path.getmtime(OUT)
    for src in SOURCES + ["fei_common.h"]:
        if os.path.getmtime(os.path.join(CSRC, src)) > out_mtime:
            return True
    return False


def build(force: bool = False, verbose: bool = True) -> str:
    """Compile every HIP source into one shared library. Returns the path."""
    if not force and not needs_build():
        return OUT
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC",
        *[os.path.join(CSRC, s) for s in SOURCES],
        "-o", OUT,
    ]
    if verbose:
        print("[fei_amd.ops.build]", " ".join(cmd), file=sys.stderr)
    result = subprocess.run(cmd, capture_output=True, text=True)
    if result.returncode != 0:
        raise RuntimeError(
            f"hipcc failed (rc={result.returncode}):\n{result.stderr[-4000:]}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT)
