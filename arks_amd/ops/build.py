"""In-tree build of the arks_amd HIP extension for gfx950.

The extension builds into arks_amd/ops/_build (IN the repo tree, so the .so
travels with gpurun snapshots) via torch.utils.cpp_extension, which drives
hipcc for the .hip sources. Cross-compiles fine on a GPU-less box.

Usage: python -m arks_amd.ops.build [-v]
"""

from __future__ import annotations

import glob
import os
import sys

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "csrc")
BUILD_DIR = os.path.join(ROOT, "_build")
EXT_NAME = "arks_amd_C"

SOURCES = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "kvcache.hip"),
    os.path.join(CSRC, "attn_decode.hip"),
    os.path.join(CSRC, "attn_prefill.hip"),
    os.path.join(CSRC, "attn_extend.hip"),
    os.path.join(CSRC, "attn_extend2.hip"),
    os.path.join(CSRC, "quant_fp8.hip"),
    os.path.join(CSRC, "skinny_gemm.hip"),
    os.path.join(CSRC, "sampling.hip"),
    os.path.join(CSRC, "moe.hip"),
    os.path.join(CSRC, "allreduce.hip"),
]


def built_so_path() -> str | None:
    matches = glob.glob(os.path.join(BUILD_DIR, f"{EXT_NAME}*.so"))
    return matches[0] if matches else None


def build(verbose: bool = False):
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils import cpp_extension

    cpp_extension.load(
        name=EXT_NAME,
        sources=SOURCES,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", "--offload-arch=gfx950"],
        build_directory=BUILD_DIR,
        verbose=verbose,
        is_python_module=False,  # we import the .so ourselves (_load.py)
        keep_intermediates=True,
    )
    so = built_so_path()
    assert so is not None, "build produced no .so"
    return so


if __name__ == "__main__":
    so = build(verbose="-v" in sys.argv)
    print(f"built: {so}")
