"""In-tree build of the gfx950 kernel extension.

Compiles every source under ops/csrc/ into ``_dlb_kernels`` with the .so
placed IN-TREE (this directory) so it travels with repo snapshots to GPU
boxes — a JIT cache under ~/.cache would not.  hipcc cross-compiles
gfx950 without a GPU present.
"""

from __future__ import annotations

import glob
import os

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")


def build(verbose: bool = False):
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load

    # Exclude "*_hip.hip": torch's extension builder writes hipify
    # passthrough copies of each .hip next to the original; globbing them
    # on a rebuild in a used tree would double-compile every kernel.
    sources = sorted(
        s for s in (glob.glob(os.path.join(CSRC, "*.cpp")) +
                    glob.glob(os.path.join(CSRC, "*.hip")))
        if not s.endswith("_hip.hip"))
    mod = load(
        name="_dlb_kernels",
        sources=sources,
        build_directory=PKG_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
    )
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print("built _dlb_kernels")
