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
    # torch's hipify SKIPS regenerating an existing passthrough copy even
    # when the source changed (observed: a stale elemwise_hip.hip kept an
    # old kernel set linked).  Drop any copy older than its source, and
    # its object, so ninja recompiles from fresh code.
    for s in sources:
        if not s.endswith(".hip"):
            continue
        h = s[:-4] + "_hip.hip"
        if os.path.exists(h) and os.path.getmtime(h) < os.path.getmtime(s):
            os.remove(h)
            obj = os.path.join(
                PKG_DIR,
                os.path.basename(h)[:-4] + ".cuda.o")
            if os.path.exists(obj):
                os.remove(obj)
    mod = load(
        name="_dlb_kernels",
        sources=sources,
        build_directory=PKG_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
    )
    # guard against a stale incremental link silently dropping a newly
    # added source (observed once with the ninja/hipify cache)
    for attr in ("conv_fwd", "gn_bwd", "lmloss_fwd", "se_fwd", "dropout",
                 "embed_fwd", "gn_dgb_reduce_multi", "attn_fwd"):
        if not hasattr(mod, attr):
            raise RuntimeError(
                f"_dlb_kernels is missing `{attr}` after build — stale "
                "incremental link; delete ops/*.o and build.ninja, rebuild")
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print("built _dlb_kernels")
