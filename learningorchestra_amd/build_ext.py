"""In-tree build of the gfx950 HIP kernel extension.

Builds ``learningorchestra_amd/csrc/*`` into ``learningorchestra_amd/_build/
_lo_C.so`` with hipcc targeting gfx950 (cross-compiles fine on a GPU-less
host). The built .so is git-ignored but ships to the GPU box with the repo
snapshot, where ``ops/_ext.py`` loads it directly (no JIT cache dependency).

Run directly (``python -m learningorchestra_amd.build_ext``) or via
``__graft_entry__.build()``.
"""
from __future__ import annotations

import os
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")
BUILD_DIR = os.path.join(PKG_DIR, "_build")
EXT_NAME = "_lo_C"

SOURCES = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "gemm.hip"),
    os.path.join(CSRC, "gemm_8phase.hip"),
    os.path.join(CSRC, "conv_pool.hip"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "softmax_ce.hip"),
    os.path.join(CSRC, "tree_hist.hip"),
    os.path.join(CSRC, "embedding.hip"),
    os.path.join(CSRC, "batchnorm.hip"),
]


def build(verbose: bool = False):
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load
    mod = load(
        name=EXT_NAME,
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=True,
    )
    # stale-cache guard (bit us once: a ninja failure can leave an older
    # cached .so importable while the sources have moved on): the linked
    # .so must be at least as new as every source it claims to contain
    so = os.path.join(BUILD_DIR, EXT_NAME + ".so")
    if os.path.exists(so):
        so_mt = os.path.getmtime(so)
        stale = [os.path.basename(s) for s in SOURCES
                 if os.path.getmtime(s) > so_mt + 1.0]
        if stale:
            raise RuntimeError(
                f"stale kernel build: {so} predates modified sources "
                f"{stale}; remove {BUILD_DIR} and rebuild")
    return mod


if __name__ == "__main__":
    build(verbose="-v" in sys.argv)
    print(f"built {os.path.join(BUILD_DIR, EXT_NAME)}.so")
