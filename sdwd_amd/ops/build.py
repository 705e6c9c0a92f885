"""Build the gfx950 HIP extension IN-TREE so the .so travels with the repo.

Usage: python -m sdwd_amd.ops.build   (also called by __graft_entry__.build).
Cross-compiles fine on a GPU-less box (hipcc --offload-arch=gfx950 via
torch.utils.cpp_extension under PYTORCH_ROCM_ARCH=gfx950).
"""
from __future__ import annotations

import glob
import os
import shutil
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(HERE, "hip", "ext.hip")
BUILD_DIR = os.path.join(HERE, "_build")
OUT_PREFIX = os.path.join(HERE, "_sdwd_hip")


def build(verbose: bool = False) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    # torch's hipify caches hip/ext_hip.hip keyed on ext.hip alone; edits to
    # the #included kernel files (attention.hip, conv.hip, ...) would ship a
    # STALE build silently. Nuke the artifact so hipify + ninja re-run.
    art = os.path.join(HERE, "hip", "ext_hip.hip")
    if os.path.exists(art):
        os.remove(art)
    from torch.utils.cpp_extension import load

    mod = load(
        name="_sdwd_hip",
        sources=[SRC],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        build_directory=BUILD_DIR,
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
        keep_intermediates=True,
    )
    # copy the built module next to ops/ for the in-tree loader
    built = sorted(glob.glob(os.path.join(BUILD_DIR, "_sdwd_hip*.so")))
    if not built:
        raise RuntimeError(f"build produced no .so in {BUILD_DIR}")
    dest = OUT_PREFIX + ".so"
    shutil.copy2(built[0], dest)
    return dest


if __name__ == "__main__":
    path = build(verbose="-v" in sys.argv)
    print(f"built {path}")
