"""Build the dppo_amd HIP extension in-tree for gfx950.

Usage: python -m dppo_amd.ops.build [--force]

Compiles every .hip under dppo_amd/ops/hip/ plus the torch binding into a
single extension dppo_amd/ops/_dppo_hip.so using torch.utils.cpp_extension
(which drives hipcc with --offload-arch=gfx950 under PYTORCH_ROCM_ARCH).
Cross-compiles fine on a GPU-less host.
"""

from __future__ import annotations

import glob
import os
import shutil
import sys

_HERE = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(_HERE, "hip")
EXT_NAME = "_dppo_hip"


def build(force: bool = False, verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    import torch
    from torch.utils.cpp_extension import load as jit_load

    sources = sorted(glob.glob(os.path.join(HIP_DIR, "*.hip"))) + sorted(
        glob.glob(os.path.join(HIP_DIR, "*.cpp"))
    )
    if not sources:
        raise RuntimeError(f"no HIP sources under {HIP_DIR}")

    build_dir = os.path.join(_HERE, "_build")
    os.makedirs(build_dir, exist_ok=True)
    if force:
        shutil.rmtree(build_dir, ignore_errors=True)
        os.makedirs(build_dir, exist_ok=True)

    mod = jit_load(
        name=EXT_NAME,
        sources=sources,
        build_directory=build_dir,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=True,
    )
    # copy the built .so in-tree so it ships with the repo snapshot
    built = glob.glob(os.path.join(build_dir, f"{EXT_NAME}*.so"))
    if not built:
        raise RuntimeError(f"build produced no .so under {build_dir}")
    dest = os.path.join(_HERE, os.path.basename(built[0]))
    shutil.copy2(built[0], dest)
    if verbose:
        print(f"[dppo_amd.ops.build] built {dest}")
    return dest


if __name__ == "__main__":
    build(force="--force" in sys.argv)
