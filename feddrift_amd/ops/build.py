"""Build the HIP extension in-tree for gfx950.

Usage: python -m feddrift_amd.ops.build
The .so lands in feddrift_amd/ops/hip/_build/ (git-ignored, but it travels
to the GPU box with the gpurun snapshot). hipcc cross-compiles without a
GPU present.
"""

from __future__ import annotations

import os

_DIR = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(_DIR, "hip", "feddrift_kernels.hip")
SRC_CNN = os.path.join(_DIR, "hip", "cnn_kernels.hip")
BUILD_DIR = os.path.join(_DIR, "hip", "_build")
MODULE_NAME = "feddrift_hip"


def build(verbose: bool = False):
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load
    mod = load(
        name=MODULE_NAME,
        sources=[SRC, SRC_CNN],
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        verbose=verbose,
        is_python_module=True,
        with_cuda=True,
    )
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print(f"built {MODULE_NAME} into {BUILD_DIR}")
