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


ASAN_BUILD_DIR = os.path.join(_DIR, "hip", "_build_asan")
ASAN_MODULE_NAME = "feddrift_hip_asan"


def build_asan(verbose: bool = False) -> str:
    """Device-AddressSanitizer build (gfx950:xnack+): the sanitizer CI
    lane loads this variant (FEDDRIFT_HIP_SO=...) and re-runs the GPU
    numerics tests under HSA_XNACK=1 with the ASAN runtime preloaded —
    certifying the kernels' memory safety beyond what numerics tests can
    see (an OOB read that does not perturb results passes those).

    hipcc drives BOTH compile and link (torch cpp_extension links with
    gcc, which cannot link clang's ASAN runtime or device code)."""
    import subprocess
    import torch.utils.cpp_extension as ce
    import sysconfig
    os.makedirs(ASAN_BUILD_DIR, exist_ok=True)
    torch_lib = os.path.join(os.path.dirname(ce.__file__), "..", "lib")
    incs = ce.include_paths() + [sysconfig.get_paths()["include"],
                                 "/opt/rocm/include"]
    out = os.path.join(ASAN_BUILD_DIR, ASAN_MODULE_NAME + ".so")
    cmd = (["hipcc", "--offload-arch=gfx950:xnack+", "-O1", "-g",
            "-fsanitize=address", "-shared-libsan", "-shared", "-fPIC",
            "-std=c++17",
            f"-DTORCH_EXTENSION_NAME={ASAN_MODULE_NAME}",
            "-DWITH_HIP", "-DTORCH_API_INCLUDE_EXTENSION_H",
            "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
            "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
            "-D__HIP_NO_HALF_CONVERSIONS__=1",
            "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1"]
           + [f"-I{i}" for i in incs]
           + [SRC, SRC_CNN,
              f"-L{os.path.abspath(torch_lib)}", "-lc10", "-lc10_hip",
              "-ltorch_cpu", "-ltorch_hip", "-ltorch", "-ltorch_python",
              "-L/opt/rocm/lib", "-lamdhip64", "-o", out])
    subprocess.run(cmd, check=True,
                   capture_output=not verbose)
    return out


if __name__ == "__main__":
    import sys
    if "--asan" in sys.argv:
        build_asan(verbose=True)
        print(f"built {ASAN_MODULE_NAME} into {ASAN_BUILD_DIR}")
    else:
        build(verbose=True)
        print(f"built {MODULE_NAME} into {BUILD_DIR}")
