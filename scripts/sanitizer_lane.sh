#!/bin/bash
# Sanitizer lane: run the GPU numerics tests against the device-ASAN
# build of the HIP kernels (gfx950:xnack+, -fsanitize=address) under
# HSA_XNACK=1. An out-of-bounds access that happens not to perturb the
# numerics would pass the parity tests silently; this lane faults it.
#
# Usage (on a GPU box): bash scripts/sanitizer_lane.sh [pytest args...]
# The ASAN .so must be pre-built: python -m feddrift_amd.ops.build --asan
set -u
cd "$(dirname "$0")/.."

ASAN_SO="feddrift_amd/ops/hip/_build_asan/feddrift_hip_asan.so"
if [ ! -f "$ASAN_SO" ]; then
  echo "sanitizer lane: $ASAN_SO missing - build with" \
       "python -m feddrift_amd.ops.build --asan" >&2
  exit 2
fi
RT=$(ls /opt/rocm/lib/llvm/lib/clang/*/lib/linux/libclang_rt.asan-x86_64.so \
     2>/dev/null | head -1)
if [ -z "$RT" ]; then
  echo "sanitizer lane: ASAN runtime not found under /opt/rocm" >&2
  exit 2
fi

export FEDDRIFT_HIP_SO="$PWD/$ASAN_SO"
export HSA_XNACK=1
export LD_PRELOAD="$RT"
# torch/python leak reports are noise here; halt_on_error keeps device
# ASAN reports fatal so the lane FAILS on the first OOB
export ASAN_OPTIONS="detect_leaks=0:halt_on_error=1:protect_shadow_gap=0"

args=("$@")
if [ ${#args[@]} -eq 0 ]; then
  args=(tests/test_gpu_ops.py tests/test_gpu_cnn.py -x -q)
fi
python -m pytest "${args[@]}"
rc=$?
echo "sanitizer lane exit: $rc (0 = all kernels clean under device ASAN)"
exit $rc
