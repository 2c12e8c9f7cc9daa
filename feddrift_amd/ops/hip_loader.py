"""Load the pre-built in-tree HIP extension (no JIT on the GPU box).

The .so must have been produced by `python -m feddrift_amd.ops.build`
(which the driver's build() hook runs on the CPU box; the snapshot carries
it to the GPU box). We intentionally do NOT fall back to a JIT rebuild here:
a missing .so on a GPU box should fail loudly, not silently rebuild or
revert to eager."""

from __future__ import annotations

import importlib.util
import os

from .build import BUILD_DIR, MODULE_NAME

_mod = None


def load():
    global _mod
    if _mod is not None:
        return _mod
    # FEDDRIFT_HIP_SO: explicit .so override (the sanitizer lane points
    # this at the ASAN build of the same extension)
    so = os.environ.get("FEDDRIFT_HIP_SO") or \
        os.path.join(BUILD_DIR, MODULE_NAME + ".so")
    if not os.path.exists(so):
        raise FileNotFoundError(
            f"{so} not found — run `python -m feddrift_amd.ops.build` first")
    # module name must match the .so's PyInit_<name> symbol
    mod_name = os.path.splitext(os.path.basename(so))[0]
    spec = importlib.util.spec_from_file_location(mod_name, so)
    _mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(_mod)
    return _mod
