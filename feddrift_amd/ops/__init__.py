"""Compute ops: batched local training / evaluation for the MLP family.

Two implementations with identical semantics:
  * ops.mlp_torch — vectorized pure-torch (device-agnostic). The numerics
    reference, and the CPU path for tests.
  * ops.hip (feddrift_hip.so (ops/hip/_build)) — hand-written CDNA4 HIP kernels (gfx950):
    the whole local-training phase of an FL round in one launch, the
    model x client accuracy sweep in one launch. Used on ROCm GPUs.

Dispatch: on a CUDA/HIP device the HIP extension is REQUIRED — we fail
loudly rather than fall back to eager so a missing .so cannot silently
produce non-native numbers (see repo policy on native-code loading).
"""

from __future__ import annotations

import os

import torch

from . import mlp_torch

_hip_mod = None
_hip_error = None


def _load_hip():
    global _hip_mod, _hip_error
    if _hip_mod is not None or _hip_error is not None:
        return _hip_mod
    try:
        from . import hip_loader
        _hip_mod = hip_loader.load()
    except Exception as e:  # noqa: BLE001
        _hip_error = e
        _hip_mod = None
    return _hip_mod


def hip_available() -> bool:
    return _load_hip() is not None


def backend_for(device: torch.device, use_hip: str = "auto"):
    """Return the op module for this device ('auto'|'always'|'never')."""
    if device.type == "cuda":
        if use_hip == "never":
            return mlp_torch
        mod = _load_hip()
        if mod is None:
            raise RuntimeError(
                "feddrift HIP extension not available on a GPU device: "
                f"{_hip_error!r}. Build it with python -m feddrift_amd.ops.build "
                "(or set use_hip_kernels=never to force the eager path "
                "explicitly).")
        from . import mlp_hip
        return mlp_hip
    return mlp_torch
