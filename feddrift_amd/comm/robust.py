"""Robust aggregation defenses: update-norm clipping + Gaussian noise.

Counterpart of the reference fedml_core/distributed/robustness/
robust_aggregation.py (RobustAggregator.norm_diff_clipping :38,
add_noise :51) and the fedavg_robust package, re-expressed on flat
parameter vectors so it composes with the engine's fused aggregation:
each client update is clipped to `norm_bound` around the previous global
model BEFORE the weighted average, and stddev-sigma noise is added AFTER.
"""

from __future__ import annotations

from typing import Optional

import torch


def norm_diff_clipping(local_flat: torch.Tensor, global_flat: torch.Tensor,
                       norm_bound: float) -> torch.Tensor:
    """Clip (local - global) to an L2 ball of radius norm_bound."""
    diff = local_flat - global_flat
    norm = torch.linalg.vector_norm(diff, dim=-1, keepdim=True)
    scale = torch.clamp(norm_bound / (norm + 1e-12), max=1.0)
    return global_flat + diff * scale


def add_noise(flat: torch.Tensor, stddev: float,
              generator: Optional[torch.Generator] = None) -> torch.Tensor:
    noise = torch.randn(flat.shape, device=flat.device, generator=generator)
    return flat + stddev * noise


def robustify_replicas(replicas: torch.Tensor, global_params: torch.Tensor,
                       rows: torch.Tensor, model_of: torch.Tensor,
                       norm_bound: float) -> None:
    """In-place clipping of trained replica rows around their global model
    row (engine hook: called between local training and aggregation)."""
    if rows.numel() == 0:
        return
    local = replicas[rows]
    glob = global_params[model_of.long()]
    replicas[rows] = norm_diff_clipping(local, glob, norm_bound)
