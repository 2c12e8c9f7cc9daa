"""Secure aggregation via pairwise additive masking.

Counterpart of the reference fedml_api/distributed/turboaggregate (MPC-
style secure aggregation): each client pair (i, j) derives a shared mask
from a common seed; client i adds +mask_ij, client j adds -mask_ij, so the
server-side SUM is exact while every individual upload is information-
theoretically masked. Masks are generated on-device; the engine's fused
aggregation then operates on masked uploads unchanged.
"""

from __future__ import annotations

from typing import Sequence

import torch


def pair_seed(base_seed: int, i: int, j: int) -> int:
    a, b = (i, j) if i < j else (j, i)
    return (base_seed * 1000003 + a * 7919 + b) & 0x7FFFFFFF


def mask_for(client: int, others: Sequence[int], n_params: int,
             base_seed: int, device, scale: float = 1.0) -> torch.Tensor:
    """Sum of signed pairwise masks for `client` against `others`."""
    total = torch.zeros(n_params, device=device)
    for other in others:
        if other == client:
            continue
        g = torch.Generator(device=device)
        g.manual_seed(pair_seed(base_seed, client, other))
        m = torch.randn(n_params, device=device, generator=g) * scale
        total += m if client < other else -m
    return total


def mask_uploads(uploads: torch.Tensor, clients: Sequence[int],
                 base_seed: int, scale: float = 1.0) -> torch.Tensor:
    """uploads [n, P] -> masked copies; sum over rows is preserved."""
    out = uploads.clone()
    for row, c in enumerate(clients):
        out[row] += mask_for(c, clients, uploads.shape[1], base_seed,
                             uploads.device, scale)
    return out
