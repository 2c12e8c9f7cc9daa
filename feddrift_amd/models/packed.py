"""Flat ("packed") parameter layout for the batched MLP compute path.

The engine keeps the K ensemble models and every per-(client, model) replica
as rows of flat fp32 tensors resident in HBM, so local training, the
model x client accuracy sweep and the weighted aggregation are single fused
kernels / collectives instead of the reference's per-model state_dict
shuffling (FedAvgEnsTrainer.py:47-95, FedAvgEnsAggregatorSoftCluster.py:148-195).

Layout (row-major, matching torch's nn.Linear storage):
  fnn: [fc1.weight (H*D) | fc1.bias (H) | fc2.weight (O*H) | fc2.bias (O)]
  lr:  [linear.weight (O*D) | linear.bias (O)]
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict

import torch


@dataclass(frozen=True)
class MLPSpec:
    kind: str          # 'fnn' | 'lr'
    d: int             # input features
    h: int             # hidden units (0 for lr)
    o: int             # output classes

    @property
    def n_params(self) -> int:
        if self.kind == "fnn":
            return self.h * self.d + self.h + self.o * self.h + self.o
        return self.o * self.d + self.o

    # slice offsets into the flat vector
    @property
    def off_w1(self) -> int:
        return 0

    @property
    def off_b1(self) -> int:
        return self.h * self.d

    @property
    def off_w2(self) -> int:
        return self.h * self.d + self.h

    @property
    def off_b2(self) -> int:
        return self.h * self.d + self.h + self.o * self.h


def spec_for(model_name: str, feature_num: int, class_num: int) -> MLPSpec:
    if model_name == "fnn":
        return MLPSpec("fnn", feature_num, feature_num * 2, class_num)
    if model_name == "lr":
        return MLPSpec("lr", feature_num, 0, class_num)
    raise NameError(model_name)


class PackedMLP:
    """Conversions between torch state_dicts and flat rows."""

    def __init__(self, spec: MLPSpec):
        self.spec = spec

    def flatten(self, sd: Dict[str, torch.Tensor]) -> torch.Tensor:
        s = self.spec
        if s.kind == "fnn":
            parts = [sd["fc1.weight"], sd["fc1.bias"],
                     sd["fc2.weight"], sd["fc2.bias"]]
        else:
            parts = [sd["linear.weight"], sd["linear.bias"]]
        return torch.cat([p.reshape(-1).float() for p in parts])

    def unflatten(self, flat: torch.Tensor) -> Dict[str, torch.Tensor]:
        s = self.spec
        f = flat.detach().cpu()
        if s.kind == "fnn":
            return {
                "fc1.weight": f[s.off_w1:s.off_b1].reshape(s.h, s.d).clone(),
                "fc1.bias": f[s.off_b1:s.off_w2].reshape(s.h).clone(),
                "fc2.weight": f[s.off_w2:s.off_b2].reshape(s.o, s.h).clone(),
                "fc2.bias": f[s.off_b2:].reshape(s.o).clone(),
            }
        return {
            "linear.weight": f[: s.o * s.d].reshape(s.o, s.d).clone(),
            "linear.bias": f[s.o * s.d:].reshape(s.o).clone(),
        }

    def views(self, flat: torch.Tensor):
        """Views of a batched flat tensor [..., P] as weight matrices."""
        s = self.spec
        lead = flat.shape[:-1]
        if s.kind == "fnn":
            w1 = flat[..., s.off_w1:s.off_b1].reshape(*lead, s.h, s.d)
            b1 = flat[..., s.off_b1:s.off_w2].reshape(*lead, s.h)
            w2 = flat[..., s.off_w2:s.off_b2].reshape(*lead, s.o, s.h)
            b2 = flat[..., s.off_b2:].reshape(*lead, s.o)
            return w1, b1, w2, b2
        w = flat[..., : s.o * s.d].reshape(*lead, s.o, s.d)
        b = flat[..., s.o * s.d:].reshape(*lead, s.o)
        return w, b
