"""Generic flat packing for arbitrary nn.Modules (CNN / ResNet path).

The engine represents every model — global ensemble rows, per-(worker,
model) replicas, merge/reinit/copy operations and the fused aggregation —
as flat fp32 vectors. For the MLP family a fixed layout (models/packed.py)
feeds the HIP kernels; for convolutional models this packer flattens the
FULL state_dict (parameters AND buffers — the reference's aggregation
averages every state_dict entry including BN running stats,
FedAvgEnsAggregatorSoftCluster.py:174-185), while optimizer updates apply
only to trainable parameters.
"""

from __future__ import annotations

from typing import Dict, List

import torch
from torch import nn


class ModulePacker:
    def __init__(self, template: nn.Module):
        sd = template.state_dict()
        self.keys: List[str] = list(sd.keys())
        self.shapes = {k: sd[k].shape for k in self.keys}
        self.dtypes = {k: sd[k].dtype for k in self.keys}
        self.numels = [sd[k].numel() for k in self.keys]
        self.n_params = int(sum(self.numels))
        # entries that the optimizer trains (parameters, in state_dict order)
        param_keys = {n for n, _ in template.named_parameters()}
        self.param_keys = [k for k in self.keys if k in param_keys]
        self.n_train_params = int(sum(sd[k].numel() for k in self.param_keys))

    def flatten(self, sd: Dict[str, torch.Tensor]) -> torch.Tensor:
        return torch.cat([sd[k].detach().reshape(-1).float()
                          for k in self.keys])

    def unflatten(self, flat: torch.Tensor) -> Dict[str, torch.Tensor]:
        out = {}
        i = 0
        f = flat.detach()
        for k, n in zip(self.keys, self.numels):
            out[k] = f[i:i + n].reshape(self.shapes[k]).to(self.dtypes[k]) \
                .cpu().clone()
            i += n
        return out

    @torch.no_grad()
    def load_into(self, module: nn.Module, flat: torch.Tensor) -> None:
        i = 0
        sd = module.state_dict()
        for k, n in zip(self.keys, self.numels):
            sd[k].copy_(flat[i:i + n].reshape(self.shapes[k]))
            i += n

    @torch.no_grad()
    def dump_from(self, module: nn.Module, out_flat: torch.Tensor) -> None:
        i = 0
        sd = module.state_dict()
        for k, n in zip(self.keys, self.numels):
            out_flat[i:i + n].copy_(sd[k].reshape(-1).float())
            i += n
