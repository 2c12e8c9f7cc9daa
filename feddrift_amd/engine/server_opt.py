"""FedOpt-style server optimizer.

Counterpart of the reference fedml_api/standalone/fedopt (server-side
optimizer over the pseudo-gradient): instead of replacing the global model
with the weighted client average, treat (average - global) as a gradient
and step a torch optimizer (looked up via utils.OptRepo, e.g. sgd with
momentum, adam, adagrad) on the flat [K, P] model bank.
"""

from __future__ import annotations

import torch

from ..utils.optrepo import OptRepo


class ServerOptimizer:
    def __init__(self, global_params: torch.Tensor, name: str = "sgd",
                 lr: float = 1.0, **kwargs):
        self.param = torch.nn.Parameter(global_params.clone())
        cls = OptRepo.name2cls(name)
        self.opt = cls([self.param], lr=lr, **kwargs)

    @torch.no_grad()
    def step(self, global_params: torch.Tensor, averaged: torch.Tensor,
             updated_mask: torch.Tensor) -> None:
        """global <- optimizer step toward `averaged` on updated rows.

        FedAvg is the special case sgd(lr=1). Rows not updated this round
        keep their parameters and accrue no optimizer state drift (their
        pseudo-gradient is zero)."""
        self.param.data.copy_(global_params)
        grad = torch.where(updated_mask.unsqueeze(1),
                           global_params - averaged,
                           torch.zeros_like(global_params))
        self.param.grad = grad
        self.opt.step()
        global_params.copy_(self.param.data)
