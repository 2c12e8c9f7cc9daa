"""Optimizer name -> class repository.

Counterpart of the reference's fedml_api/standalone/fedopt/optrepo.py (the
one component the reference actually unit-tests,
tests/fedml_api/standalone/fedavg/test_optrepo.py): case-insensitive lookup
of torch.optim classes by name, with helpful errors and a listing helper.
Used by the FedOpt server optimizer (engine/server_opt.py).
"""

from __future__ import annotations

from typing import Dict, List, Type

import torch.optim as optim


class OptRepo:
    repo: Dict[str, Type[optim.Optimizer]] = {
        name.lower(): cls
        for name, cls in vars(optim).items()
        if isinstance(cls, type) and issubclass(cls, optim.Optimizer)
        and cls is not optim.Optimizer
    }

    @classmethod
    def name2cls(cls, name: str) -> Type[optim.Optimizer]:
        try:
            return cls.repo[name.lower()]
        except KeyError:
            raise KeyError(
                f"unknown optimizer '{name}'; supported: "
                f"{sorted(cls.repo)}") from None

    @classmethod
    def supported_parameters(cls, name: str) -> List[str]:
        import inspect
        sig = inspect.signature(cls.name2cls(name).__init__)
        return [p for p in sig.parameters if p not in ("self", "params")]
