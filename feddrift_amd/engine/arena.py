"""Device arena: all client data for one training iteration, resident in HBM.

Every (client, iteration) segment and every retrain-view segment is uploaded
once at iteration start; batches are (offset, length) windows into two flat
tensors (X [N, D] fp32, Y [N] i64). No host<->device data movement happens
inside the round loop — the reference instead moves every batch and every
model CPU<->GPU every round (FedAvgEnsTrainer.py:51-87).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

import numpy as np
import torch

from ..data.loader import Segment


@dataclass
class SegRef:
    offset: int
    n: int
    windows: List[Tuple[int, int]]   # (global offset, length)


class DeviceArena:
    def __init__(self, feature_num: int, device: torch.device):
        self.d = feature_num
        self.device = device
        self._xs: List[np.ndarray] = []
        self._ys: List[np.ndarray] = []
        self._n = 0
        self.x: torch.Tensor | None = None
        self.y: torch.Tensor | None = None

    def add(self, seg: Segment) -> SegRef:
        assert self.x is None, "arena is frozen"
        off = self._n
        self._xs.append(seg.x.reshape(-1, self.d) if seg.n else
                        np.zeros((0, self.d), np.float32))
        self._ys.append(seg.y)
        self._n += seg.n
        return SegRef(off, seg.n, [(off + s, ln) for s, ln in seg.windows])

    def freeze(self) -> None:
        x = np.concatenate(self._xs, 0) if self._xs else \
            np.zeros((0, self.d), np.float32)
        y = np.concatenate(self._ys, 0) if self._ys else \
            np.zeros((0,), np.int64)
        self.x = torch.from_numpy(np.ascontiguousarray(x)).to(self.device)
        self.y = torch.from_numpy(np.ascontiguousarray(y)).to(self.device)
        self._xs = self._ys = None
