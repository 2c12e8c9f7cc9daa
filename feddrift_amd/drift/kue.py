"""KUE state: per-model random feature masks + worst-model tracking.

Semantics per reference FedAvgEnsDataLoader.py:32-56 (KueState): each of the
K ensemble members sees a random subset of features (r ~ U[1, feature_num]
features chosen without replacement); the kappa-worst member's mask and
parameters are re-drawn each iteration (FedAvgEnsAggregatorKue.py:47-57).
"""

from __future__ import annotations

import numpy as np


class KueState:
    def __init__(self, model_num: int, feature_num: int,
                 rng: np.random.RandomState | None = None):
        self.model_num = model_num
        self.feature_num = feature_num
        self.worst_idx = 0
        self.masks = np.zeros((model_num, feature_num), dtype=bool)
        self._rng = rng or np.random.RandomState()
        for m in range(model_num):
            self.initialize_mask(m)

    def initialize_mask(self, model_idx: int) -> None:
        r = self._rng.randint(low=1, high=self.feature_num + 1)
        used = self._rng.choice(self.feature_num, size=r, replace=False)
        self.masks[model_idx, :] = False
        self.masks[model_idx, used] = True

    def set_worst_idx(self, model_idx: int) -> None:
        self.worst_idx = model_idx

    def get_worst_idx(self) -> int:
        return self.worst_idx

    def get_masks(self) -> np.ndarray:
        return self.masks
