"""MultiModelAcc state (FedDrift-Eager precursor 'mmacc' + oracles
'mmgeni'/'mmgeniex').

Semantics per reference FedAvgEnsDataLoader.py:317-449 (MultiModelAccState):
per-client best-model selection with drift detection onto the next free
model slot; oracle variants read the change-point matrix directly.
Model scoring is supplied by the engine (score_fn(model_key, client) ->
accuracy on that client's newest data).
"""

from __future__ import annotations

import json
from typing import Dict, List, Optional

import numpy as np


class MultiModelAccState:
    def __init__(self, client_num: int, model_num: int = 2,
                 delta: float = 0.1):
        self.client_num = client_num
        self.model_num = model_num
        self.delta = delta
        self.train_data_dict: Dict[int, List[List[int]]] = {
            m: [[] for _ in range(client_num)] for m in range(model_num)}
        self.models: Dict[int, Optional[np.ndarray]] = {}
        self.train_model_idx: Dict[int, int] = {}
        self.test_model_idx: Dict[int, int] = {}
        self.acc_dict: Dict[int, float] = {}

    def run_model_select(self, score_fn, curr_iter: int) -> None:
        if curr_iter == 0:
            for c in range(self.client_num):
                self.train_data_dict[0][c].append(0)
                self.train_model_idx[c] = 0
                self.test_model_idx[c] = 0
            return
        next_free_model = -1
        for m in range(self.model_num):
            if m not in self.models:
                next_free_model = m
                break
        for c in range(self.client_num):
            model_acc = {m: score_fn(m, c) for m in self.models}
            # best_acc starts below any real accuracy: when EVERY model
            # scores exactly 0.0 on this client (possible early on hard
            # tasks) the reference's `best_acc = 0.0` start leaves
            # best_model at -1 and crashes on the dict access
            # (FedAvgEnsDataLoader.py:350-390); pick the first model
            # instead — identical behavior whenever any score is > 0.
            best_model, best_acc = -1, -1.0
            for m, a in model_acc.items():
                if a > best_acc:
                    best_acc, best_model = a, m
            if best_model == -1:       # no models registered yet
                best_model = 0
            if self.acc_dict[c] - best_acc > self.delta and \
                    next_free_model != -1:
                best_model = next_free_model
            self.train_data_dict[best_model][c].append(curr_iter)
            self.train_model_idx[c] = best_model
            self.test_model_idx[c] = best_model

    def model_select_geni(self, curr_iter: int, change_points: np.ndarray,
                          time_stretch: int) -> None:
        for c in range(self.client_num):
            best = int(change_points[curr_iter // time_stretch][c])
            self.train_data_dict[best][c].append(curr_iter)
            self.train_model_idx[c] = best
            self.test_model_idx[c] = best

    def model_select_geniex(self, curr_iter: int, change_points: np.ndarray,
                            time_stretch: int) -> None:
        min_cp = 1000000
        for t in range(change_points.shape[0]):
            if any(change_points[t]):
                min_cp = t * time_stretch
                break
        for c in range(self.client_num):
            train_model = int(change_points[curr_iter // time_stretch][c])
            if curr_iter >= min_cp:
                test_model = int(
                    change_points[(curr_iter + 1) // time_stretch][c])
            else:
                test_model = train_model
            self.train_data_dict[train_model][c].append(curr_iter)
            self.train_model_idx[c] = train_model
            self.test_model_idx[c] = test_model

    def set_model(self, key: int, flat: Optional[np.ndarray]) -> None:
        self.models[key] = flat

    def set_acc(self, client: int, acc: float) -> None:
        self.acc_dict[client] = acc

    def get_train_data_by_model(self, key: int) -> str:
        train_data = self.train_data_dict[key]
        if not any(len(dl) > 0 for dl in train_data):
            return ""
        return json.dumps(train_data)

    def get_test_model_idx(self, client_idx: int) -> int:
        return self.test_model_idx[client_idx]

    def get_train_model_idx(self, client_idx: int) -> int:
        return self.train_model_idx[client_idx]
