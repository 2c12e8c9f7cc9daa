"""DriftSurf stable/reactive two-model state machine.

Semantics per reference FedAvgEnsDataLoader.py:146-266 (DriftSurfState).
The reference stores live torch models inside the pickled state and scores
them itself; here the state stores flat parameter vectors (numpy) and
scoring is supplied by the engine (a batched accuracy sweep), keeping the
decision logic identical:

  * stable: enter reactive when acc_pred < acc_best - delta or
    acc_pred < acc_stab - delta/2 (:222-229)
  * reactive (r=3 steps): track per-step accuracies; model_key follows the
    better of pred/reac (:235-249); on exit, reac replaces pred if its mean
    acc won (:250-260)
  * each model trains on a greedy window of <= wl iteration ids (:179-183)
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np


class DriftSurfState:
    def __init__(self, delta: float = 0.1, r: int = 3, wl: int = 10):
        self.reac_len = r
        self.delta = delta
        self.win_len = wl
        # flat parameter vectors (numpy) per key; None = uninitialized
        self.models: Dict[str, Optional[np.ndarray]] = {
            "pred": None, "stab": None, "reac": None}
        self.train_data_dict: Dict[str, Optional[List[int]]] = {
            "pred": [0], "stab": [0], "reac": None}
        self.train_keys = ["pred", "stab"]
        self.acc_best = 0.0
        self.acc_dict = None
        self.reac_ctr = None
        self.state = "stab"
        self.model_key = "pred"

    def _append_train_data(self, key: str, iter_id: int) -> None:
        self.train_data_dict[key].append(iter_id)
        if len(self.train_data_dict[key]) > self.win_len:
            self.train_data_dict[key].pop(0)

    def _reset(self, key: str) -> None:
        self.models[key] = None
        self.train_data_dict[key] = []

    def get_train_keys(self) -> List[str]:
        return self.train_keys

    def get_train_data(self, key: str) -> List[int]:
        return self.train_data_dict[key]

    def get_model_key(self) -> str:
        return self.model_key

    def set_model(self, key: str, flat: Optional[np.ndarray]) -> None:
        self.models[key] = flat

    def run_ds_algo(self, score_fn, curr_iter: int) -> None:
        """score_fn(key) -> accuracy of self.models[key] on the newest data
        batch (0 when the model is None / has no data)."""
        acc_pred = score_fn("pred")
        if acc_pred > self.acc_best:
            self.acc_best = acc_pred
        if self.state == "stab":
            if len(self.train_data_dict["stab"]) == 0:
                acc_stab = 0.0
            else:
                acc_stab = score_fn("stab")
            if (acc_pred < self.acc_best - self.delta) or \
               (acc_pred < acc_stab - self.delta / 2):
                self.state = "reac"
                self._reset("reac")
                self.reac_ctr = 0
                self.acc_dict = {"pred": np.zeros(self.reac_len),
                                 "reac": np.zeros(self.reac_len)}
            else:
                self._append_train_data("pred", curr_iter)
                self._append_train_data("stab", curr_iter)
                self.train_keys = ["pred", "stab"]
        if self.state == "reac":
            if self.reac_ctr > 0:
                acc_reac = score_fn("reac")
                self.acc_dict["pred"][self.reac_ctr - 1] = acc_pred
                self.acc_dict["reac"][self.reac_ctr - 1] = acc_reac
                self.model_key = "reac" if acc_reac > acc_pred else "pred"
            self._append_train_data("pred", curr_iter)
            self._append_train_data("reac", curr_iter)
            self.train_keys = ["pred", "reac"]
            self.reac_ctr += 1
            if self.reac_ctr == self.reac_len:
                self.state = "stab"
                self._reset("stab")
                if np.mean(self.acc_dict["pred"]) < np.mean(self.acc_dict["reac"]):
                    self.models["pred"] = self.models["reac"]
                    self.train_data_dict["pred"] = self.train_data_dict["reac"]
                    self.acc_best = float(np.amax(self.acc_dict["reac"]))
                    self.model_key = "pred"
                self.acc_dict = None
                self.reac_ctr = None
