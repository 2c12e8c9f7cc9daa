"""Adaptive-FedAvg learning-rate controller.

Double-EMA variance-ratio estimator over the flattened averaged parameters;
eta = min(eta0, eta0 * gamma_hat / t). Semantics per reference
FedAvgEnsDataLoader.py:75-126 (AdaState).
"""

from __future__ import annotations

import copy

import numpy as np


class AdaState:
    def __init__(self, init_lr: float = 1e-2, beta1: float = 0.5,
                 beta2: float = 0.5, beta3: float = 0.5):
        self.init_lr = init_lr
        self.beta1 = beta1
        self.beta2 = beta2
        self.beta3 = beta3
        self.eta = init_lr
        self.mu = None
        self.s = 0.0
        self.gam = 0.0

    def update(self, theta: np.ndarray, t: int) -> None:
        t = t + 1  # count from 1
        prev_mu = self.mu if self.mu is not None else np.zeros(theta.shape)
        prev_s = self.s
        prev_gam = self.gam
        if t != 1:
            prev_muh = prev_mu / (1 - self.beta1 ** (t - 1))
            prev_sh = prev_s / (1 - self.beta2 ** (t - 1))
        else:
            prev_muh = 0
            prev_sh = 0
        new_mu = self.beta1 * prev_mu + (1 - self.beta1) * theta
        new_s = self.beta2 * prev_s + (1 - self.beta2) * np.mean(
            (theta - prev_muh) * (theta - prev_muh))
        new_sh = new_s / (1 - self.beta2 ** t)
        ratio = new_sh / prev_sh if prev_sh != 0 else 1
        new_gam = self.beta3 * prev_gam + (1 - self.beta3) * ratio
        new_gamh = new_gam / (1 - self.beta3 ** t)
        self.eta = min(self.init_lr, (self.init_lr * new_gamh) / t)
        self.mu = copy.deepcopy(new_mu)
        self.s = new_s
        self.gam = new_gam

    def current_lr(self) -> float:
        return self.eta
