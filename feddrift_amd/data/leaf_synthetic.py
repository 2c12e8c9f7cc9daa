"""LEAF-style SYNTHETIC(alpha, beta) federated dataset generator.

Counterpart of the reference data/synthetic_{a}_{b}/generate_synthetic.py
(LEAF's classic non-IID generator): each client k draws a logistic model
W_k ~ N(u_k, 1), b_k ~ N(u_k, 1) with u_k ~ N(0, alpha), and features
x ~ N(v_k, Sigma) with client mean v_k ~ N(B_k, 1), B_k ~ N(0, beta);
Sigma = diag(j^-1.2). alpha controls model heterogeneity, beta feature
heterogeneity. Sample counts follow a log-normal power law.
"""

from __future__ import annotations

from typing import Dict, Tuple

import numpy as np


def generate_synthetic(alpha: float, beta: float, n_clients: int = 30,
                       dim: int = 60, n_classes: int = 10,
                       seed: int = 0, min_samples: int = 5,
                       mean_samples: float = 4.0
                       ) -> Dict[int, Tuple[np.ndarray, np.ndarray]]:
    rng = np.random.default_rng(seed)
    samples = (rng.lognormal(mean_samples, 2.0, n_clients).astype(int)
               + min_samples)
    sigma = np.diag(np.power(np.arange(1, dim + 1, dtype=np.float64), -1.2))
    out = {}
    for k in range(n_clients):
        u_k = rng.normal(0, alpha)
        b_big = rng.normal(0, beta)
        v_k = rng.normal(b_big, 1.0, dim)
        W = rng.normal(u_k, 1.0, (dim, n_classes))
        b = rng.normal(u_k, 1.0, n_classes)
        x = rng.multivariate_normal(v_k, sigma, samples[k])
        logits = x @ W + b
        probs = np.exp(logits - logits.max(axis=1, keepdims=True))
        probs /= probs.sum(axis=1, keepdims=True)
        y = np.array([rng.choice(n_classes, p=p) for p in probs])
        out[k] = (x.astype(np.float32), y.astype(np.int64))
    return out
