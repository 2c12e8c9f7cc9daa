"""Drift-dataset generators (SEA / SINE / CIRCLE / MNIST-drift) + change points.

Semantics follow the reference generators (statistical parity, not byte parity):
  * SEA: fedml_api/data_preprocessing/sea/data_loader.py:37-82. The shipped
    concept CSVs (data/sea/concept{1..4}.csv) were measured to be
    x ~ U[0,10]^3, label = [f2+f3 > theta] with a 10% label flip, and
    theta = (8, 9, 7, 9) — concepts 2 and 4 are statistically identical in
    the reference's own data files; we reproduce that rule rather than the
    textbook SEA (f1+f2, theta4=9.5).
  * SINE: sine/data_loader.py:37-47 — y = [x2 <= sin x1] or inverted.
  * CIRCLE: circle/data_loader.py:37-44 — inside/outside one of two circles.
  * MNIST-drift: MNIST/data_loader_cont.py:151-214 — concept k in {1,2,3}
    swaps label pairs (1,2),(3,4),(5,6). Real MNIST pixels are unavailable
    offline; we synthesize class-conditional Gaussian images of the same
    shape (784 features in [0,1], 10 classes) so the drift structure and
    tensor shapes match.
  * change-point matrix: (T+1) x num_clients ints = concept per (time, client)
    (data/changepoints/*.cp; 'rand' drawn as in sea/data_loader.py:49-64).

Data is written per (client, iteration) as CSV files named
client_{c}_iter_{t}.csv, interoperable with the reference layout.
"""

from __future__ import annotations

import os

import numpy as np

SEA_THETAS = (8.0, 9.0, 7.0, 9.0)
SEA_NOISE = 0.1


def sample_sea(n: int, concept: int, rng: np.random.Generator) -> np.ndarray:
    x = rng.uniform(0.0, 10.0, size=(n, 3))
    y = (x[:, 1] + x[:, 2] > SEA_THETAS[concept]).astype(np.float64)
    flip = rng.random(n) < SEA_NOISE
    y[flip] = 1.0 - y[flip]
    return np.concatenate([x, y[:, None]], axis=1)


def sample_sine(n: int, concept: int, rng: np.random.Generator) -> np.ndarray:
    x = rng.random((n, 2))
    below = x[:, 1] <= np.sin(x[:, 0])
    y = below.astype(np.float64) if concept == 0 else (~below).astype(np.float64)
    return np.concatenate([x, y[:, None]], axis=1)


def sample_circle(n: int, concept: int, rng: np.random.Generator) -> np.ndarray:
    x = rng.random((n, 2))
    cx, cy, r = (0.2, 0.5, 0.15) if concept == 0 else (0.6, 0.5, 0.25)
    z = (x[:, 0] - cx) ** 2 + (x[:, 1] - cy) ** 2 - r * r
    y = (z > 0).astype(np.float64)
    return np.concatenate([x, y[:, None]], axis=1)


# synthetic stand-in for MNIST pixels: per-class mean images drawn once from a
# fixed seed so every rank/process sees the same class structure.
_MNIST_PROTO = None


def _mnist_prototypes() -> np.ndarray:
    global _MNIST_PROTO
    if _MNIST_PROTO is None:
        prng = np.random.default_rng(2718)
        _MNIST_PROTO = prng.random((10, 784)) * 0.8
    return _MNIST_PROTO


_MNIST_SWAPS = {1: (1, 2), 2: (3, 4), 3: (5, 6)}


def sample_mnist(n: int, concept: int, rng: np.random.Generator) -> np.ndarray:
    proto = _mnist_prototypes()
    y = rng.integers(0, 10, size=n).astype(np.float64)
    x = proto[y.astype(int)] + rng.normal(0.0, 0.25, size=(n, 784))
    x = np.clip(x, 0.0, 1.0)
    if concept != 0:
        a, b = _MNIST_SWAPS[concept]
        ya = y == a
        yb = y == b
        y[ya] = b
        y[yb] = a
    return np.concatenate([x, y[:, None]], axis=1)


# synthetic stand-ins for the image benchmark configs (no network access):
# class-conditional Gaussian images with label-swap drift concepts, shaped
# like CIFAR-10 (3x32x32, ResNet-18 config) and FederatedEMNIST (28x28,
# 62 classes, CNN ensemble config)
_PROTO_CACHE = {}


def _prototypes(n_classes: int, dim: int, seed: int) -> np.ndarray:
    key = (n_classes, dim, seed)
    if key not in _PROTO_CACHE:
        prng = np.random.default_rng(seed)
        _PROTO_CACHE[key] = prng.random((n_classes, dim)) * 0.8
    return _PROTO_CACHE[key]


def _sample_image_like(n: int, concept: int, rng: np.random.Generator,
                       n_classes: int, dim: int, seed: int) -> np.ndarray:
    proto = _prototypes(n_classes, dim, seed)
    y = rng.integers(0, n_classes, size=n).astype(np.float64)
    x = proto[y.astype(int)] + rng.normal(0.0, 0.25, size=(n, dim))
    x = np.clip(x, 0.0, 1.0)
    if concept != 0:
        a, b = _MNIST_SWAPS.get(concept, (1, 2))
        ya, yb = y == a, y == b
        y[ya] = b
        y[yb] = a
    return np.concatenate([x, y[:, None]], axis=1)


def sample_cifar(n, concept, rng):
    return _sample_image_like(n, concept, rng, 10, 3072, 1414)


def sample_femnist(n, concept, rng):
    return _sample_image_like(n, concept, rng, 62, 784, 1703)


def sample_fmow(n, concept, rng):
    # synthetic twin of the FMoW real-drift features (real path:
    # data/real.py FmowIndexStore over the reference partition layout)
    return _sample_image_like(n, concept, rng, 62, 1024, 2027)


TEXT_SEQ_LEN = 20
TEXT_VOCAB = 30


def sample_text(n: int, concept: int, rng: np.random.Generator) -> np.ndarray:
    """Concept drift over TEXT: rows are char-id context windows
    (TEXT_SEQ_LEN ids) + next-char label, drawn from an order-1 Markov
    chain; a concept change PERMUTES the chain (a 'language shift'), so
    the drift algorithms see the same staggered-concept structure the
    numeric datasets have (data/text_synthetic.py holds the chain
    machinery; CharLSTM in models/rnn.py is the matching model)."""
    from .text_synthetic import _client_chain, _sample_chain
    base = np.random.default_rng(4242)
    # peaky transitions (low Dirichlet concentration): a learnable
    # language with a high Bayes next-char accuracy
    chain = _client_chain(base, TEXT_VOCAB, concentration=0.05)
    if concept != 0:
        perm = np.random.default_rng(1000 + concept).permutation(TEXT_VOCAB)
        chain = chain[perm][:, perm]
    seqs = _sample_chain(rng, chain, n, TEXT_SEQ_LEN)
    return seqs.astype(np.float64)


_SAMPLERS = {"sea": sample_sea, "sine": sample_sine, "circle": sample_circle,
             "MNIST": sample_mnist, "cifar": sample_cifar,
             "femnist": sample_femnist, "text": sample_text,
             "fmow": sample_fmow}

FEATURE_NUM = {"sea": 3, "sine": 2, "circle": 2, "MNIST": 784,
               "cifar": 3072, "femnist": 784, "text": TEXT_SEQ_LEN,
               "fmow": 1024}
CLASS_NUM = {"sea": 2, "sine": 2, "circle": 2, "MNIST": 10,
             "cifar": 10, "femnist": 62, "text": TEXT_VOCAB,
             "fmow": 62}

_SEA_COLS = ["f1", "f2", "f3", "label"]


def change_points_path(data_dir: str, name: str) -> str:
    # our tree: <data_dir>/changepoints/<name>.cp
    return os.path.join(data_dir, "changepoints", f"{name}.cp")


def load_change_points(data_dir: str, name: str) -> np.ndarray:
    return np.loadtxt(change_points_path(data_dir, name), dtype=int)


def make_random_change_points(train_iteration: int, num_client: int,
                              drift_together: int, stretch: int) -> np.ndarray:
    """Single random 0->1 change point per client
    (reference sea/data_loader.py:49-64; uses the global np RNG on purpose:
    the reference seeds np.random with dummy_arg before calling this)."""
    hi = max(2, train_iteration // stretch)
    if drift_together == 1:
        cp = np.random.randint(1, hi)
        cps = [cp] * num_client
    else:
        cps = [np.random.randint(1, hi) for _ in range(num_client)]
    mat = np.zeros((train_iteration // stretch + 1, num_client), dtype=int)
    for c, t in enumerate(cps):
        mat[t:, c] = 1
    return mat


def generate_data(dataset: str, data_dir: str, train_iteration: int,
                  num_client: int, drift_together: int,
                  sample_per_client_iter: int, noise_prob: float,
                  stretch_factor: int, change_point_str: str = "rand") -> None:
    """Write client_{c}_iter_{t}.csv for t in [0, train_iteration].

    The extra label-noise flip with probability noise_prob matches
    sea/data_loader.py:76 (binary flip) and MNIST add_noise (random other
    digit, data_loader_cont.py:40-48).
    """
    dataset = "MNIST" if dataset.lower() == "mnist" else dataset
    ds_dir = os.path.join(data_dir, dataset)
    os.makedirs(ds_dir, exist_ok=True)
    os.makedirs(os.path.join(data_dir, "changepoints"), exist_ok=True)

    if change_point_str == "rand":
        mat = make_random_change_points(train_iteration, num_client,
                                        drift_together, stretch_factor)
        np.savetxt(change_points_path(data_dir, "rand"), mat, fmt="%u")
    change_point = load_change_points(data_dir, change_point_str)

    sampler = _SAMPLERS[dataset]
    # real-pixel source: when the LEAF json layout is present under the
    # dataset dir (a user brought the reference's data/MNIST/train+test
    # along), draw from it with the reference MNIST_Data semantics
    # instead of the synthetic prototypes (data/real.py)
    if dataset in ("MNIST", "femnist"):
        from .real import LeafSampleSource, leaf_layout_present
        if leaf_layout_present(ds_dir):
            sampler = LeafSampleSource(ds_dir).generate_sample
    rng = np.random.default_rng(np.random.randint(0, 2**31))
    n_classes = CLASS_NUM[dataset]

    header = ",".join(_SEA_COLS) if dataset == "sea" else \
        ",".join(str(i) for i in range(FEATURE_NUM[dataset])) + ",label"

    for it in range(train_iteration + 1):
        for c in range(num_client):
            concept = int(change_point[it // stretch_factor][c])
            arr = sampler(sample_per_client_iter, concept, rng)
            if noise_prob > 0:
                flip = rng.random(len(arr)) < noise_prob
                if n_classes == 2:
                    arr[flip, -1] = 1.0 - arr[flip, -1]
                else:
                    for i in np.nonzero(flip)[0]:
                        choices = np.delete(np.arange(n_classes), int(arr[i, -1]))
                        arr[i, -1] = rng.choice(choices)
            path = os.path.join(ds_dir, f"client_{c}_iter_{it}.csv")
            np.savetxt(path, arr, delimiter=",", header=header, comments="",
                       fmt="%.9g")
