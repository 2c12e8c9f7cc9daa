"""Data layer: generators, retrain DSL, batchification."""

import json
import os

import numpy as np
import pytest

from feddrift_amd.data.generators import (SEA_THETAS, generate_data,
                                          load_change_points, sample_circle,
                                          sample_mnist, sample_sea,
                                          sample_sine)
from feddrift_amd.data.loader import (RawStore, batchify, load_all_data,
                                      load_retrain_data,
                                      resolve_retrain_rows)


def test_sea_statistics():
    rng = np.random.default_rng(0)
    arr = sample_sea(20000, 0, rng)
    x, y = arr[:, :3], arr[:, 3]
    assert x.min() >= 0 and x.max() <= 10
    # label = [f2+f3 > 8] with 10% flips (measured from the reference's
    # shipped concept CSVs — see data/generators.py)
    s = x[:, 1] + x[:, 2]
    agree = ((s > SEA_THETAS[0]).astype(float) == y).mean()
    assert 0.88 < agree < 0.92


def test_sine_circle_mnist_shapes():
    rng = np.random.default_rng(1)
    assert sample_sine(50, 0, rng).shape == (50, 3)
    assert sample_circle(50, 1, rng).shape == (50, 3)
    m = sample_mnist(20, 1, rng)
    assert m.shape == (20, 785)
    assert set(np.unique(m[:, -1])).issubset(set(range(10)))


def test_mnist_label_swap():
    rng = np.random.default_rng(2)
    base = sample_mnist(5000, 0, rng)
    rng2 = np.random.default_rng(2)
    swapped = sample_mnist(5000, 1, rng2)
    # same draws, labels 1<->2 swapped
    y0, y1 = base[:, -1], swapped[:, -1]
    assert np.array_equal(y1[y0 == 1], np.full((y0 == 1).sum(), 2))
    assert np.array_equal(y1[y0 == 2], np.full((y0 == 2).sum(), 1))
    assert np.array_equal(y0[(y0 != 1) & (y0 != 2)],
                          y1[(y0 != 1) & (y0 != 2)])


def test_generate_and_load_csv(tmp_path):
    d = str(tmp_path)
    os.makedirs(os.path.join(d, "changepoints"))
    mat = np.zeros((4, 3), dtype=int)
    mat[2:, 1] = 1
    np.savetxt(os.path.join(d, "changepoints", "T.cp"), mat, fmt="%u")
    np.random.seed(0)
    generate_data("sea", d, 3, 3, 0, 100, 0.0, 1, "T")
    store = RawStore(d, "sea", 3)
    x, y = store.get(0, 0)
    assert x.shape == (100, 3) and y.shape == (100,)
    assert load_change_points(d, "T").shape == (4, 3)


def _mini_store():
    store = RawStore("/nonexistent", "sea", 2)
    for c in range(2):
        for t in range(4):
            n = 10 * (t + 1)
            x = np.full((n, 3), float(c * 10 + t), np.float32)
            y = np.arange(n) % 2
            store.put(c, t, x, y)
    return store


def test_retrain_dsl_windows():
    store = _mini_store()
    rng = np.random.default_rng(0)
    # win-2 at iter 2: iterations 1,2 -> 20+30 rows
    x, y = resolve_retrain_rows(store, 0, 2, "win-2", rng)
    assert len(y) == 50
    assert set(np.unique(x[:, 0])) == {1.0, 2.0}
    # all at iter 2: 10+20+30
    x, _ = resolve_retrain_rows(store, 0, 2, "all", rng)
    assert len(x) == 60
    # sel-0,2
    x, _ = resolve_retrain_rows(store, 1, 2, "sel-0,2", rng)
    assert set(np.unique(x[:, 0])) == {10.0, 12.0}
    # clientsel
    spec = json.dumps([[0], [1, 2]])
    x, _ = resolve_retrain_rows(store, 1, 2, "clientsel-" + spec, rng)
    assert set(np.unique(x[:, 0])) == {11.0, 12.0}
    # weight-linear duplicates rows: iter t repeated t+1 times
    x, _ = resolve_retrain_rows(store, 0, 1, "weight-linear", rng)
    assert len(x) == 10 * 1 + 20 * 2
    # weight-exp: 2**t copies
    x, _ = resolve_retrain_rows(store, 0, 2, "weight-exp", rng)
    assert len(x) == 10 * 1 + 20 * 2 + 30 * 4
    # poisson keeps the newest iteration's row count
    x, _ = resolve_retrain_rows(store, 0, 2, "poisson", rng)
    assert len(x) == 30


def test_batchify_windows():
    rng = np.random.default_rng(0)
    x = np.arange(25 * 3, dtype=np.float32).reshape(25, 3)
    y = np.arange(25, dtype=np.int64) % 2
    seg = batchify(x, y, 10, rng)
    assert seg.windows == [(0, 10), (10, 10), (20, 5)]
    # shuffle is a permutation
    assert sorted(seg.y.tolist()) == sorted(y.tolist())
    assert np.allclose(np.sort(seg.x[:, 0]), np.sort(x[:, 0]))


def test_prequential_test_set():
    store = _mini_store()
    rng = np.random.default_rng(0)
    view = load_retrain_data(store, 2, 10, "win-1", rng)
    # test data is ALWAYS iteration t+1 (retrain.py:79-83)
    assert view.test[0].n == 40
    assert np.unique(view.test[0].x[:, 0]) == [3.0]
    assert view.train[0].n == 30


def test_leaf_synthetic_generator():
    from feddrift_amd.data.leaf_synthetic import generate_synthetic
    data = generate_synthetic(0.5, 0.5, n_clients=8, dim=20, n_classes=5,
                              seed=0)
    assert len(data) == 8
    for k, (x, y) in data.items():
        assert x.shape[1] == 20 and x.shape[0] == len(y) >= 5
        assert y.min() >= 0 and y.max() < 5
    # heterogeneity: different clients see different label distributions
    h0 = np.bincount(data[0][1], minlength=5) / len(data[0][1])
    h1 = np.bincount(data[1][1], minlength=5) / len(data[1][1])
    assert np.abs(h0 - h1).sum() > 0.1


def test_reference_style_csv_interop(tmp_path):
    """CSV files written the reference's way (pandas to_csv, header row,
    float label column — sea/data_loader.py:80-82, MNIST
    data_loader_cont.py:85-88) load correctly through our RawStore."""
    import pandas as pd
    d = tmp_path / "sea"
    d.mkdir()
    df = pd.DataFrame({"f1": [1.5, 2.5], "f2": [3.5, 4.5],
                       "f3": [5.5, 6.5], "label": [0.0, 1.0]})
    df.to_csv(d / "client_0_iter_0.csv", index=False)
    # MNIST-style: integer column names, float labels
    arr = np.random.default_rng(0).random((3, 4))
    df2 = pd.DataFrame(np.column_stack([arr, [0.0, 7.0, 3.0]]))
    df2.to_csv(d / "client_1_iter_0.csv", index=False)

    store = RawStore(str(tmp_path), "sea", 2)
    x, y = store.get(0, 0)
    assert x.shape == (2, 3) and y.tolist() == [0, 1]
    x2, y2 = store.get(1, 0)
    assert x2.shape == (3, 4) and y2.tolist() == [0, 7, 3]
