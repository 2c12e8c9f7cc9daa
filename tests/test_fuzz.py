"""Configuration fuzz: random tiny timelines across the full algorithm /
dataset / option surface must run without crashing and produce finite
metrics (robustness against config corners a user may hit)."""

import dataclasses
import os

import numpy as np
import pytest

from feddrift_amd.config import Config
from feddrift_amd.data.generators import generate_data
from feddrift_amd.engine.timeline import run_timeline

ALGOS = [
    ("softcluster", "H_A_C_1_10_0"), ("softcluster", "H_B_D_2_06_08"),
    ("softcluster", "hard"), ("softcluster", "hard-r"),
    ("softcluster", "softmax_1"), ("softcluster", "mmacc_04"),
    ("softcluster", "cfl_0.2_all"), ("softclusterreset", "softmax_0"),
    ("softclusterwin-1", "H_A_C_1_10_0"),
    ("aue", ""), ("auepc", ""), ("kue", ""), ("driftsurf", "5"),
    ("ada", "all_round"), ("ada", "win-1_iter"), ("exp", ""), ("lin", ""),
    ("mmacc", ""), ("single", ""),
]


@pytest.fixture(scope="module")
def fuzz_data(tmp_path_factory):
    dirs = {}
    for ds in ["sea", "sine", "circle"]:
        d = str(tmp_path_factory.mktemp(ds))
        os.makedirs(os.path.join(d, "changepoints"), exist_ok=True)
        mat = np.zeros((6, 7), dtype=int)
        mat[2:, :3] = 1
        mat[4:, 3] = 1
        np.savetxt(os.path.join(d, "changepoints", "T.cp"), mat, fmt="%u")
        np.random.seed(7)
        generate_data(ds, d, 5, 7, 0, 90, 0.05, 1, "T")
        dirs[ds] = d
    return dirs


def test_fuzz_configs(fuzz_data, tmp_path):
    rng = np.random.default_rng(123)
    for trial in range(24):
        algo, arg = ALGOS[int(rng.integers(0, len(ALGOS)))]
        ds = ["sea", "sine", "circle"][int(rng.integers(0, 3))]
        n_workers = int(rng.integers(3, 8))      # may be < clients: sampling
        cfg = Config(
            model=["fnn", "lr"][int(rng.integers(0, 2))],
            dataset=ds, data_dir=fuzz_data[ds],
            client_num_in_total=7,
            client_num_per_round=n_workers,
            batch_size=int(rng.integers(16, 91)),
            client_optimizer=["adam", "sgd"][int(rng.integers(0, 2))],
            lr=0.01, epochs=int(rng.integers(1, 5)),
            comm_round=int(rng.integers(2, 5)),
            total_train_iteration=int(rng.integers(2, 4)),
            concept_num=int(rng.integers(2, 4)),
            concept_drift_algo=algo, concept_drift_algo_arg=arg,
            retrain_data=["win-1", "win-2", "all",
                          "weight-linear"][int(rng.integers(0, 4))],
            change_points="T", dummy_arg=int(rng.integers(0, 5)),
            frequency_of_the_test=int(rng.integers(1, 3)),
            ci=int(rng.integers(0, 2)),
            log_dir=str(tmp_path / f"t{trial}"), report_client=0)
        os.makedirs(cfg.log_dir, exist_ok=True)
        out = run_timeline(cfg)
        assert np.isfinite(out["avg_test_acc"]), (trial, algo, arg, ds)
        assert 0.0 <= out["avg_test_acc"] <= 1.0, (trial, algo, arg)
