#!/usr/bin/env python3
"""Long-running CPU configuration soak: random timelines across the full
algorithm / dataset / model / option surface (the same generator as
tests/test_fuzz.py, scaled up and widened). Crashes or non-finite
metrics fail loudly; use for pre-release robustness sweeps.

Usage: python scripts/soak_cpu.py [N_TRIALS] [SEED]
"""

import os
import sys
import tempfile
import time

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.config import Config
from feddrift_amd.data.generators import generate_data
from feddrift_amd.engine.timeline import run_timeline

ALGOS = [
    ("softcluster", "H_A_C_1_10_0"), ("softcluster", "H_B_D_2_06_08"),
    ("softcluster", "H_A_F_1_06_0"), ("softcluster", "H_A_E_1_10_0"),
    ("softcluster", "hard"), ("softcluster", "hard-r"),
    ("softcluster", "softmax_1"), ("softcluster", "mmacc_04"),
    ("softcluster", "geni"), ("softcluster", "gmm"),
    ("softcluster", "cfl_0.2_all"), ("softclusterreset", "softmax_0"),
    ("softclusterwin-1", "H_A_C_1_10_0"),
    ("aue", ""), ("auepc", ""), ("kue", ""), ("driftsurf", "5"),
    ("dsurf", ""), ("ada", "all_round"), ("ada", "win-1_iter"),
    ("exp", ""), ("lin", ""), ("mmacc", ""), ("mmgeni", ""),
    ("mmgeniex", ""), ("single", ""),
]
RETRAIN = ["win-1", "win-2", "all", "weight-linear", "weight-exp",
           "poisson"]


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 100
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 123
    rng = np.random.default_rng(seed)
    root = tempfile.mkdtemp(prefix="soak_")
    dirs = {}
    for ds in ["sea", "sine", "circle"]:
        d = os.path.join(root, ds)
        os.makedirs(os.path.join(d, "changepoints"), exist_ok=True)
        import shutil
        cps = os.path.join(os.path.dirname(__file__), "..", "data",
                           "changepoints")
        for f in os.listdir(cps):
            if f.endswith(".cp"):
                shutil.copy(os.path.join(cps, f),
                            os.path.join(d, "changepoints", f))
        mat = np.zeros((6, 7), dtype=int)       # staggered 2-3 concept
        mat[2:, :3] = 1
        mat[4:, 3] = 1
        np.savetxt(os.path.join(d, "changepoints", "T.cp"), mat, fmt="%u")
        np.random.seed(seed)
        generate_data(ds, d, 4, 7, 0, 60, 0.0, 1, "T")
        dirs[ds] = d
    t0 = time.time()
    for trial in range(n):
        algo, arg = ALGOS[int(rng.integers(0, len(ALGOS)))]
        ds = ["sea", "sine", "circle"][int(rng.integers(0, 3))]
        cfg = Config(
            model=["fnn", "lr"][int(rng.integers(0, 2))],
            dataset=ds, data_dir=dirs[ds],
            client_num_in_total=7,
            client_num_per_round=int(rng.integers(3, 8)),
            batch_size=int(rng.integers(8, 91)),
            client_optimizer=["adam", "sgd"][int(rng.integers(0, 2))],
            lr=0.01, epochs=int(rng.integers(1, 5)),
            comm_round=int(rng.integers(2, 6)),
            total_train_iteration=int(rng.integers(2, 5)),
            # the F (per-client init) variant starts one model per
            # client, so its cap must cover the client count
            concept_num=(7 if "_F_" in arg else int(rng.integers(2, 5))),
            concept_drift_algo=algo, concept_drift_algo_arg=arg,
            retrain_data=RETRAIN[int(rng.integers(0, len(RETRAIN)))],
            change_points="T", dummy_arg=int(rng.integers(0, 5)),
            frequency_of_the_test=int(rng.integers(1, 3)),
            ci=int(rng.integers(0, 2)),
            time_stretch=int(rng.choice([1, 1, 2])),
            log_dir=os.path.join(root, f"t{trial}"), report_client=0)
        os.makedirs(cfg.log_dir, exist_ok=True)
        out = run_timeline(cfg)
        acc = out["avg_test_acc"]
        assert np.isfinite(acc) and 0.0 <= acc <= 1.0, (trial, algo, arg)
        if (trial + 1) % 20 == 0:
            print(f"{trial + 1}/{n} ok ({time.time() - t0:.0f}s)",
                  flush=True)
    print(f"soak OK: {n} configs in {time.time() - t0:.0f}s")


if __name__ == "__main__":
    main()
