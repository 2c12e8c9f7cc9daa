#!/usr/bin/env python3
"""Canonical SEA-4 drift experiments (reference README.md:45-48 config):
10 clients, FNN, 200 rounds x 5 epochs, batch 500, lr 0.01, 100 samples per
(client, iteration), 10 iterations, change-point matrix A — across the
drift-algorithm surface. Writes results JSON with the north-star metric
(avg Test/Acc over the drift timeline)."""

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.data.generators import generate_data
from feddrift_amd.engine.timeline import run_timeline

ALGOS = [
    ("softcluster", "H_A_C_1_10_0"),    # FedDrift (delta=0.10)
    ("softcluster", "H_A_C_1_06_0"),    # FedDrift (delta=0.06)
    ("softcluster", "H_A_F_1_06_0"),    # FedDrift (per-client init)
    ("softcluster", "mmacc_06"),        # FedDrift-Eager
    ("softcluster", "hard"),            # IFCA
    ("softcluster", "geni"),            # clustering oracle
    ("aue", ""),
    ("auepc", ""),
    ("kue", ""),
    ("driftsurf", ""),
    ("ada", "win-1_round"),
    ("exp", ""),
    ("lin", ""),
    ("mmacc", ""),                      # FedDrift-Eager precursor
    ("mmgeniex", ""),                   # oracle
    ("single", "win-1"),                # oblivious win-1
    ("single", "all"),                  # oblivious all
]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="gpurun_out/sea_results.json")
    p.add_argument("--data_dir", default="/tmp/sea_exp")
    p.add_argument("--rounds", type=int, default=200)
    p.add_argument("--iters", type=int, default=10)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--algos", default="")   # comma filter, e.g. 'softcluster'
    p.add_argument("--cp", default="A")      # change-point matrix name
    p.add_argument("--dataset", default="sea")  # sea | sine | circle
    a = p.parse_args()

    comm = Communicator()
    np.random.seed(a.seed)
    if comm.is_root:
        import shutil
        repo_cp = os.path.join(os.path.dirname(__file__), "..", "data",
                               "changepoints")
        os.makedirs(os.path.join(a.data_dir, "changepoints"), exist_ok=True)
        for f in os.listdir(repo_cp):
            if f.endswith(".cp"):
                shutil.copy(os.path.join(repo_cp, f),
                            os.path.join(a.data_dir, "changepoints", f))
        generate_data(a.dataset, a.data_dir, a.iters, 10, 0, 100, 0.0, 1, a.cp)
    comm.barrier()

    results = {}
    for algo, arg in ALGOS:
        if a.algos and not any(tok and tok in f"{algo}:{arg}"
                               for tok in a.algos.split(",")):
            continue
        name = f"{algo}:{arg}" if arg else algo
        log_dir = os.path.join(a.data_dir, "run_" + name.replace(":", "_"))
        os.makedirs(log_dir, exist_ok=True)
        # the F (per-client-init) FedDrift variant starts one model per
        # client, so the ensemble cap equals the client count
        k_cap = 10 if "_F_" in arg else 4
        cfg = Config(
            model="fnn", dataset=a.dataset, data_dir=a.data_dir,
            client_num_in_total=10, client_num_per_round=10,
            batch_size=500, lr=0.01, epochs=5, comm_round=a.rounds,
            total_train_iteration=a.iters, concept_num=k_cap,
            concept_drift_algo=algo,
            concept_drift_algo_arg=arg,
            retrain_data=arg if algo == "single" else "win-1",
            change_points=a.cp, dummy_arg=a.seed, sample_num=100,
            log_dir=log_dir, report_client=0)
        t0 = time.time()
        out = run_timeline(cfg, comm)
        dt = time.time() - t0
        results[name] = {
            "avg_test_acc": out["avg_test_acc"],
            "per_iteration": [round(x, 4) for x in
                              out["per_iteration_test_acc"]],
            "wall_s": round(dt, 1),
        }
        if comm.is_root:
            print(f"{name:32s} avg={out['avg_test_acc']:.4f}  ({dt:.0f}s)")
    if comm.is_root:
        os.makedirs(os.path.dirname(a.out), exist_ok=True)
        with open(a.out, "w") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
