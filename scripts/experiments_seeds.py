#!/usr/bin/env python3
"""Seed-variance study of the north-star metric: canonical SEA-4 FedDrift
(softcluster H_A_C_1_10_0, change-point A) across seeds — the defensible
parity target is statistical (same metric +- noise over seeds), stated in
SURVEY.md section 7."""

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


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="gpurun_out/seed_variance.json")
    p.add_argument("--seeds", type=int, default=10)
    p.add_argument("--rounds", type=int, default=200)
    p.add_argument("--algo", default="softcluster")
    p.add_argument("--arg", default="H_A_C_1_10_0")
    p.add_argument("--dataset", default="sea")
    a = p.parse_args()
    comm = Communicator()

    accs = []
    for seed in range(a.seeds):
        data_dir = f"/tmp/seed_{a.dataset}_{seed}"
        if comm.is_root:
            import shutil
            repo_cp = os.path.join(os.path.dirname(__file__), "..", "data",
                                   "changepoints")
            os.makedirs(os.path.join(data_dir, "changepoints"),
                        exist_ok=True)
            for f in os.listdir(repo_cp):
                if f.endswith(".cp"):
                    shutil.copy(os.path.join(repo_cp, f),
                                os.path.join(data_dir, "changepoints", f))
            np.random.seed(seed)
            generate_data(a.dataset, data_dir, 10, 10, 0, 100, 0.0, 1, "A")
        comm.barrier()
        log_dir = os.path.join(data_dir, "run")
        os.makedirs(log_dir, exist_ok=True)
        cfg = Config(model="fnn", dataset=a.dataset, data_dir=data_dir,
                     client_num_in_total=10, client_num_per_round=10,
                     batch_size=500, lr=0.01, epochs=5,
                     comm_round=a.rounds, total_train_iteration=10,
                     concept_num=4, concept_drift_algo=a.algo,
                     concept_drift_algo_arg=a.arg, change_points="A",
                     dummy_arg=seed, sample_num=100, log_dir=log_dir,
                     report_client=0)
        out = run_timeline(cfg, comm)
        accs.append(out["avg_test_acc"])
        if comm.is_root:
            print(f"seed {seed}: {out['avg_test_acc']:.4f}")
    if comm.is_root:
        arr = np.array(accs)
        summary = {"algo": f"{a.algo}:{a.arg}", "seeds": a.seeds,
                   "mean": float(arr.mean()), "std": float(arr.std()),
                   "min": float(arr.min()), "max": float(arr.max()),
                   "values": [round(v, 4) for v in accs]}
        print(json.dumps(summary))
        os.makedirs(os.path.dirname(a.out), exist_ok=True)
        with open(a.out, "w") as f:
            json.dump(summary, f, indent=1)


if __name__ == "__main__":
    main()
