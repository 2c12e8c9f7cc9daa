#!/usr/bin/env python3
"""MNIST label-swap drift sweep (BASELINE config 3 shape): CNN_DropOut,
10 clients, 4 concepts (identity + 3 label swaps), staggered drift.
Clustered methods should dominate oblivious baselines (conflicting label
swaps destroy a shared model) — the paper's main figure."""

import argparse
import dataclasses
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.data.generators import sample_mnist
from feddrift_amd.data.loader import DriftDataset
from feddrift_amd.engine.fljob import FLJob
from feddrift_amd.engine.timeline import clean_state_files
from feddrift_amd.eval.metrics import MetricLogger

ALGOS = [
    ("softcluster", "H_A_C_1_10_0"),    # FedDrift
    ("softcluster", "mmacc_06"),        # FedDrift-Eager
    ("softcluster", "hard"),            # IFCA
    ("aue", ""),
    ("driftsurf", ""),
    ("mmgeniex", ""),                   # oracle
    ("single", "win-1"),
    ("single", "all"),
]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="gpurun_out/mnist_results.json")
    p.add_argument("--rounds", type=int, default=20)
    p.add_argument("--iters", type=int, default=6)
    p.add_argument("--clients", type=int, default=10)
    p.add_argument("--samples", type=int, default=200)
    p.add_argument("--algos", default="")
    a = p.parse_args()
    comm = Communicator()

    # staggered drift: client c switches to concept c%4 at iteration 3
    # (write a matching change-point matrix for the oracle variants)
    rng = np.random.default_rng(0)
    ds = DriftDataset(data_dir="/tmp/mnist_x", dataset="MNIST",
                      num_client=a.clients)
    cp = np.zeros((a.iters + 1, a.clients), dtype=int)
    for c in range(a.clients):
        cp[3:, c] = c % 4
    os.makedirs("/tmp/mnist_x/changepoints", exist_ok=True)
    np.savetxt("/tmp/mnist_x/changepoints/S.cp", cp, fmt="%u")
    for c in range(a.clients):
        for t in range(a.iters + 1):
            arr = sample_mnist(a.samples, int(cp[t, c]), rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])

    results = {}
    for algo, arg in ALGOS:
        if a.algos and a.algos not in algo:
            continue
        name = f"{algo}:{arg}" if arg else algo
        log_dir = "/tmp/mnist_x/run_" + name.replace(":", "_")
        os.makedirs(log_dir, exist_ok=True)
        cfg = Config(model="cnn", dataset="MNIST", data_dir="/tmp/mnist_x",
                     client_num_in_total=a.clients,
                     client_num_per_round=a.clients,
                     batch_size=100, lr=0.003, epochs=5,
                     comm_round=a.rounds, total_train_iteration=a.iters,
                     concept_num=4,
                     concept_drift_algo=algo, concept_drift_algo_arg=arg,
                     retrain_data=arg if algo == "single" else "win-1",
                     change_points="S", log_dir=log_dir, report_client=0)
        if comm.is_root:
            clean_state_files(cfg)
        comm.barrier()
        t0 = time.time()
        accs = []
        for it in range(a.iters):
            icfg = dataclasses.replace(cfg, curr_train_iteration=it)
            logger = MetricLogger(log_dir, enabled=comm.is_root,
                                  to_file=False)
            job = FLJob(icfg, comm, logger, dataset=ds)
            job.run()
            accs.append(logger.mean("Test/Acc"))
        dt = time.time() - t0
        results[name] = {"avg_test_acc": float(np.mean(accs)),
                         "per_iteration": [round(x, 4) for x in accs],
                         "wall_s": round(dt, 1)}
        if comm.is_root:
            print(f"{name:32s} avg={np.mean(accs):.4f}  ({dt:.0f}s)")
    if comm.is_root:
        os.makedirs(os.path.dirname(a.out), exist_ok=True)
        with open(a.out, "w") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
