#!/usr/bin/env python3
"""BASELINE.json configs 3-5 demonstrations on synthetic drift data:

  3. MNIST label-swap drift, CNN_DropOut, FedDrift-Eager (mmacc_06)
  4. CIFAR-shaped label-swap drift, ResNet-18, IFCA (softcluster hard)
  5. FEMNIST-scale: 3400 clients, K-model AUE ensemble, everything
     resident in HBM (lr tower; the CNN ensemble at this client count is
     a multi-node job)

Each runs a shortened timeline and reports avg Test/Acc + per-round wall
time. Sizes are chosen to finish in minutes on one MI355X; pass --full
for the reference-length runs."""

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.data.generators import sample_cifar, sample_femnist, sample_mnist
from feddrift_amd.data.loader import DriftDataset
from feddrift_amd.engine.timeline import run_timeline
from feddrift_amd.eval.metrics import MetricLogger
from feddrift_amd.engine.fljob import FLJob
import dataclasses


def build_ds(dataset, sampler, n_clients, iters, n, concept_of, seed=0):
    ds = DriftDataset(data_dir="/nonexistent", dataset=dataset,
                      num_client=n_clients)
    rng = np.random.default_rng(seed)
    for c in range(n_clients):
        for t in range(iters + 1):
            arr = sampler(n, concept_of(c, t), rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])
    return ds


def run_cfg(name, cfg, ds, comm, results):
    t0 = time.time()
    accs = []
    from feddrift_amd.engine.timeline import clean_state_files
    if comm.is_root:
        clean_state_files(cfg)
    comm.barrier()
    for it in range(cfg.total_train_iteration):
        icfg = dataclasses.replace(cfg, curr_train_iteration=it)
        logger = MetricLogger(cfg.log_dir, enabled=comm.is_root,
                              to_file=False)
        job = FLJob(icfg, comm, logger, dataset=ds)
        job.run()
        accs.append(logger.mean("Test/Acc"))
    dt = time.time() - t0
    n_rounds = cfg.comm_round * cfg.total_train_iteration
    results[name] = {
        "avg_test_acc": float(np.mean(accs)),
        "per_iteration": [round(a, 4) for a in accs],
        "wall_s": round(dt, 1),
        "rounds": n_rounds,
        "ms_per_round": round(dt / n_rounds * 1e3, 1),
    }
    if comm.is_root:
        print(f"{name:36s} acc={np.mean(accs):.4f} "
              f"({dt:.0f}s, {dt / n_rounds * 1e3:.0f} ms/round)")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="gpurun_out/config_results.json")
    p.add_argument("--full", action="store_true")
    p.add_argument("--skip", default="")
    a = p.parse_args()
    skip = set(a.skip.split(","))
    comm = Communicator()
    results = {}
    tmp = "/tmp/cfg_runs"
    os.makedirs(tmp, exist_ok=True)

    # -- config 3: MNIST label-swap CNN, FedDrift-Eager -----------------
    if "mnist" not in skip:
        iters = 6 if not a.full else 10
        ds = build_ds("MNIST", sample_mnist, 10, iters, 200,
                      lambda c, t: (c % 4) if t >= 3 else 0)
        cfg = Config(model="cnn", dataset="MNIST", data_dir="/nonexistent",
                     client_num_in_total=10, client_num_per_round=10,
                     batch_size=100, lr=0.003, epochs=5,
                     comm_round=20 if not a.full else 100,
                     total_train_iteration=iters, concept_num=4,
                     concept_drift_algo="softcluster",
                     concept_drift_algo_arg="mmacc_06",
                     log_dir=tmp, report_client=0)
        run_cfg("mnist_cnn_feddrift_eager", cfg, ds, comm, results)

    # -- config 4: CIFAR-shaped ResNet-18, IFCA --------------------------
    if "cifar" not in skip:
        n_cl = 20 if not a.full else 100
        iters = 3
        ds = build_ds("cifar", sample_cifar, n_cl, iters, 128,
                      lambda c, t: (c % 2) if t >= 2 else 0)
        cfg = Config(model="resnet", dataset="cifar",
                     data_dir="/nonexistent",
                     client_num_in_total=n_cl, client_num_per_round=n_cl,
                     batch_size=64, lr=0.001, epochs=2,
                     comm_round=8 if not a.full else 50,
                     total_train_iteration=iters, concept_num=2,
                     concept_drift_algo="softcluster",
                     concept_drift_algo_arg="hard",
                     log_dir=tmp, report_client=0)
        run_cfg("cifar_resnet18_ifca", cfg, ds, comm, results)

    # -- config 5: FEMNIST-scale AUE ensemble, 3400 clients --------------
    if "femnist" not in skip:
        n_cl = 3400 if not a.full else 3400
        iters = 3
        ds = build_ds("femnist", sample_femnist, n_cl, iters, 100,
                      lambda c, t: (c % 4) if t >= 2 else 0)
        cfg = Config(model="lr", dataset="femnist", data_dir="/nonexistent",
                     client_num_in_total=n_cl, client_num_per_round=n_cl,
                     batch_size=100, lr=0.01, epochs=5,
                     comm_round=10 if not a.full else 50,
                     total_train_iteration=iters, concept_num=4,
                     ensemble_window=4,
                     concept_drift_algo="aue",
                     log_dir=tmp, report_client=0)
        run_cfg("femnist3400_aue_ensemble", cfg, ds, comm, results)

    # -- config 5b: FEMNIST CNN ensemble (vmap-batched, 400 clients) -----
    if "femnistcnn" not in skip:
        n_cl = 400
        iters = 2
        ds = build_ds("femnist", sample_femnist, n_cl, iters, 100,
                      lambda c, t: (c % 4) if t >= 2 else 0, seed=1)
        cfg = Config(model="cnn", dataset="femnist",
                     data_dir="/nonexistent",
                     client_num_in_total=n_cl, client_num_per_round=n_cl,
                     batch_size=100, lr=0.003, epochs=5,
                     comm_round=30 if not a.full else 100,
                     total_train_iteration=iters, concept_num=4,
                     ensemble_window=4, concept_drift_algo="aue",
                     log_dir=tmp, report_client=0)
        run_cfg("femnist400_cnn_aue_vmap", cfg, ds, comm, results)

    if comm.is_root:
        os.makedirs(os.path.dirname(a.out), exist_ok=True)
        with open(a.out, "w") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
