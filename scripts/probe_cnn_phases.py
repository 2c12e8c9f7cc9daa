#!/usr/bin/env python3
"""Phase breakdown of a config-3-shaped CNN round (HIP vs vmap engines),
with device sync around each phase so the time lands where the work is."""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.data.generators import sample_mnist
from feddrift_amd.data.loader import DriftDataset
from feddrift_amd.engine.fljob import FLJob
from feddrift_amd.eval.metrics import MetricLogger


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rounds", type=int, default=10)
    p.add_argument("--clients", type=int, default=10)
    p.add_argument("--out", default="")
    a = p.parse_args()
    comm = Communicator()
    n_cl = a.clients
    ds = DriftDataset(data_dir="/nonexistent", dataset="MNIST",
                      num_client=n_cl)
    rng = np.random.default_rng(0)
    for c in range(n_cl):
        for t in range(5):
            arr = sample_mnist(200, c % 4 if t >= 3 else 0, rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])
    cfg = Config(model="cnn", dataset="MNIST", data_dir="/nonexistent",
                 client_num_in_total=n_cl, client_num_per_round=n_cl,
                 batch_size=100, lr=0.003, epochs=5, comm_round=a.rounds,
                 total_train_iteration=4, curr_train_iteration=3,
                 concept_num=4, concept_drift_algo="softcluster",
                 concept_drift_algo_arg="mmacc_06",
                 log_dir="/tmp/probe_cnn", report_client=0, bench_mode=1)
    os.makedirs("/tmp/probe_cnn", exist_ok=True)
    job = FLJob(cfg, comm, MetricLogger(enabled=False, to_file=False),
                dataset=ds)
    print("engine:", type(job.mod_engine).__name__,
          "n_models:", job.n_models)
    phases = {}

    def run_round(r, client_idx):
        t = {}
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        plan = job.algo.plan(job, r, client_idx)
        torch.cuda.synchronize()
        t["plan"] = time.perf_counter() - t0
        t0 = time.perf_counter()
        job.train(plan)
        torch.cuda.synchronize()
        t["train"] = time.perf_counter() - t0
        t0 = time.perf_counter()
        job.algo.aggregate(job, r, plan, client_idx)
        job.algo.post_aggregate(job, r)
        torch.cuda.synchronize()
        t["aggregate"] = time.perf_counter() - t0
        t0 = time.perf_counter()
        job.algo.test(job, r)
        torch.cuda.synchronize()
        t["test"] = time.perf_counter() - t0
        return t

    client_idx = job.client_sampling(0)
    run_round(0, client_idx)  # warmup
    for r in range(1, a.rounds):
        t = run_round(r, client_idx)
        for k, v in t.items():
            phases[k] = phases.get(k, 0.0) + v
    n = a.rounds - 1
    tot = sum(phases.values())
    print(f"per-round total {tot / n * 1e3:.1f} ms")
    for k, v in phases.items():
        print(f"  {k:10s} {v / n * 1e3:8.2f} ms")
    if a.out:
        with open(a.out, "w") as f:
            json.dump({k: v / n * 1e3 for k, v in phases.items()}, f)


if __name__ == "__main__":
    main()
