#!/usr/bin/env python3
"""Single-GPU load sweep: FedDrift rounds/sec vs client count (bench.py's
steady-state workload at increasing scale — per-round work grows linearly
with clients; shows the engine's throughput envelope on one MI355X)."""

import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import bench
from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.eval.metrics import MetricLogger
from feddrift_amd.engine.fljob import FLJob


def run_scale(comm, n_clients, steps=200, warmup=40):
    cfg = Config(model="fnn", dataset="sea", data_dir="/nonexistent",
                 client_num_in_total=n_clients,
                 client_num_per_round=n_clients, batch_size=bench.SEQ,
                 client_optimizer="adam", lr=0.01, epochs=bench.EPOCHS,
                 comm_round=10 ** 9,
                 total_train_iteration=bench.CURR_ITER + 1,
                 curr_train_iteration=bench.CURR_ITER,
                 concept_num=bench.N_MODELS,
                 concept_drift_algo="softcluster",
                 concept_drift_algo_arg="H_A_F_1_06_0", change_points="A",
                 dummy_arg=0, report_client=0, bench_mode=1)
    ds = bench.build_dataset(n_clients, seed=1234)
    job = FLJob(cfg, comm, MetricLogger(enabled=True, to_file=False),
                dataset=ds)
    idx = np.arange(n_clients)
    for r in range(warmup):
        bench.one_round(job, r, idx)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for r in range(warmup, warmup + steps):
        bench.one_round(job, r, idx)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return steps / dt


def main():
    comm = Communicator()
    out = {}
    for c, steps in [(10, 400), (50, 300), (200, 200), (1000, 60),
                     (3400, 20)]:
        rps = run_scale(comm, c, steps=steps, warmup=max(5, steps // 5))
        out[str(c)] = {"rounds_per_sec": round(rps, 1),
                       "client_rounds_per_sec": round(rps * c, 1)}
        print(f"clients={c:5d}: {rps:8.1f} rounds/s "
              f"({rps * c:10.0f} client-rounds/s)")
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/bench_sweep.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
