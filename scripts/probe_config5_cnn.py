#!/usr/bin/env python3
"""Config-5-as-named probe: 3400 clients, CNN_DropOut K<=4 AUE ensemble —
measure per-round wall time on one GPU (the 8-GPU node divides the pairs
and the eval sweep 8 ways)."""

import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.data.generators import sample_femnist
from feddrift_amd.data.loader import DriftDataset
from feddrift_amd.engine.fljob import FLJob
from feddrift_amd.engine.timeline import clean_state_files
from feddrift_amd.eval.metrics import MetricLogger
import dataclasses


def main():
    n_cl = 3400
    ds = DriftDataset(data_dir="/nonexistent", dataset="femnist",
                      num_client=n_cl)
    rng = np.random.default_rng(0)
    for c in range(n_cl):
        for t in range(3):
            arr = sample_femnist(100, c % 4 if t >= 1 else 0, rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])
    cfg = Config(model="cnn", dataset="femnist", data_dir="/nonexistent",
                 client_num_in_total=n_cl, client_num_per_round=n_cl,
                 batch_size=100, lr=0.003, epochs=5, comm_round=3,
                 total_train_iteration=3, curr_train_iteration=1,
                 concept_num=4, ensemble_window=4,
                 concept_drift_algo="aue", log_dir="/tmp/c5",
                 report_client=0)
    os.makedirs("/tmp/c5", exist_ok=True)
    comm = Communicator()
    clean_state_files(cfg)
    logger = MetricLogger("/tmp/c5", enabled=comm.is_root, to_file=False)
    t0 = time.time()
    job = FLJob(cfg, comm, logger, dataset=ds)
    setup = time.time() - t0
    client_idx = np.arange(n_cl)
    torch.cuda.synchronize()
    t0 = time.time()
    for r in range(2):
        plan = job.algo.plan(job, r, client_idx)
        job.train(plan)
        job.algo.aggregate(job, r, plan, client_idx)
        job.algo.post_aggregate(job, r)
        job.algo.test(job, r)
    torch.cuda.synchronize()
    per_round = (time.time() - t0) / 2
    hbm = torch.cuda.memory_allocated() / 2**30
    out = {"clients": n_cl, "model": "cnn", "K": job.n_models,
           "setup_s": round(setup, 1), "s_per_round": round(per_round, 2),
           "hbm_gib": round(hbm, 2)}
    print(json.dumps(out))
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/config5_cnn_probe.json", "w") as f:
        json.dump(out, f)


if __name__ == "__main__":
    main()
