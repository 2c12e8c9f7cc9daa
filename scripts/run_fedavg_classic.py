#!/usr/bin/env python3
"""Classic one-shot FedAvg benchmarks (non-drift).

Counterpart of fedml_experiments/distributed/fedavg and the vanilla
fedml_api/distributed/fedavg package: a static dataset partitioned across
clients (homo or Dirichlet hetero), R rounds of FedAvg with client
sampling, accuracy on a held-out test split. Runs on the same engine as
the drift path: the train partition becomes iteration 0 and the test
split iteration 1 (prequential eval of iteration t+1 == classic test-set
eval when T=1).

Example:
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      scripts/run_fedavg_classic.py --dataset cifar --model resnet \
      --client_num_in_total 100 --client_num_per_round 10 \
      --partition_method hetero --comm_round 100
"""

import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.data.generators import _SAMPLERS, CLASS_NUM
from feddrift_amd.data.loader import DriftDataset
from feddrift_amd.data.partition import partition
from feddrift_amd.engine.fljob import FLJob
from feddrift_amd.eval.metrics import MetricLogger


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--dataset", default="cifar")
    p.add_argument("--model", default="resnet")
    p.add_argument("--client_num_in_total", type=int, default=16)
    p.add_argument("--client_num_per_round", type=int, default=8)
    p.add_argument("--partition_method", default="hetero")
    p.add_argument("--partition_alpha", type=float, default=0.5)
    p.add_argument("--n_train", type=int, default=4000)
    p.add_argument("--n_test_per_client", type=int, default=100)
    p.add_argument("--comm_round", type=int, default=20)
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch_size", type=int, default=64)
    p.add_argument("--lr", type=float, default=0.003)
    p.add_argument("--client_optimizer", default="adam")
    p.add_argument("--server_optimizer", default="avg")
    p.add_argument("--server_lr", type=float, default=1.0)
    p.add_argument("--robust_norm_bound", type=float, default=0.0)
    p.add_argument("--robust_noise", type=float, default=0.0)
    p.add_argument("--seed", type=int, default=0)
    a = p.parse_args()

    comm = Communicator()
    rng = np.random.default_rng(a.seed)
    sampler = _SAMPLERS[a.dataset]
    full = sampler(a.n_train, 0, rng)
    labels = full[:, -1].astype(int)
    parts = partition(a.partition_method, labels, a.client_num_in_total,
                      a.partition_alpha, seed=a.seed)

    ds = DriftDataset(data_dir="/nonexistent", dataset=a.dataset,
                      num_client=a.client_num_in_total)
    for c, idx in parts.items():
        ds.store.put(c, 0, full[idx, :-1], full[idx, -1])
        test = sampler(a.n_test_per_client, 0, rng)
        ds.store.put(c, 1, test[:, :-1], test[:, -1])

    cfg = Config(model=a.model, dataset=a.dataset, data_dir="/nonexistent",
                 client_num_in_total=a.client_num_in_total,
                 client_num_per_round=a.client_num_per_round,
                 batch_size=a.batch_size, lr=a.lr,
                 client_optimizer=a.client_optimizer, epochs=a.epochs,
                 comm_round=a.comm_round, total_train_iteration=1,
                 curr_train_iteration=0, concept_num=1,
                 concept_drift_algo="single", retrain_data="win-1",
                 dummy_arg=a.seed, report_client=0,
                 server_optimizer=a.server_optimizer, server_lr=a.server_lr,
                 robust_norm_bound=a.robust_norm_bound,
                 robust_noise=a.robust_noise,
                 log_dir="/tmp/fedavg_classic")
    os.makedirs(cfg.log_dir, exist_ok=True)
    logger = MetricLogger(cfg.log_dir, enabled=comm.is_root, to_file=False)
    job = FLJob(cfg, comm, logger, dataset=ds)
    job.run()
    if comm.is_root:
        s = logger.series("Test/Acc")
        print(f"final Test/Acc = {s[-1]:.4f} "
              f"(best {max(s):.4f} over {len(s)} rounds)")


if __name__ == "__main__":
    main()
