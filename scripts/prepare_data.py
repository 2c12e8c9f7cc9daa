#!/usr/bin/env python3
"""Data preparation: per-(client, iteration) CSVs + change points.

Argument surface matches the reference prepare_data.py invocation
(run_fedavg_distributed_pytorch.sh:35-47)."""

import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.data.generators import generate_data


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--dataset", type=str, default="sea")
    p.add_argument("--data_dir", type=str, default="./data")
    p.add_argument("--sample_num", type=int, default=500)
    p.add_argument("--noise_prob", type=float, default=0.0)
    p.add_argument("--partition_method", type=str, default="homo")
    p.add_argument("--client_num_in_total", type=int, default=10)
    p.add_argument("--client_num_per_round", type=int, default=10)
    p.add_argument("--batch_size", type=int, default=500)
    p.add_argument("--train_iteration", type=int, default=10)
    p.add_argument("--drift_together", type=int, default=0)
    p.add_argument("--time_stretch", type=int, default=1)
    p.add_argument("--change_points", type=str, default="rand")
    p.add_argument("--dummy_arg", type=int, default=0)
    a = p.parse_args()

    np.random.seed(a.dummy_arg)

    # seed the canonical change-point matrices (A..F, W..Z, R0..R9) into the
    # data dir, like the reference's data/changepoints checkout
    import shutil
    repo_cp = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..",
                           "data", "changepoints")
    dst_cp = os.path.join(a.data_dir, "changepoints")
    os.makedirs(dst_cp, exist_ok=True)
    if os.path.isdir(repo_cp):
        for f in os.listdir(repo_cp):
            if f.endswith(".cp") and not os.path.exists(
                    os.path.join(dst_cp, f)):
                shutil.copy(os.path.join(repo_cp, f),
                            os.path.join(dst_cp, f))

    generate_data(a.dataset, a.data_dir, a.train_iteration,
                  a.client_num_in_total, a.drift_together, a.sample_num,
                  a.noise_prob, a.time_stretch, a.change_points)
    print(f"prepared {a.dataset} data under {a.data_dir}")


if __name__ == "__main__":
    main()
