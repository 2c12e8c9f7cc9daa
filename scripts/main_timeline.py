#!/usr/bin/env python3
"""Whole-timeline entry: all training iterations in one set of processes
(one per GPU). Checkpoint files are still written per iteration, so a run
can be resumed at iteration granularity with scripts/main_fedavg.py."""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import config_from_argv
from feddrift_amd.engine.timeline import run_timeline


def main():
    cfg = config_from_argv()
    comm = Communicator(backend=cfg.backend)
    out = run_timeline(cfg, comm)
    if comm.is_root:
        print(f"avg Test/Acc over drift timeline: {out['avg_test_acc']:.4f}")
        print("per-iteration:",
              [round(a, 4) for a in out["per_iteration_test_acc"]])


if __name__ == "__main__":
    main()
