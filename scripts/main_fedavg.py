#!/usr/bin/env python3
"""Per-iteration entry (reference-style operation): runs ONE training
iteration, reading/writing the cross-iteration checkpoint files.

Equivalent of fedml_experiments/distributed/fedavg_cont_ens/main_fedavg.py,
launched once per --curr_train_iteration by the outer shell loop. Under
torchrun each rank drives one GPU (env RANK/WORLD_SIZE/LOCAL_RANK)."""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from feddrift_amd.config import config_from_argv
from feddrift_amd.engine.fljob import run_iteration
from feddrift_amd.engine.timeline import clean_state_files


def main():
    cfg = config_from_argv()
    comm = Communicator(backend=cfg.backend)
    if cfg.curr_train_iteration == 0 and comm.is_root:
        clean_state_files(cfg)
    comm.barrier()
    job = run_iteration(cfg, comm)
    if comm.is_root:
        s = job.logger.series("Test/Acc")
        avg = sum(s) / len(s) if s else float("nan")
        print(f"iteration {cfg.curr_train_iteration}: "
              f"avg Test/Acc over rounds = {avg:.4f}")


if __name__ == "__main__":
    main()
