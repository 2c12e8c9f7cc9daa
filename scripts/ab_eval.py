#!/usr/bin/env python3
"""Same-box A/B of the templated small-eval kernel vs the generic eval
kernel (FEDDRIFT_NO_SMALL_EVAL toggles the C++ dispatch).  Run twice in
one process is impossible (the toggle is read once per process), so this
script is launched per-arm by the driver shell loop; each arm prints one
line: arm, clients, rounds/s."""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.comm import Communicator
from scripts.bench_sweep import run_scale


def main():
    arm = "generic" if os.environ.get("FEDDRIFT_NO_SMALL_EVAL") == "1" \
        else "small"
    comm = Communicator()
    sizes = [(10, 600), (200, 300), (3400, 25)]
    if len(sys.argv) > 1:                     # "clients:steps,clients:steps"
        sizes = [tuple(int(v) for v in part.split(":"))
                 for part in sys.argv[1].split(",")]
    for c, steps in sizes:
        rps = run_scale(comm, c, steps=steps, warmup=max(10, steps // 5))
        print(f"AB {arm} clients={c} rps={rps:.1f}", flush=True)


if __name__ == "__main__":
    main()
