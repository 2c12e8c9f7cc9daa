#!/usr/bin/env python3
"""Microbenchmark of the fused train kernel: time vs (G pairs, E steps,
batch n) to localize launch overhead vs per-step vs per-sample cost."""

import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.models.packed import spec_for
from feddrift_amd.ops import mlp_hip, mlp_torch


def bench(spec, G, E, n, iters=200):
    dev = torch.device("cuda:0")
    N = 300000
    x = torch.rand(N, spec.d, device=dev) * 10
    y = torch.randint(0, spec.o, (N,), device=dev)
    K = max(1, G // 10)
    glob = torch.randn(K, spec.n_params, device=dev) * 0.3
    rows = torch.arange(G, device=dev)
    model_of = (rows % K).to(torch.int32)
    sw = torch.ones(G, device=dev)
    off = torch.randint(0, N - n - 1, (G, E), device=dev)
    ln = torch.full((G, E), n, device=dev)
    reps = torch.zeros(G, spec.n_params, device=dev)
    opt = mlp_torch.make_opt_state("adam", G, spec.n_params, 0.01, 0.001,
                                   dev)
    partial = torch.zeros(K, spec.n_params + 1, device=dev)

    def run():
        partial.zero_()
        mlp_hip.train_fused(spec, reps, rows, x, y, off, ln, opt,
                            in_params=glob, model_of=model_of, sample_w=sw,
                            partial=partial)

    for _ in range(20):
        run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        run()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6   # us


def main():
    spec = spec_for("fnn", 3, 2)
    out = {}
    for G, E, n in [(10, 5, 500), (10, 1, 500), (10, 5, 100), (10, 5, 64),
                    (1, 5, 500), (100, 5, 500), (1000, 5, 500),
                    (10000, 5, 500), (10, 10, 500)]:
        us = bench(spec, G, E, n)
        out[f"G{G}_E{E}_n{n}"] = round(us, 1)
        print(f"G={G:6d} E={E:2d} n={n:4d}: {us:9.1f} us/launch")
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/microbench_train.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
