#!/usr/bin/env python3
"""Randomized edge-case soak for the CNN kernel pipeline on a GPU box:
odd fleet sizes, batch 1, zero-length steps, O=62, masks, both
optimizers, chunked pairs, eval windows of every length — each config
checked against the vmap engine (SGD exact; Adam bounded per the
eps-amplification analysis in tests/test_gpu_cnn.py)."""
import copy
import sys
import time

import numpy as np
import torch

sys.path.append(".")
from feddrift_amd.engine.fljob import TrainPlan
from feddrift_amd.models import zoo
from feddrift_amd.models.generic_packer import ModulePacker
from feddrift_amd.ops.cnn_hip import CnnHipEngine
from feddrift_amd.ops.module_vmap import VmapEngine

DEV = torch.device("cuda:0")
N_CONFIGS = int(sys.argv[1]) if len(sys.argv) > 1 else 16
SEED = int(sys.argv[2]) if len(sys.argv) > 2 else 20260914
rng = np.random.default_rng(SEED)
t0 = time.time()

for trial in range(N_CONFIGS):
    O = int(rng.choice([10, 62]))
    K = int(rng.integers(1, 4))
    nW = int(rng.integers(1, 4))
    G = nW * K
    E = int(rng.integers(1, 4))
    bmax = int(rng.choice([1, 2, 7, 64]))
    N = 300
    use_mask = bool(rng.random() < 0.3)
    opt_kind = "sgd" if rng.random() < 0.5 else "adam"
    torch.manual_seed(trial)
    proto = zoo.CNN_DropOut(only_digits=(O == 10))
    packer = ModulePacker(proto)
    P = packer.n_params
    gp = torch.randn(K, P, device=DEV) * 0.05
    x = torch.randn(N, 784, device=DEV)
    y = torch.from_numpy(rng.integers(0, O, N)).to(DEV)
    step_off = rng.integers(0, N - bmax, (G, E)).astype(np.int64)
    step_len = rng.integers(0, bmax + 1, (G, E)).astype(np.int64)
    if step_len.max() == 0:
        step_len[0, 0] = 1
    plan = TrainPlan(np.arange(G), step_off, step_len,
                     np.ones((nW, K), np.float32))
    xm = None
    if use_mask:
        xm = (torch.rand(G, 784, device=DEV) > 0.2).float()

    p1 = copy.deepcopy(proto)
    p2 = copy.deepcopy(proto)
    p2.dropout_1.p = 0.0
    p2.dropout_2.p = 0.0
    hipE = CnnHipEngine(p1, packer, DEV)
    hipE.dropout_override = (0.0, 0.0)
    if rng.random() < 0.3:
        hipE.WS_BUDGET = 1  # force pair chunking
    vmapE = VmapEngine(p2, packer, DEV)
    res = {}
    for name, eng in (("hip", hipE), ("vmap", vmapE)):
        reps = torch.zeros(G, P, device=DEV)
        opt = eng.make_opt_state(opt_kind, G, 0.02, 1e-3)
        eng.train(gp.clone(), reps, plan, opt, x, y, K, x_mask=xm)
        torch.cuda.synchronize()
        res[name] = reps.clone()
    d = (res["hip"] - res["vmap"]).abs()
    tol = 1e-4 if opt_kind == "sgd" else 0.15
    assert d.max().item() < tol, (trial, opt_kind, d.max().item())
    if opt_kind == "adam":
        assert d.mean().item() < 1e-4, (trial, d.mean().item())

    # eval sweep vs vmap
    Wn = int(rng.integers(1, 12))
    tr = torch.from_numpy(rng.integers(0, K, Wn)).to(DEV)
    ti = torch.from_numpy(rng.integers(0, 4, Wn)).to(DEV)
    ln = torch.from_numpy(rng.integers(1, 131, Wn)).to(DEV)
    off = torch.from_numpy(
        np.array([rng.integers(0, N - int(l)) for l in ln.cpu()])).to(DEV)
    hipE.EVAL_SLOT_BUDGET = int(rng.choice([7, 64, 8192]))
    a = hipE.eval_tasks_stacked(gp, tr, ti, off, ln, 4, want_mse=True,
                                x_arena=x, y_arena=y)
    b = vmapE.eval_tasks_stacked(gp, tr, ti, off, ln, 4, want_mse=True,
                                 x_arena=x, y_arena=y)
    torch.cuda.synchronize()
    assert torch.equal(a[0], b[0]) and torch.equal(a[1], b[1]), trial
    assert (a[2] - b[2]).abs().max().item() < 1e-2, trial

    print(f"trial {trial}: O={O} G={G} E={E} bmax={bmax} opt={opt_kind} "
          f"mask={use_mask} OK")

print(f"fuzz OK: {N_CONFIGS} configs in {time.time() - t0:.0f}s")
