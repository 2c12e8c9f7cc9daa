#!/usr/bin/env python3
"""Localize the Adam-step mismatch: print epoch, layer, and the full
update computation at the worst element."""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.engine.fljob import TrainPlan
from feddrift_amd.models import zoo
from feddrift_amd.models.generic_packer import ModulePacker
from feddrift_amd.ops.cnn_hip import CnnHipEngine
from feddrift_amd.ops.mlp_torch import _apply_update

DEV = torch.device("cuda:0")
torch.manual_seed(0)
rng0 = np.random.default_rng(0)
proto = zoo.CNN_DropOut()
packer = ModulePacker(proto)
P = packer.n_params
K, G, E = 3, 6, 1
gp = torch.randn(K, P, device=DEV) * 0.05
N = 400
x = torch.randn(N, 784, device=DEV)
y = torch.from_numpy(rng0.integers(0, 10, N)).to(DEV)
rows = np.arange(G, dtype=np.int64)
plan0 = TrainPlan(rows, np.zeros((G, 1), np.int64),
                  np.ones((G, 1), np.int64), np.ones((2, K), np.float32))

hipE = CnnHipEngine(proto, packer, DEV)
hipE.dropout_override = (0.0, 0.0)
lr, wd = 0.03, 1e-3
opt = hipE.make_opt_state("adam", G, lr, wd)
st_ref = {"m": torch.zeros(G, P, device=DEV),
          "v": torch.zeros(G, P, device=DEV),
          "vmax": torch.zeros(G, P, device=DEV),
          "t": torch.zeros(G, dtype=torch.int32, device=DEV)}
rng = np.random.default_rng(123)
gp_cur = gp.clone()
rows_t = torch.as_tensor(rows, device=DEV)
lr_t = torch.full((G,), lr, device=DEV)
for epoch in range(3):
    step_off = rng.integers(0, N - 8, (G, 1)).astype(np.int64)
    step_len = rng.integers(1, 9, (G, 1)).astype(np.int64)
    eplan = TrainPlan(rows, step_off, step_len, plan0.sample_num)
    reps_a = torch.zeros(G, P, device=DEV)
    m_before = opt["m"].clone()
    v_before = opt["v"].clone()
    vm_before = opt["vmax"].clone()
    hipE.train(gp_cur.clone(), reps_a, eplan, opt, x, y, K)
    torch.cuda.synchronize()
    g_exact = hipE._ws["grad"][:G].clone()
    w_ref = gp_cur[rows_t % K].clone()
    w_before = w_ref.clone()
    _apply_update("adam", lr_t, wd, st_ref, w_ref, g_exact)
    d = (reps_a - w_ref).abs()
    err = d.max().item()
    gi, pi = np.unravel_index(int(d.argmax().cpu()), d.shape)
    i = 0
    lname = "?"
    for kk, nn in zip(packer.keys, packer.numels):
        if i <= pi < i + nn:
            lname = kk
            break
        i += nn
    print(f"epoch {epoch}: err={err:.3e} at pair {gi} {lname}[{pi - i}]")
    print(f"  w_before={w_before[gi, pi].item():+.9e}")
    print(f"  g={g_exact[gi, pi].item():+.9e}")
    print(f"  m_bef(hip)={m_before[gi, pi].item():+.9e} "
          f"v_bef={v_before[gi, pi].item():+.9e} "
          f"vm_bef={vm_before[gi, pi].item():+.9e}")
    print(f"  hip: m={opt['m'][gi, pi].item():+.9e} "
          f"v={opt['v'][gi, pi].item():+.9e} "
          f"vmax={opt['vmax'][gi, pi].item():+.9e} "
          f"w={reps_a[gi, pi].item():+.9e} t={opt['t'][gi].item()}")
    print(f"  ref: m={st_ref['m'][gi, pi].item():+.9e} "
          f"v={st_ref['v'][gi, pi].item():+.9e} "
          f"vmax={st_ref['vmax'][gi, pi].item():+.9e} "
          f"w={w_ref[gi, pi].item():+.9e} t={st_ref['t'][gi].item()}")
    md = (opt["m"] - st_ref["m"]).abs().max().item()
    vd = (opt["v"] - st_ref["v"]).abs().max().item()
    print(f"  max m diff={md:.3e} v diff={vd:.3e}")
    gp_cur = reps_a[:K].clone()
