#!/usr/bin/env python3
"""Localize CNN kernel gradient mismatches: one SGD step with lr=1 on a
single pair makes (param_before - param_after) == the gradient; compare
per-layer slices against torch autograd."""
import os
import sys

import numpy as np
import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from feddrift_amd.engine.fljob import TrainPlan
from feddrift_amd.models import zoo
from feddrift_amd.models.generic_packer import ModulePacker
from feddrift_amd.ops.cnn_hip import CnnHipEngine

DEV = torch.device("cuda:0")
torch.manual_seed(0)
rng = np.random.default_rng(0)

proto = zoo.CNN_DropOut()
packer = ModulePacker(proto)
P = packer.n_params
gp = (torch.randn(1, P, device=DEV) * 0.05)
N = 64
B = 6
x = torch.randn(N, 784, device=DEV)
y = torch.from_numpy(rng.integers(0, 10, N)).to(DEV)

rows = np.array([0], dtype=np.int64)
step_off = np.array([[0]], dtype=np.int64)
step_len = np.array([[B]], dtype=np.int64)
plan = TrainPlan(rows, step_off, step_len, np.ones((1, 1), np.float32))

eng = CnnHipEngine(proto, packer, DEV)
eng.dropout_override = (0.0, 0.0)
reps = torch.zeros(1, P, device=DEV)
opt = eng.make_opt_state("sgd", 1, 1.0, 0.0)
eng.train(gp.clone(), reps, plan, opt, x, y, 1)
torch.cuda.synchronize()
g_hip = (gp[0] - reps[0])

# autograd reference
import copy
mod = copy.deepcopy(proto).to(DEV)
mod.dropout_1.p = 0.0
mod.dropout_2.p = 0.0
packer.load_into(mod, gp[0])
mod.train()
loss = F.cross_entropy(mod(x[:B]), y[:B])
loss.backward()
g_ref = torch.cat([p.grad.reshape(-1) for _, p in mod.named_parameters()])

i = 0
for k, n in zip(packer.keys, packer.numels):
    d = (g_hip[i:i + n] - g_ref[i:i + n])
    rel = d.abs().max() / (g_ref[i:i + n].abs().max() + 1e-12)
    print(f"{k:20s} maxabs={d.abs().max().item():.3e} "
          f"ref_max={g_ref[i:i+n].abs().max().item():.3e} rel={rel:.3e}")
    i += n
print("TOTAL maxabs", (g_hip - g_ref).abs().max().item())

# also check the forward activations: run kernels' forward via eval dump
tr = torch.zeros(1, dtype=torch.int64, device=DEV)
ti = torch.zeros(1, dtype=torch.int64, device=DEV)
off = torch.zeros(1, dtype=torch.int64, device=DEV)
ln = torch.full((1,), B, dtype=torch.int64, device=DEV)
out_probs = []
def sink(start, end, wo, wl, slot, outp):
    out_probs.append(outp.clone())
eng._x_arena, eng._y_arena = x, y
eng._eval_sweep(gp, tr, ti, off, ln, 1, 2, dump_sink=sink)
torch.cuda.synchronize()
mod.eval()
with torch.no_grad():
    s_ref = mod(x[:B])
print("fwd probs maxabs", (out_probs[0] - s_ref).abs().max().item())
