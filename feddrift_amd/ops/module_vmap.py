"""Vmap-batched module training: all (client, model) pairs in ONE batched
autograd step.

For buffer-free modules (CNN_DropOut — convs, dropout, linears) the
per-pair training loop of ops/module_engine.py collapses into
torch.func.vmap(grad(...)) over stacked parameters: every pair's forward+
backward runs as grouped MIOpen convolutions in a single graph, and the
Adam(amsgrad, wd) update applies to the whole [G, P] block at once (the
same update math as ops/mlp_torch.py — exact reference semantics).
Dropout uses randomness='different' (independent masks per pair, as the
reference's per-process training would draw).

Modules with buffers (BatchNorm ResNets) keep the sequential ModuleEngine:
running-stat updates are stateful and outside vmap's functional model.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch
import torch.nn.functional as F
from torch import nn
from torch.func import functional_call, grad, vmap

from ..models.generic_packer import ModulePacker
from .mlp_torch import _apply_update


def vmap_compatible(module: nn.Module) -> bool:
    """True when state_dict == parameters (no buffers)."""
    return len(list(module.buffers())) == 0


class VmapEngine:
    def __init__(self, template: nn.Module, packer: ModulePacker,
                 device: torch.device):
        assert vmap_compatible(template), "buffers need ModuleEngine"
        self.module = template.to(device)
        self.packer = packer
        self.device = device
        self.names = [n for n, _ in self.module.named_parameters()]
        self.shapes = [p.shape for _, p in self.module.named_parameters()]
        self.numels = [p.numel() for _, p in self.module.named_parameters()]
        # flat layout must equal state_dict order (packer) — true when no
        # buffers exist and named_parameters follows state_dict order
        assert self.names == packer.keys

        def loss_fn(param_list, x, y, mask, inv_n):
            params = dict(zip(self.names, param_list))
            logits = functional_call(self.module, params, (x,))
            ce = F.cross_entropy(logits, y, reduction="none")
            return (ce * mask).sum() * inv_n

        self._grad_fn = vmap(grad(loss_fn),
                             in_dims=(0, 0, 0, 0, 0),
                             randomness="different")

        def fwd_fn(param_list, x):
            params = dict(zip(self.names, param_list))
            return functional_call(self.module, params, (x,))

        self._fwd_fn = vmap(fwd_fn, in_dims=(0, 0))

    def _param_views(self, flat: torch.Tensor):
        """[G, P] flat -> list of [G, *shape] views (zero copy)."""
        out = []
        i = 0
        G = flat.shape[0]
        for shape, n in zip(self.shapes, self.numels):
            out.append(flat[:, i:i + n].reshape(G, *shape))
            i += n
        return out

    def make_opt_state(self, kind: str, n_rows: int, lr: float, wd: float):
        # full-P state (no buffers, so every entry is trainable)
        from .mlp_torch import make_opt_state
        return make_opt_state(kind, n_rows, self.packer.n_params, lr, wd,
                              self.device)

    # activation memory bound: vmapped conv activations scale with
    # pairs x batch; chunk the pair dimension (pairs are independent, so
    # chunking is numerically exact)
    MAX_PAIRS = 256

    def train(self, global_params: torch.Tensor, replicas: torch.Tensor,
              plan, opt: Dict, x_arena: torch.Tensor, y_arena: torch.Tensor,
              n_models: int, x_mask: Optional[torch.Tensor] = None) -> None:
        n_pairs = len(plan.rows)
        if n_pairs > self.MAX_PAIRS:
            import dataclasses
            for g0 in range(0, n_pairs, self.MAX_PAIRS):
                sl = slice(g0, g0 + self.MAX_PAIRS)
                sub = dataclasses.replace(
                    plan, rows=plan.rows[sl], step_off=plan.step_off[sl],
                    step_len=plan.step_len[sl])
                self.train(global_params, replicas, sub, opt, x_arena,
                           y_arena, n_models,
                           x_mask=x_mask[sl] if x_mask is not None else None)
            return
        rows = torch.as_tensor(plan.rows, dtype=torch.int64,
                               device=self.device)
        G = rows.numel()
        if G == 0:
            return
        self.module.train()
        model_of = rows % n_models
        work = global_params[model_of].clone()          # [G, P]
        step_off = torch.as_tensor(plan.step_off, dtype=torch.int64,
                                   device=self.device)
        step_len = torch.as_tensor(plan.step_len, dtype=torch.int64,
                                   device=self.device)
        if opt["kind"] == "adam":
            st = {"m": opt["m"][rows], "v": opt["v"][rows],
                  "vmax": opt["vmax"][rows], "t": opt["t"][rows]}
        else:
            st = {}
        lr = opt["lr"][rows]
        E = step_off.shape[1]
        for e in range(E):
            off = step_off[:, e]
            ln = step_len[:, e]
            bmax = int(ln.max())
            if bmax == 0:
                continue
            ar = torch.arange(bmax, device=self.device).unsqueeze(0)
            mask = (ar < ln.unsqueeze(1)).float()
            idx = off.unsqueeze(1) + torch.minimum(
                ar, (ln - 1).clamp(min=0).unsqueeze(1))
            x = x_arena[idx.reshape(-1)].reshape(G, bmax, -1)
            if x_mask is not None:
                x = x * x_mask.unsqueeze(1)
            y = y_arena[idx.reshape(-1)].reshape(G, bmax)
            inv_n = 1.0 / ln.clamp(min=1).float()
            grads = self._grad_fn(self._param_views(work), x, y, mask,
                                  inv_n)
            gflat = torch.cat([g.reshape(G, -1) for g in grads], dim=1)
            sel = ln > 0
            if bool(sel.all()):
                _apply_update(opt["kind"], lr, opt.get("wd", 0.0), st,
                              work, gflat)
            else:
                sub = sel.nonzero(as_tuple=True)[0]
                w_sub = work[sub]
                st_sub = {k: v[sub] for k, v in st.items()}
                _apply_update(opt["kind"], lr[sub], opt.get("wd", 0.0),
                              st_sub, w_sub, gflat[sub])
                work[sub] = w_sub
                for k in st:
                    st[k][sub] = st_sub[k]
        replicas[rows] = work
        if opt["kind"] == "adam":
            for k in st:
                opt[k][rows] = st[k]

    @torch.no_grad()
    def eval_tasks_stacked(self, params: torch.Tensor, task_row, task_id,
                           win_off, win_len, n_tasks: int,
                           want_mse: bool = False, x_arena=None,
                           y_arena=None,
                           x_mask: Optional[torch.Tensor] = None,
                           max_samples: int = 16384) -> torch.Tensor:
        rowsn = 4 if want_mse else 3
        out = torch.zeros(rowsn, n_tasks, dtype=torch.float64,
                          device=self.device)
        W = task_row.numel()
        if W == 0:
            return out
        self.module.eval()
        # bound activation memory: windows per sweep so that
        # windows x batch <= max_samples
        bmax_all = int(win_len.max())
        max_windows = max(1, max_samples // max(1, bmax_all))
        for w0 in range(0, W, max_windows):
            tr = task_row[w0:w0 + max_windows]
            ti = task_id[w0:w0 + max_windows]
            wo = win_off[w0:w0 + max_windows]
            wl = win_len[w0:w0 + max_windows]
            Wc = tr.numel()
            bmax = int(wl.max())
            ar = torch.arange(bmax, device=self.device).unsqueeze(0)
            mask = ar < wl.unsqueeze(1)
            idx = wo.unsqueeze(1) + torch.minimum(
                ar, (wl - 1).clamp(min=0).unsqueeze(1))
            x = x_arena[idx.reshape(-1)].reshape(Wc, bmax, -1)
            if x_mask is not None:
                xm = x_mask[w0:w0 + max_windows]
                x = x * (xm.unsqueeze(1) if xm.dim() == 2 else xm)
            y = y_arena[idx.reshape(-1)].reshape(Wc, bmax)
            p = params[tr]
            logits = self._fwd_fn(self._param_views(p),
                                  x.reshape(Wc, bmax, -1))
            logits = logits.reshape(Wc, bmax, -1)
            pred = logits.argmax(-1)
            corr = ((pred == y) & mask).sum(dim=1).double()
            ce = F.cross_entropy(
                logits.reshape(Wc * bmax, -1), y.reshape(-1),
                reduction="none").reshape(Wc, bmax)
            out[0].scatter_add_(0, ti, corr)
            out[1].scatter_add_(0, ti, mask.sum(dim=1).double())
            out[2].scatter_add_(0, ti, (ce * mask).sum(dim=1).double())
            if want_mse:
                prob = torch.softmax(logits, -1)
                pt = prob.gather(-1, y.unsqueeze(-1)).squeeze(-1)
                out[3].scatter_add_(
                    0, ti, (((1 - pt) ** 2) * mask).sum(dim=1).double())
        return out

    # cold paths reuse the sequential engine's implementations
    def _seq(self):
        if not hasattr(self, "_seq_engine"):
            from .module_engine import ModuleEngine
            self._seq_engine = ModuleEngine(self.module, self.packer,
                                            self.device)
        return self._seq_engine

    def ens_vote_eval(self, *args, **kw):
        return self._seq().ens_vote_eval(*args, **kw)

    def confusion_tasks(self, *args, **kw):
        return self._seq().confusion_tasks(*args, **kw)
