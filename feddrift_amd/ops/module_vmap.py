"""Vmap-batched module training: all (client, model) pairs in ONE batched
autograd step.

The per-pair training loop of ops/module_engine.py collapses into
torch.func.vmap(grad(...)) over stacked state: every pair's forward+
backward runs as grouped MIOpen convolutions in a single graph, and the
Adam(amsgrad, wd) update applies to the whole [G, Pp] parameter block at
once (the same update math as ops/mlp_torch.py — exact reference
semantics). Dropout uses randomness='different' (independent masks per
pair, as the reference's per-process training would draw).

Covers BOTH buffer-free modules (CNN_DropOut) and BatchNorm models
(ResNet / MobileNet / DenseNet): the flat row is the FULL state_dict
(generic_packer.py), parameters and buffers are exposed to
functional_call as separate batched pytrees, and train-mode
F.batch_norm updates each pair's running stats IN PLACE through the
batched buffer views — per-model stats evolve exactly as the
sequential engine's eager forwards would (verified by parity tests).
The optimizer update applies only to the parameter positions of the
flat row (buffers move only through BN's own momentum updates and
through FedAvg aggregation, which averages full state_dicts like the
reference — FedAvgEnsAggregatorSoftCluster.py:174-185)."""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn.functional as F
from torch import nn
from torch.func import functional_call, grad, vmap

from ..models.generic_packer import ModulePacker
from .mlp_torch import _apply_update

_BN_BUFFERS = ("running_mean", "running_var", "num_batches_tracked")


def vmap_compatible(module: nn.Module) -> bool:
    """True when the module can run on the vmap engine: either buffer-free
    (state_dict == parameters) or every buffer is a BatchNorm running
    statistic (updated in place through batched views under vmap).
    Recurrent modules are excluded: aten::lstm has no vmap batching rule
    on the inference path (RuntimeError under the eval sweep), so RNNs
    run on the sequential engine."""
    for m in module.modules():
        if isinstance(m, nn.RNNBase):
            return False
    for name, _ in module.named_buffers():
        if not name.rsplit(".", 1)[-1] in _BN_BUFFERS:
            return False
    return True


class VmapEngine:
    def __init__(self, template: nn.Module, packer: ModulePacker,
                 device: torch.device):
        assert vmap_compatible(template), "non-BN buffers need ModuleEngine"
        self.module = template.to(device)
        self.packer = packer
        self.device = device
        # flat layout follows packer.keys (full state_dict order); split
        # each key into trainable-parameter vs buffer role
        pnames = {n for n, _ in self.module.named_parameters()}
        self.keys = packer.keys
        self.shapes = [packer.shapes[k] for k in self.keys]
        self.numels = list(packer.numels)
        self.is_param = [k in pnames for k in self.keys]
        self.pnames = [k for k in self.keys if k in pnames]
        self.bnames = [k for k in self.keys if k not in pnames]
        # flat positions of the trainable parameters (optimizer targets)
        idx, i = [], 0
        for k, n in zip(self.keys, self.numels):
            if k in pnames:
                idx.append(torch.arange(i, i + n))
            i += n
        self.param_idx = (torch.cat(idx) if idx else
                          torch.empty(0, dtype=torch.int64)).to(device)
        self.has_buffers = len(self.bnames) > 0

        def loss_fn(plist, blist, x, y, mask, inv_n):
            tensors = dict(zip(self.pnames + self.bnames,
                               list(plist) + list(blist)))
            logits = functional_call(self.module, tensors, (x,))
            ce = F.cross_entropy(logits, y, reduction="none")
            return (ce * mask).sum() * inv_n

        self._grad_fn = vmap(grad(loss_fn),
                             in_dims=(0, 0, 0, 0, 0, 0),
                             randomness="different")

        def fwd_fn(plist, blist, x):
            tensors = dict(zip(self.pnames + self.bnames,
                               list(plist) + list(blist)))
            return functional_call(self.module, tensors, (x,))

        self._fwd_fn = vmap(fwd_fn, in_dims=(0, 0, 0))

    def _views(self, flat: torch.Tensor):
        """[G, P] flat -> (param views, buffer views), zero copy, in
        state_dict order within each role."""
        pv, bv = [], []
        i = 0
        G = flat.shape[0]
        for shape, n, isp in zip(self.shapes, self.numels, self.is_param):
            v = flat[:, i:i + n].reshape(G, *shape)
            (pv if isp else bv).append(v)
            i += n
        return pv, bv

    def make_opt_state(self, kind: str, n_rows: int, lr: float, wd: float):
        # optimizer state covers trainable parameters only (like the
        # sequential engine; buffers are not optimizer targets)
        from .mlp_torch import make_opt_state
        return make_opt_state(kind, n_rows, self.packer.n_train_params,
                              lr, wd, self.device)

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
        work = global_params[model_of].clone()          # [G, P_full]
        pv, bv = self._views(work)
        # BN's in-place running-stat update must write through CONTIGUOUS
        # batched tensors (an update through a strided view into `work`
        # does not land); stage buffers out, copy back after the E steps
        bufs = [v.contiguous() for v in bv]
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
            # grads come back for the parameter pytree only; BN updates
            # its running stats through `bufs` in place during the forward
            zero = (ln == 0)
            saved = ([b[zero].clone() for b in bufs]
                     if (self.has_buffers and bool(zero.any())) else None)
            grads = self._grad_fn(pv, bufs, x, y, mask, inv_n)
            if saved is not None:
                # a zero-length step is SKIPPED by reference semantics:
                # undo the padded forward's running-stat update for those
                # pairs (their loss was masked; params stay via `sel`)
                for b, s in zip(bufs, saved):
                    b[zero] = s
            gflat = torch.cat([g.reshape(G, -1) for g in grads], dim=1)
            if self.has_buffers:
                wp = work[:, self.param_idx]
            else:
                wp = work
            sel = ln > 0
            if bool(sel.all()):
                _apply_update(opt["kind"], lr, opt.get("wd", 0.0), st,
                              wp, gflat)
            else:
                sub = sel.nonzero(as_tuple=True)[0]
                w_sub = wp[sub]
                st_sub = {k: v[sub] for k, v in st.items()}
                _apply_update(opt["kind"], lr[sub], opt.get("wd", 0.0),
                              st_sub, w_sub, gflat[sub])
                wp[sub] = w_sub
                for k in st:
                    st[k][sub] = st_sub[k]
            if self.has_buffers:
                work[:, self.param_idx] = wp
        if self.has_buffers:
            for v, b in zip(bv, bufs):       # stats back into the flat rows
                v.copy_(b)
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
            pv, bv = self._views(p)
            bv = [v.contiguous() for v in bv]
            logits = self._fwd_fn(pv, bv, x.reshape(Wc, bmax, -1))
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
