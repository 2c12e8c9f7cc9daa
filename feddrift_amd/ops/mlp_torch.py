"""Vectorized torch implementation of the batched MLP ops.

Semantics mirror the reference's per-model eager loops
(FedAvgEnsTrainer.py:65-85: forward, CrossEntropyLoss, backward,
SGD/Adam(amsgrad, wd) step) but batched over all (client, model) pairs at
once. This module is the numerics ground truth the HIP kernels are tested
against, and the CPU execution path.

All tensors live on one device; nothing moves host<->device inside a round.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn.functional as F

from ..models.packed import MLPSpec

ADAM_B1 = 0.9
ADAM_B2 = 0.999
ADAM_EPS = 1e-8


def make_opt_state(kind: str, n_pairs: int, n_params: int, lr: float,
                   wd: float, device) -> Dict:
    st = {"kind": kind, "lr": torch.full((n_pairs,), lr, device=device),
          "wd": wd}
    if kind == "adam":
        st["m"] = torch.zeros(n_pairs, n_params, device=device)
        st["v"] = torch.zeros(n_pairs, n_params, device=device)
        st["vmax"] = torch.zeros(n_pairs, n_params, device=device)
        st["t"] = torch.zeros(n_pairs, dtype=torch.int32, device=device)
    return st


def _forward_fnn(spec: MLPSpec, params: torch.Tensor, x: torch.Tensor):
    """params [G,P], x [G,B,D] -> (z1, a, z2)."""
    G = params.shape[0]
    w1 = params[:, spec.off_w1:spec.off_b1].reshape(G, spec.h, spec.d)
    b1 = params[:, spec.off_b1:spec.off_w2].reshape(G, 1, spec.h)
    w2 = params[:, spec.off_w2:spec.off_b2].reshape(G, spec.o, spec.h)
    b2 = params[:, spec.off_b2:].reshape(G, 1, spec.o)
    z1 = torch.baddbmm(b1, x, w1.transpose(1, 2))
    a = torch.relu(z1)
    z2 = torch.baddbmm(b2, a, w2.transpose(1, 2))
    return z1, a, z2


def _forward_lr(spec: MLPSpec, params: torch.Tensor, x: torch.Tensor):
    G = params.shape[0]
    w = params[:, : spec.o * spec.d].reshape(G, spec.o, spec.d)
    b = params[:, spec.o * spec.d:].reshape(G, 1, spec.o)
    z = torch.baddbmm(b, x, w.transpose(1, 2))
    p = torch.sigmoid(z)
    return z, p


def forward_logits(spec: MLPSpec, params: torch.Tensor, x: torch.Tensor):
    """Model output as fed to CrossEntropyLoss / argmax by the reference.

    For 'lr' that is sigmoid(linear) (reference lr.py:10) — kept as-is.
    """
    if spec.kind == "fnn":
        return _forward_fnn(spec, params, x)[2]
    return _forward_lr(spec, params, x)[1]


def _apply_update(kind: str, lr: torch.Tensor, wd: float, st: Dict,
                  params: torch.Tensor, grads: torch.Tensor):
    """In-place optimizer step on contiguous params [G,P].

    Adam matches torch.optim.Adam(amsgrad=True, weight_decay=wd) exactly.
    st holds the CONTIGUOUS gathered state {m, v, vmax, t} for these rows.
    """
    lr = lr.unsqueeze(-1)
    if kind == "sgd":
        params -= lr * grads
        return
    g = grads + wd * params
    st["t"] += 1
    t = st["t"]
    st["m"].mul_(ADAM_B1).add_(g, alpha=1 - ADAM_B1)
    st["v"].mul_(ADAM_B2).addcmul_(g, g, value=1 - ADAM_B2)
    torch.maximum(st["vmax"], st["v"], out=st["vmax"])
    tf = t.float().unsqueeze(-1)
    bc1 = 1 - torch.pow(torch.tensor(ADAM_B1, device=params.device), tf)
    bc2 = 1 - torch.pow(torch.tensor(ADAM_B2, device=params.device), tf)
    denom = (st["vmax"] / bc2).sqrt_().add_(ADAM_EPS)
    params -= lr * (st["m"] / bc1) / denom


def train_fused(spec: MLPSpec,
                params_all: torch.Tensor,      # [R, P] fp32, updated in place
                rows: torch.Tensor,            # [G] int64 rows that train
                x_arena: torch.Tensor,         # [N, D] fp32
                y_arena: torch.Tensor,         # [N] int64
                step_off: torch.Tensor,        # [G, E] int64 window starts
                step_len: torch.Tensor,        # [G, E] int64 window lengths
                opt: Dict,
                x_mask: Optional[torch.Tensor] = None  # [G, D] input mask
                ) -> None:
    """E local optimizer steps per pair, each on its own minibatch window.

    Mirrors FedAvgEnsTrainer.py:65-85 / FedAvgEnsTrainerSoftCluster.py:91-113
    (the host picks the random batch windows; see engine/fljob.py).
    opt holds full per-row buffers {kind, lr[R], wd, m/v/vmax[R,P], t[R]}.
    A zero step_len means the step is SKIPPED for that pair (no optimizer
    state advance — reference FedAvgEnsTrainerExp.py:73-74 `continue`).
    x_mask (KUE feature masks, FedAvgEnsTrainerKue.py:65-97) multiplies the
    inputs elementwise before the forward pass.
    """
    G, E = step_off.shape
    if G == 0:
        return
    dev = params_all.device
    params = params_all[rows]
    if opt["kind"] == "adam":
        st = {"m": opt["m"][rows], "v": opt["v"][rows],
              "vmax": opt["vmax"][rows], "t": opt["t"][rows]}
    else:
        st = {}
    lr = opt["lr"][rows]
    for e in range(E):
        off = step_off[:, e]
        ln = step_len[:, e]
        bmax = int(ln.max().item())
        if bmax == 0:
            continue
        ar = torch.arange(bmax, device=dev).unsqueeze(0)        # [1,B]
        mask = ar < ln.unsqueeze(1)                             # [G,B]
        idx = off.unsqueeze(1) + torch.minimum(
            ar, (ln - 1).clamp(min=0).unsqueeze(1))             # [G,B]
        x = x_arena[idx.reshape(-1)].reshape(G, bmax, spec.d)
        y = y_arena[idx.reshape(-1)].reshape(G, bmax)
        if x_mask is not None:
            x = x * x_mask.unsqueeze(1)

        if spec.kind == "fnn":
            z1, a, z2 = _forward_fnn(spec, params, x)
            dz2 = torch.softmax(z2, dim=-1)
            dz2.scatter_add_(-1, y.unsqueeze(-1),
                             -torch.ones_like(dz2[..., :1]))
            dz2 = dz2 * (mask.unsqueeze(-1) / ln.reshape(G, 1, 1))
            G_ = params.shape[0]
            w2 = params[:, spec.off_w2:spec.off_b2].reshape(G_, spec.o, spec.h)
            dw2 = torch.bmm(dz2.transpose(1, 2), a)             # [G,O,H]
            db2 = dz2.sum(dim=1)                                # [G,O]
            da = torch.bmm(dz2, w2)                             # [G,B,H]
            dz1 = da * (z1 > 0)
            dw1 = torch.bmm(dz1.transpose(1, 2), x)             # [G,H,D]
            db1 = dz1.sum(dim=1)                                # [G,H]
            grads = torch.cat([dw1.reshape(G_, -1), db1,
                               dw2.reshape(G_, -1), db2], dim=1)
        else:
            z, p = _forward_lr(spec, params, x)
            dp = torch.softmax(p, dim=-1)
            dp.scatter_add_(-1, y.unsqueeze(-1),
                            -torch.ones_like(dp[..., :1]))
            dp = dp * (mask.unsqueeze(-1) / ln.reshape(G, 1, 1))
            dz = dp * p * (1 - p)
            dw = torch.bmm(dz.transpose(1, 2), x)
            db = dz.sum(dim=1)
            grads = torch.cat([dw.reshape(G, -1), db], dim=1)

        sel = ln > 0
        if bool(sel.all()):
            _apply_update(opt["kind"], lr, opt.get("wd", 0.0), st, params,
                          grads)
        else:
            sub = sel.nonzero(as_tuple=True)[0]
            p_sub = params[sub]
            st_sub = {k: v[sub] for k, v in st.items()}
            _apply_update(opt["kind"], lr[sub], opt.get("wd", 0.0), st_sub,
                          p_sub, grads[sub])
            params[sub] = p_sub
            for k in st:
                st[k][sub] = st_sub[k]

    # scatter the trained rows and optimizer state back
    params_all[rows] = params
    if opt["kind"] == "adam":
        opt["m"][rows] = st["m"]
        opt["v"][rows] = st["v"]
        opt["vmax"][rows] = st["vmax"]
        opt["t"][rows] = st["t"]


@torch.no_grad()
def eval_tasks(spec: MLPSpec,
               params: torch.Tensor,           # [M, P]
               x_arena: torch.Tensor, y_arena: torch.Tensor,
               task_row: torch.Tensor,         # [W] int64: params row per window
               task_id: torch.Tensor,          # [W] int64: output slot per window
               win_off: torch.Tensor,          # [W] int64
               win_len: torch.Tensor,          # [W] int64
               n_tasks: int,
               want_mse: bool = False,
               x_mask: Optional[torch.Tensor] = None):
    """Accuracy / summed CE loss (/ summed AUE MSE) per task.

    Equivalent to the reference _infer loops
    (FedAvgEnsAggregatorSoftCluster.py:305-330, FedAvgEnsDataLoader.py:1087-1108,
    FedAvgEnsAggregatorAue.py:_mse) with windows evaluated in one batched
    pass and accumulated per task.
    """
    dev = params.device
    correct = torch.zeros(n_tasks, dtype=torch.float64, device=dev)
    total = torch.zeros(n_tasks, dtype=torch.float64, device=dev)
    loss_sum = torch.zeros(n_tasks, dtype=torch.float64, device=dev)
    mse_sum = torch.zeros(n_tasks, dtype=torch.float64, device=dev) \
        if want_mse else None
    W = task_row.shape[0]
    if W == 0:
        return correct, total, loss_sum, mse_sum

    bmax = int(win_len.max().item())
    ar = torch.arange(bmax, device=dev).unsqueeze(0)
    mask = ar < win_len.unsqueeze(1)                            # [W,B]
    idx = win_off.unsqueeze(1) + torch.minimum(
        ar, (win_len - 1).clamp(min=0).unsqueeze(1))
    x = x_arena[idx.reshape(-1)].reshape(W, bmax, spec.d)
    y = y_arena[idx.reshape(-1)].reshape(W, bmax)
    if x_mask is not None:
        x = x * (x_mask.unsqueeze(1) if x_mask.dim() == 2 else x_mask)

    p = params[task_row]                                        # [W,P]
    logits = forward_logits(spec, p, x)                         # [W,B,O]
    pred = logits.argmax(dim=-1)
    corr = ((pred == y) & mask).sum(dim=1).double()
    # per-sample CE, summed (reference sums loss.item()*batch_size of means)
    ce = F.cross_entropy(logits.reshape(-1, spec.o), y.reshape(-1),
                         reduction="none").reshape(W, bmax)
    lsum = (ce * mask).sum(dim=1).double()
    correct.scatter_add_(0, task_id, corr)
    total.scatter_add_(0, task_id, mask.sum(dim=1).double())
    loss_sum.scatter_add_(0, task_id, lsum)
    if want_mse:
        prob = torch.softmax(logits, dim=-1)
        ptrue = prob.gather(-1, y.unsqueeze(-1)).squeeze(-1)
        ms = (((1.0 - ptrue) ** 2) * mask).sum(dim=1).double()
        mse_sum.scatter_add_(0, task_id, ms)
    return correct, total, loss_sum, mse_sum


@torch.no_grad()
def ens_vote_eval(spec: MLPSpec,
                  params: torch.Tensor,        # [M, P]
                  weights: torch.Tensor,       # [M] (zero => excluded)
                  x_arena: torch.Tensor, y_arena: torch.Tensor,
                  windows, mode: str = "hard",
                  masks: Optional[torch.Tensor] = None):
    """Weighted-vote ensemble accuracy over one client's windows.

    mode='hard': each model adds `weight` to its argmax class
    (AUE, FedAvgEnsAggregatorAue.py:256-283).
    mode='soft': votes += weight * softmax(logits), with optional per-model
    feature masks (KUE, FedAvgEnsAggregatorKue.py:234-264).
    Returns (correct, total) floats.
    """
    dev = params.device
    M = params.shape[0]
    correct = 0.0
    total = 0.0
    for off, ln in windows:
        if ln <= 0:
            continue
        x = x_arena[off:off + ln]                        # [B, D]
        y = y_arena[off:off + ln]
        votes = torch.zeros(ln, spec.o, device=dev)
        for m in range(M):
            w = float(weights[m].item())
            if w == 0.0:
                continue
            xm = x * masks[m] if masks is not None else x
            logits = forward_logits(spec, params[m:m + 1],
                                    xm.unsqueeze(0)).squeeze(0)
            if mode == "hard":
                pred = logits.argmax(dim=-1)
                votes.scatter_add_(
                    1, pred.unsqueeze(1),
                    torch.full((ln, 1), w, device=dev))
            else:
                votes += w * torch.softmax(logits, dim=-1)
        overall = votes.argmax(dim=-1)
        correct += float((overall == y).sum().item())
        total += float(ln)
    return correct, total


@torch.no_grad()
def confusion_tasks(spec: MLPSpec,
                    params: torch.Tensor,
                    x_arena: torch.Tensor, y_arena: torch.Tensor,
                    task_row: torch.Tensor, task_id: torch.Tensor,
                    win_off: torch.Tensor, win_len: torch.Tensor,
                    n_tasks: int, n_classes: int,
                    x_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Per-task confusion matrices A[y, y_hat]
    (KUE kappa weights, FedAvgEnsAggregatorKue.py:266-303)."""
    dev = params.device
    A = torch.zeros(n_tasks, n_classes, n_classes, dtype=torch.float64,
                    device=dev)
    W = task_row.shape[0]
    if W == 0:
        return A
    bmax = int(win_len.max().item())
    ar = torch.arange(bmax, device=dev).unsqueeze(0)
    mask = ar < win_len.unsqueeze(1)
    idx = win_off.unsqueeze(1) + torch.minimum(
        ar, (win_len - 1).clamp(min=0).unsqueeze(1))
    x = x_arena[idx.reshape(-1)].reshape(W, bmax, spec.d)
    y = y_arena[idx.reshape(-1)].reshape(W, bmax)
    if x_mask is not None:
        x = x * (x_mask.unsqueeze(1) if x_mask.dim() == 2 else x_mask)
    logits = forward_logits(spec, params[task_row], x)
    pred = logits.argmax(dim=-1)
    flat = (task_id.unsqueeze(1) * n_classes * n_classes +
            y * n_classes + pred)[mask]
    A.reshape(-1).scatter_add_(
        0, flat.reshape(-1),
        torch.ones_like(flat, dtype=torch.float64).reshape(-1))
    return A


@torch.no_grad()
def eval_tasks_stacked(spec: MLPSpec, params: torch.Tensor,
                       x_arena: torch.Tensor, y_arena: torch.Tensor,
                       task_row: torch.Tensor, task_id: torch.Tensor,
                       win_off: torch.Tensor, win_len: torch.Tensor,
                       n_tasks: int, want_mse: bool = False,
                       x_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
    """[3 or 4, n_tasks] float64 stacked result (see mlp_hip)."""
    c, t, l, m = eval_tasks(spec, params, x_arena, y_arena, task_row,
                            task_id, win_off, win_len, n_tasks,
                            want_mse=want_mse, x_mask=x_mask)
    parts = [c, t, l] + ([m] if m is not None else [])
    return torch.stack(parts)


@torch.no_grad()
def ens_vote_multi(spec: MLPSpec,
                   params: torch.Tensor,        # [M, P]
                   weights: torch.Tensor,       # [M] or [T, M]
                   x_arena: torch.Tensor, y_arena: torch.Tensor,
                   task_id: torch.Tensor,       # [W]
                   win_off: torch.Tensor, win_len: torch.Tensor,
                   n_tasks: int, mode: str = "hard",
                   masks: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Batched weighted-vote ensemble accuracy over MANY tasks at once
    (replaces per-client _infer_ens loops: FedAvgEnsAggregatorAue.py:256,
    AuePc per-client weights :260, Kue soft vote :234). weights may be
    per-task [T, M] (AUE-PC). Returns [2, n_tasks] = (correct, total)."""
    dev = params.device
    out = torch.zeros(2, n_tasks, dtype=torch.float64, device=dev)
    W = task_id.shape[0]
    if W == 0:
        return out
    M = params.shape[0]
    bmax = int(win_len.max().item())
    ar = torch.arange(bmax, device=dev).unsqueeze(0)
    smask = ar < win_len.unsqueeze(1)
    idx = win_off.unsqueeze(1) + torch.minimum(
        ar, (win_len - 1).clamp(min=0).unsqueeze(1))
    x = x_arena[idx.reshape(-1)].reshape(W, bmax, spec.d)
    y = y_arena[idx.reshape(-1)].reshape(W, bmax)
    if weights.dim() == 1:
        wt = weights.unsqueeze(0).expand(W, M)
    else:
        wt = weights[task_id]                       # [W, M]
    votes = torch.zeros(W, bmax, spec.o, device=dev)
    for m in range(M):
        wm = wt[:, m]
        if float(wm.abs().max()) == 0.0:
            continue
        xm = x * masks[m] if masks is not None else x
        logits = forward_logits(spec, params[m:m + 1].expand(W, -1), xm)
        if mode == "hard":
            votes.scatter_add_(
                2, logits.argmax(-1, keepdim=True),
                wm.reshape(W, 1, 1).expand(W, bmax, 1).contiguous())
        else:
            votes += wm.reshape(W, 1, 1) * torch.softmax(logits, -1)
    corr = ((votes.argmax(-1) == y) & smask).sum(dim=1).double()
    out[0].scatter_add_(0, task_id, corr)
    out[1].scatter_add_(0, task_id, smask.sum(dim=1).double())
    return out
