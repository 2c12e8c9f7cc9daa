"""GPU op module: hand-written CDNA4 HIP kernels for the hot paths.

Same API as ops.mlp_torch. train_fused and eval_tasks (the per-round hot
loops: all local training in one launch, all accuracy sweeps in one launch)
run on the HIP kernels in ops/hip/feddrift_kernels.hip; the cold paths
(ensemble-vote testing, KUE confusion matrices — a handful of calls per
iteration, batched across clients) run on torch GPU ops by design.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from ..models.packed import MLPSpec
from . import hip_loader, mlp_torch

make_opt_state = mlp_torch.make_opt_state
forward_logits = mlp_torch.forward_logits
ens_vote_eval = mlp_torch.ens_vote_eval

_KIND = {"fnn": 0, "lr": 1}

# The train kernel keeps weights+grads (and per-sample chunk state) in LDS;
# the eval kernel keeps weights in LDS. Models beyond these sizes are
# GEMM-shaped (e.g. the 784x62 FEMNIST tower) and run on torch GPU ops —
# batched rocBLAS GEMMs, the right tool for plain library matmuls on
# CDNA4 (a hand-rolled HBM-resident kernel variant measured 5.4x SLOWER
# than this path at 3400 clients and was removed). This is a documented
# size-based dispatch, not a silent fallback; the drift-path flagship
# models are far below the caps.
LDS_BUDGET_FLOATS = 150 * 1024 // 4
TRAIN_MAX_P = (LDS_BUDGET_FLOATS - 4096) // 2
EVAL_MAX_P = LDS_BUDGET_FLOATS - 64


def _fits_train(spec: MLPSpec) -> bool:
    return spec.n_params <= TRAIN_MAX_P


def _fits_eval(spec: MLPSpec) -> bool:
    return spec.n_params <= EVAL_MAX_P


def train_fused(spec: MLPSpec, params_all: torch.Tensor, rows: torch.Tensor,
                x_arena: torch.Tensor, y_arena: torch.Tensor,
                step_off: torch.Tensor, step_len: torch.Tensor, opt: Dict,
                x_mask: Optional[torch.Tensor] = None,
                in_params: Optional[torch.Tensor] = None,
                model_of: Optional[torch.Tensor] = None,
                sample_w: Optional[torch.Tensor] = None,
                partial: Optional[torch.Tensor] = None) -> None:
    """in_params/model_of: stage initial weights straight from the global
    model rows (fuses the round's model broadcast into the launch);
    sample_w/partial: fused weighted aggregation partial sums."""
    if rows.numel() == 0:
        return
    if not _fits_train(spec):
        if in_params is not None:
            params_all[rows] = in_params[model_of.long()]
        mlp_torch.train_fused(spec, params_all, rows, x_arena, y_arena,
                              step_off, step_len, opt, x_mask=x_mask)
        if partial is not None and sample_w is not None:
            P = params_all.shape[1]
            mo = model_of.long()
            partial.index_add_(
                0, mo,
                torch.cat([sample_w.unsqueeze(1) * params_all[rows],
                           sample_w.unsqueeze(1)], dim=1))
        return
    mod = hip_loader.load()
    adam = opt["kind"] == "adam"
    mod.train_fused(
        params_all, rows.contiguous(), x_arena, y_arena,
        step_off.contiguous(), step_len.contiguous(),
        spec.d, spec.h, spec.o, _KIND[spec.kind],
        x_mask.contiguous() if x_mask is not None else None,
        opt["m"] if adam else None,
        opt["v"] if adam else None,
        opt["vmax"] if adam else None,
        opt["t"] if adam else None,
        opt["lr"], float(opt.get("wd", 0.0)),
        in_params, model_of, sample_w, partial)


def apply_aggregate(global_params: torch.Tensor, partial: torch.Tensor,
                    mask: Optional[torch.Tensor] = None,
                    totals_out: Optional[torch.Tensor] = None
                    ) -> torch.Tensor:
    """Masked weighted-average update; ALSO zero-drains `partial` for the
    next round's fused accumulation and parks the per-model totals in
    totals_out (allocated if not given). Returns totals_out."""
    if totals_out is None:
        totals_out = torch.empty(global_params.shape[0],
                                 device=global_params.device)
    hip_loader.load().apply_aggregate(global_params, partial, mask,
                                      totals_out)
    return totals_out


def eval_tasks(spec: MLPSpec, params: torch.Tensor,
               x_arena: torch.Tensor, y_arena: torch.Tensor,
               task_row: torch.Tensor, task_id: torch.Tensor,
               win_off: torch.Tensor, win_len: torch.Tensor, n_tasks: int,
               want_mse: bool = False,
               x_mask: Optional[torch.Tensor] = None):
    # NOTE: callers that build task lists on the host chunk long windows
    # there (engine TaskList, cached per iteration) — more workgroups,
    # shorter latency chains, zero per-call device work.
    out = eval_tasks_stacked(spec, params, x_arena, y_arena, task_row,
                             task_id, win_off, win_len, n_tasks, want_mse,
                             x_mask)
    return out[0], out[1], out[2], (out[3] if want_mse else None)


def ens_vote_multi(spec: MLPSpec, params: torch.Tensor,
                   weights: torch.Tensor, x_arena: torch.Tensor,
                   y_arena: torch.Tensor, task_id: torch.Tensor,
                   win_off: torch.Tensor, win_len: torch.Tensor,
                   n_tasks: int, mode: str = "hard",
                   masks: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Batched weighted-vote ensemble accuracy in ONE kernel launch
    (AUE _infer_ens / AUE-PC per-client weights / KUE soft vote —
    FedAvgEnsAggregatorAue.py:256-283, Kue:234-264). weights [M] or
    per-task [T, M]."""
    if not _fits_eval(spec) or spec.o > 64:
        return mlp_torch.ens_vote_multi(
            spec, params, weights, x_arena, y_arena, task_id, win_off,
            win_len, n_tasks, mode=mode, masks=masks)
    mod = hip_loader.load()
    W = task_id.shape[0]
    M = params.shape[0]
    wt = weights.to(params.device).float()
    if wt.dim() == 1:
        wt = wt.unsqueeze(0).expand(W, M)
    else:
        wt = wt[task_id]
    return mod.ens_vote_multi(
        params.contiguous(), wt.contiguous(), x_arena, y_arena,
        task_id.contiguous(), win_off.contiguous(), win_len.contiguous(),
        n_tasks, spec.d, spec.h, spec.o, _KIND[spec.kind],
        0 if mode == "hard" else 1,
        masks.contiguous().float() if masks is not None else None)


def confusion_tasks(spec: MLPSpec, params: torch.Tensor,
                    x_arena: torch.Tensor, y_arena: torch.Tensor,
                    task_row: torch.Tensor, task_id: torch.Tensor,
                    win_off: torch.Tensor, win_len: torch.Tensor,
                    n_tasks: int, n_classes: int,
                    x_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Per-task confusion matrices in ONE kernel launch (KUE kappa,
    FedAvgEnsAggregatorKue.py:266-303)."""
    if not _fits_eval(spec) or n_classes != spec.o or spec.o > 64:
        return mlp_torch.confusion_tasks(
            spec, params, x_arena, y_arena, task_row, task_id, win_off,
            win_len, n_tasks, n_classes, x_mask=x_mask)
    mod = hip_loader.load()
    return mod.confusion_tasks(
        params.contiguous(), x_arena, y_arena, task_row.contiguous(),
        task_id.contiguous(), win_off.contiguous(), win_len.contiguous(),
        n_tasks, n_classes, spec.d, spec.h, spec.o, _KIND[spec.kind],
        x_mask.contiguous().float() if x_mask is not None else None)


def eval_tasks_stacked(spec: MLPSpec, params: torch.Tensor,
                       x_arena: torch.Tensor, y_arena: torch.Tensor,
                       task_row: torch.Tensor, task_id: torch.Tensor,
                       win_off: torch.Tensor, win_len: torch.Tensor,
                       n_tasks: int, want_mse: bool = False,
                       x_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
    """[3 or 4, n_tasks] float64: correct/total/loss(/mse) rows in ONE
    contiguous buffer (ready for all_reduce, no stack)."""
    if not _fits_eval(spec):
        return mlp_torch.eval_tasks_stacked(
            spec, params, x_arena, y_arena, task_row, task_id, win_off,
            win_len, n_tasks, want_mse=want_mse, x_mask=x_mask)
    mod = hip_loader.load()
    return mod.eval_tasks(
        params.contiguous(), x_arena, y_arena,
        task_row.contiguous(), task_id.contiguous(),
        win_off.contiguous(), win_len.contiguous(), n_tasks,
        spec.d, spec.h, spec.o, _KIND[spec.kind], want_mse,
        x_mask.contiguous() if x_mask is not None else None)
