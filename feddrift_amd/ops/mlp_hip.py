"""GPU op module: hand-written CDNA4 HIP kernels for the hot paths.

Same API as ops.mlp_torch. train_fused and eval_tasks (the per-round hot
loops: all local training in one launch, all accuracy sweeps in one launch)
run on the HIP kernels in ops/hip/feddrift_kernels.hip; the cold paths
(ensemble-vote testing, KUE confusion matrices — a few times per
iteration) currently run on torch GPU ops and will be ported next.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from ..models.packed import MLPSpec
from . import hip_loader, mlp_torch

make_opt_state = mlp_torch.make_opt_state
forward_logits = mlp_torch.forward_logits
ens_vote_eval = mlp_torch.ens_vote_eval
confusion_tasks = mlp_torch.confusion_tasks

_KIND = {"fnn": 0, "lr": 1}


def train_fused(spec: MLPSpec, params_all: torch.Tensor, rows: torch.Tensor,
                x_arena: torch.Tensor, y_arena: torch.Tensor,
                step_off: torch.Tensor, step_len: torch.Tensor, opt: Dict,
                x_mask: Optional[torch.Tensor] = None) -> None:
    if rows.numel() == 0:
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
        opt["lr"], float(opt.get("wd", 0.0)))


_EVAL_CHUNK = 128


def _chunk_windows(task_row, task_id, win_off, win_len, x_mask):
    """Split long windows into <=128-sample pieces: more workgroups, shorter
    per-thread latency chains (the kernel accumulates per task via
    atomicAdd, so chunking is free)."""
    if win_len.numel() == 0 or int(win_len.max()) <= _EVAL_CHUNK:
        return task_row, task_id, win_off, win_len, x_mask
    nch = (win_len + (_EVAL_CHUNK - 1)) // _EVAL_CHUNK
    task_row = task_row.repeat_interleave(nch)
    task_id = task_id.repeat_interleave(nch)
    base_off = win_off.repeat_interleave(nch)
    base_len = win_len.repeat_interleave(nch)
    if x_mask is not None and x_mask.dim() == 2:
        x_mask = x_mask.repeat_interleave(nch, dim=0)
    # position of each chunk within its window
    csum = torch.cumsum(nch, 0)
    start = torch.repeat_interleave(csum - nch, nch)
    pos = torch.arange(task_row.numel(), device=task_row.device) - start
    off = base_off + pos * _EVAL_CHUNK
    ln = torch.minimum(base_len - pos * _EVAL_CHUNK,
                       torch.full_like(base_len, _EVAL_CHUNK))
    return task_row, task_id, off, ln, x_mask


def eval_tasks(spec: MLPSpec, params: torch.Tensor,
               x_arena: torch.Tensor, y_arena: torch.Tensor,
               task_row: torch.Tensor, task_id: torch.Tensor,
               win_off: torch.Tensor, win_len: torch.Tensor, n_tasks: int,
               want_mse: bool = False,
               x_mask: Optional[torch.Tensor] = None):
    mod = hip_loader.load()
    task_row, task_id, win_off, win_len, x_mask = _chunk_windows(
        task_row, task_id, win_off, win_len, x_mask)
    correct, total, loss, mse = mod.eval_tasks(
        params.contiguous(), x_arena, y_arena,
        task_row.contiguous(), task_id.contiguous(),
        win_off.contiguous(), win_len.contiguous(), n_tasks,
        spec.d, spec.h, spec.o, _KIND[spec.kind], want_mse,
        x_mask.contiguous() if x_mask is not None else None)
    return correct, total, loss, (mse if want_mse else None)
