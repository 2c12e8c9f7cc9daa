"""Module execution path: local training + eval sweeps for torch-module
models (CNN_DropOut, ResNet-18) on MIOpen/rocBLAS through torch.

Same round semantics as the fused MLP path (engine/fljob.py): per active
(worker, model) pair, E optimizer steps on randomly picked minibatch
windows; Adam(amsgrad, wd)/SGD state persists across rounds per pair
(reference FedAvgEnsTrainer.py:23-33 keeps one optimizer per (worker,
model) for the whole iteration). One live module instance per rank is
re-loaded from the flat rows — all model state stays in flat HBM tensors.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch
import torch.nn.functional as F
from torch import nn

from ..models.generic_packer import ModulePacker
from .mlp_torch import ADAM_B1, ADAM_B2, ADAM_EPS


class ModuleEngine:
    def __init__(self, template: nn.Module, packer: ModulePacker,
                 device: torch.device):
        self.module = template.to(device)
        self.packer = packer
        self.device = device
        self.params = [p for _, p in self.module.named_parameters()]

    # -- optimizer state over trainable parameters only ------------------
    def make_opt_state(self, kind: str, n_rows: int, lr: float, wd: float):
        pp = self.packer.n_train_params
        st = {"kind": kind, "lr": torch.full((n_rows,), lr,
                                             device=self.device), "wd": wd}
        if kind == "adam":
            st["m"] = torch.zeros(n_rows, pp, device=self.device)
            st["v"] = torch.zeros(n_rows, pp, device=self.device)
            st["vmax"] = torch.zeros(n_rows, pp, device=self.device)
            st["t"] = torch.zeros(n_rows, dtype=torch.int32,
                                  device=self.device)
        return st

    def _pvec(self) -> torch.Tensor:
        return torch.cat([p.detach().reshape(-1) for p in self.params])

    def _gvec(self) -> torch.Tensor:
        return torch.cat([
            (p.grad.reshape(-1) if p.grad is not None
             else torch.zeros(p.numel(), device=self.device))
            for p in self.params])

    @torch.no_grad()
    def _write_pvec(self, vec: torch.Tensor) -> None:
        i = 0
        for p in self.params:
            p.copy_(vec[i:i + p.numel()].reshape(p.shape))
            i += p.numel()

    def _step(self, opt: Dict, row: int, grads: torch.Tensor) -> None:
        pvec = self._pvec()
        lr = float(opt["lr"][row])
        if opt["kind"] == "sgd":
            pvec -= lr * grads
        else:
            g = grads + opt["wd"] * pvec
            opt["t"][row] += 1
            t = float(opt["t"][row])
            m = opt["m"][row]
            v = opt["v"][row]
            m.mul_(ADAM_B1).add_(g, alpha=1 - ADAM_B1)
            v.mul_(ADAM_B2).addcmul_(g, g, value=1 - ADAM_B2)
            vmax = torch.maximum(opt["vmax"][row], v)
            opt["vmax"][row] = vmax
            denom = (vmax / (1 - ADAM_B2 ** t)).sqrt().add_(ADAM_EPS)
            pvec -= lr * (m / (1 - ADAM_B1 ** t)) / denom
        self._write_pvec(pvec)

    # -- round phases ----------------------------------------------------
    def train(self, global_params: torch.Tensor, replicas: torch.Tensor,
              plan, opt: Dict, x_arena: torch.Tensor, y_arena: torch.Tensor,
              n_models: int, x_mask: Optional[torch.Tensor] = None) -> None:
        mod = self.module
        mod.train()
        for gi, row in enumerate(plan.rows):
            m_idx = int(row) % n_models
            self.packer.load_into(mod, global_params[m_idx])
            E = plan.step_off.shape[1]
            for e in range(E):
                n = int(plan.step_len[gi, e])
                if n == 0:
                    continue
                off = int(plan.step_off[gi, e])
                x = x_arena[off:off + n]
                if x_mask is not None:
                    x = x * x_mask[gi]
                y = y_arena[off:off + n]
                for p in self.params:
                    p.grad = None
                with torch.enable_grad():
                    loss = F.cross_entropy(mod(x), y)
                    loss.backward()
                self._step(opt, int(row), self._gvec())
            self.packer.dump_from(mod, replicas[int(row)])

    @torch.no_grad()
    def eval_tasks_stacked(self, params: torch.Tensor, task_row, task_id,
                           win_off, win_len, n_tasks: int,
                           want_mse: bool = False,
                           x_arena=None, y_arena=None,
                           x_mask: Optional[torch.Tensor] = None,
                           max_batch: int = 4096) -> torch.Tensor:
        rowsn = 4 if want_mse else 3
        out = torch.zeros(rowsn, n_tasks, dtype=torch.float64,
                          device=self.device)
        if task_row.numel() == 0:
            return out
        mod = self.module
        mod.eval()
        tr = task_row.cpu().numpy()
        ti = task_id.cpu().numpy()
        wo = win_off.cpu().numpy()
        wl = win_len.cpu().numpy()
        order = np.argsort(tr, kind="stable")
        cur_row = -1
        for w in order:
            r = int(tr[w])
            if r != cur_row:
                self.packer.load_into(mod, params[r])
                cur_row = r
            off, ln, tid = int(wo[w]), int(wl[w]), int(ti[w])
            for c0 in range(0, ln, max_batch):
                bc = min(max_batch, ln - c0)
                x = x_arena[off + c0:off + c0 + bc]
                if x_mask is not None:
                    x = x * x_mask[w]
                y = y_arena[off + c0:off + c0 + bc]
                logits = mod(x)
                pred = logits.argmax(-1)
                out[0, tid] += (pred == y).sum().double()
                out[1, tid] += float(bc)
                out[2, tid] += F.cross_entropy(
                    logits, y, reduction="sum").double()
                if want_mse:
                    prob = torch.softmax(logits, -1)
                    pt = prob.gather(1, y.unsqueeze(1)).squeeze(1)
                    out[3, tid] += (((1 - pt) ** 2).sum()).double()
        return out

    @torch.no_grad()
    def ens_vote_eval(self, params: torch.Tensor, weights: torch.Tensor,
                      x_arena, y_arena, windows, mode: str = "hard",
                      masks: Optional[torch.Tensor] = None):
        mod = self.module
        mod.eval()
        M = params.shape[0]
        correct = total = 0.0
        for off, ln in windows:
            if ln <= 0:
                continue
            x = x_arena[off:off + ln]
            y = y_arena[off:off + ln]
            votes = torch.zeros(ln, dtype=torch.float32, device=self.device)
            votes = None
            for m in range(M):
                wgt = float(weights[m])
                if wgt == 0.0:
                    continue
                self.packer.load_into(mod, params[m])
                xm = x * masks[m] if masks is not None else x
                logits = mod(xm)
                if votes is None:
                    votes = torch.zeros(ln, logits.shape[-1],
                                        device=self.device)
                if mode == "hard":
                    votes.scatter_add_(
                        1, logits.argmax(-1, keepdim=True),
                        torch.full((ln, 1), wgt, device=self.device))
                else:
                    votes += wgt * torch.softmax(logits, -1)
            if votes is not None:
                correct += float((votes.argmax(-1) == y).sum())
            total += float(ln)
        return correct, total

    @torch.no_grad()
    def confusion_tasks(self, params: torch.Tensor, x_arena, y_arena,
                        task_row, task_id, win_off, win_len, n_tasks: int,
                        n_classes: int,
                        x_mask: Optional[torch.Tensor] = None):
        A = torch.zeros(n_tasks, n_classes, n_classes, dtype=torch.float64,
                        device=self.device)
        mod = self.module
        mod.eval()
        for w in range(task_row.numel()):
            self.packer.load_into(mod, params[int(task_row[w])])
            off, ln = int(win_off[w]), int(win_len[w])
            x = x_arena[off:off + ln]
            if x_mask is not None:
                x = x * (x_mask[w] if x_mask.dim() == 2 else x_mask)
            y = y_arena[off:off + ln]
            pred = mod(x).argmax(-1)
            tid = int(task_id[w])
            flat = y * n_classes + pred
            A[tid].reshape(-1).scatter_add_(
                0, flat, torch.ones_like(flat, dtype=torch.float64))
        return A
