"""CnnHipEngine: hand-written CDNA4 kernel path for the CNN_DropOut model.

Same engine interface as ops.module_vmap.VmapEngine (train /
eval_tasks_stacked / make_opt_state / ens_vote_eval / confusion_tasks),
but every hot op runs on the kernels in ops/hip/cnn_kernels.hip instead of
MIOpen-through-vmap:

  * train: one fused multi-kernel epoch sequence batched over all
    (client, model) pairs — weights live in flat HBM rows, gradients use
    exclusive-owner writes, Adam(amsgrad, wd) is fused (reference
    semantics: fedml_api/distributed/fedavg_ens/FedAvgEnsTrainer.py:23-95
    with the cv/cnn.py:113-135 forward, conv activations ABSENT).
  * eval: two-stage batched sweep (per-sample conv stage with x1 in LDS,
    per-window fc stage amortizing the fc1 weight stream), with fused
    accuracy/CE/mse, confusion-matrix, and probability-dump tails.
  * ensemble vote (AUE/KUE): probability dump per model + batched torch
    vote accumulation — no per-task Python loop
    (FedAvgEnsAggregatorAue.py:256-283, FedAvgEnsAggregatorKue.py:234-264).

Dropout uses counter-based mask hashes (statistical parity with the
reference's torch RNG draws; dropout_override=(0, 0) disables it for the
GPU parity tests, which compare against the vmap/eager engines in fp32).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from . import hip_loader

# CNN_DropOut geometry (must match cnn_kernels.hip)
X1N = 32 * 26 * 26
Z2N = 64 * 24 * 24
NF = 9216
NH = 128
D_IN = 784

EV_ACC, EV_CONF, EV_DUMP = 0, 1, 2


def is_cnn_dropout(module: torch.nn.Module) -> bool:
    from ..models.zoo import CNN_DropOut
    return isinstance(module, CNN_DropOut)


class CnnHipEngine:
    """Flat-row CNN training/eval on the hand-written gfx950 kernels."""

    # workspace budget (bytes) for the per-epoch activation/grad arena;
    # pairs are chunked so the arena fits (chunks are numerically exact:
    # pairs are independent). Sized for 288 GB of HBM3E: bigger chunks =
    # fewer launch sequences per round at FEMNIST-scale fleets.
    WS_BUDGET = 24 << 30

    def __init__(self, template: torch.nn.Module, packer, device):
        assert is_cnn_dropout(template)
        self.module = template.to(device)   # kept for cold paths/debug
        self.packer = packer
        self.device = device
        self.O = template.linear_2.out_features
        self.P = packer.n_params
        self.mod = hip_loader.load()
        self.dropout_override: Optional[tuple] = None
        self._seed_counter = 0
        self._ws = None

    # -- optimizer state (same layout as the module engines) -------------
    def make_opt_state(self, kind: str, n_rows: int, lr: float, wd: float):
        from .mlp_torch import make_opt_state
        return make_opt_state(kind, n_rows, self.P, lr, wd, self.device)

    # -- training ---------------------------------------------------------
    def _dropout_ps(self):
        if self.dropout_override is not None:
            return self.dropout_override
        return (float(self.module.dropout_1.p), float(self.module.dropout_2.p))

    def _workspace(self, G: int, B: int):
        ws = self._ws
        if ws is not None and ws["G"] >= G and ws["B"] >= B:
            return ws
        dev = self.device
        f = lambda *shape: torch.empty(*shape, device=dev)  # noqa: E731
        ws = {
            "G": G, "B": B,
            "x1": f(G * B * X1N),
            "a2": f(G * B * NF),
            "pidx": torch.empty(G * B * NF, dtype=torch.uint8, device=dev),
            "z1": f(G * B * NH),
            "a1": f(G * B * NH),
            "dz2": f(G * B * 64),
            "dz1": f(G * B * NH),
            "da2": f(G * B * NF),
            "zz2": f(G * B * Z2N),
            "dx1": f(G * B * X1N),
            "c1part": f(G * B * 320),
            "wtf": f(G * 9 * 2048),
            "wtd": f(G * 9 * 2048),
            "z1part": f(G * 8 * B * NH),
            "w2ms": self._w2ms(G),
            "w2part": f(G * self._w2ms(G) * 9 * 2048),
            "b2part": f(G * B * 64),
            "grad": f(G, self.P),
        }
        self._ws = ws
        return ws

    @staticmethod
    def _w2ms(G: int) -> int:
        """conv2-wgrad m-split count: G*w2ms blocks at 2 blocks/CU need
        >= 512 resident slots plus queued tail to fill 256 CUs (env
        FEDDRIFT_W2MS overrides for A/B runs)."""
        import os
        ov = os.environ.get("FEDDRIFT_W2MS")
        if ov:
            return max(1, int(ov))
        return min(64, max(1, 1024 // max(G, 1)))

    def _chunk_pairs(self, B: int) -> int:
        per_pair = (2 * X1N + 3 * NF + Z2N + 2 * NH + 64) * B * 4 \
            + NF * B + self.P * 4
        return max(1, min(2048, int(self.WS_BUDGET // max(1, per_pair))))

    def train(self, global_params: torch.Tensor, replicas: torch.Tensor,
              plan, opt: Dict, x_arena: torch.Tensor, y_arena: torch.Tensor,
              n_models: int, x_mask: Optional[torch.Tensor] = None) -> None:
        n_pairs = len(plan.rows)
        if n_pairs == 0:
            return
        B = int(plan.step_len.max()) if plan.step_len.size else 0
        if B == 0:
            # nothing trains (reference skips n==0 batches) but the round
            # still BROADCASTS: replicas get the global model, like vmap
            rows = torch.as_tensor(plan.rows, dtype=torch.int64,
                                   device=self.device)
            replicas[rows] = global_params[rows % n_models]
            return
        chunk = self._chunk_pairs(B)
        if n_pairs > chunk:
            import dataclasses
            for g0 in range(0, n_pairs, chunk):
                sl = slice(g0, g0 + chunk)
                sub = dataclasses.replace(
                    plan, rows=plan.rows[sl], step_off=plan.step_off[sl],
                    step_len=plan.step_len[sl])
                self._train_chunk(global_params, replicas, sub, opt,
                                  x_arena, y_arena, n_models,
                                  x_mask[sl] if x_mask is not None else None,
                                  g_base=g0)
            return
        self._train_chunk(global_params, replicas, plan, opt, x_arena,
                          y_arena, n_models, x_mask, g_base=0)

    def _train_chunk(self, global_params, replicas, plan, opt, x_arena,
                     y_arena, n_models, x_mask, g_base: int) -> None:
        dev = self.device
        rows = torch.as_tensor(plan.rows, dtype=torch.int64, device=dev)
        G = rows.numel()
        B = int(plan.step_len.max())
        model_of = rows % n_models
        work = global_params[model_of].clone()          # [G, P]
        if B == 0:
            # a chunk whose every pair has zero-length steps throughout:
            # kernel grids would be dim3(0) (invalid launch). Broadcast
            # only — matches the unchunked kernel, which skips len-0
            # steps per pair but still returns the staged weights.
            replicas[rows] = work
            return
        ws = self._workspace(G, B)
        step_off = torch.as_tensor(plan.step_off, dtype=torch.int64,
                                   device=dev)
        step_len = torch.as_tensor(plan.step_len, dtype=torch.int64,
                                   device=dev)
        adam = opt["kind"] == "adam"
        p1, p2 = self._dropout_ps()
        E = step_off.shape[1]
        self._seed_counter += 1
        seed = (self._seed_counter * 0x100000001B3) & ((1 << 63) - 1)
        xm = x_mask.contiguous() if x_mask is not None else None
        for e in range(E):
            self.mod.cnn_train_epoch(
                work, ws["grad"], rows, x_arena, y_arena,
                step_off, step_len, e, xm,
                ws["x1"], ws["a2"], ws["pidx"], ws["z1"], ws["a1"],
                ws["dz2"], ws["dz1"], ws["da2"], ws["zz2"], ws["dx1"],
                ws["c1part"], ws["wtf"], ws["wtd"], ws["z1part"],
                ws["w2part"], ws["b2part"],
                opt["m"] if adam else None,
                opt["v"] if adam else None,
                opt["vmax"] if adam else None,
                opt["t"] if adam else None,
                opt["lr"], float(opt.get("wd", 0.0)),
                p1, p2, seed, g_base, B, self.O, ws["w2ms"])
        replicas[rows] = work

    # -- evaluation -------------------------------------------------------
    # bound the pooled-activation staging arena: windows per sweep chunked
    # so slots * NF floats stays within budget
    # slots per sweep chunk: bounds the x1e/z2e/a2e staging arenas
    # (~280 KB per slot across the three)
    EVAL_SLOT_BUDGET = 8192

    def _eval_sweep(self, params, task_row, task_id, win_off, win_len,
                    n_tasks, mode, want_mse=False, x_mask=None,
                    dump_sink=None):
        """Chunked batched sweep. Windows are sorted by model row (stable)
        so the fc1 MFMA GEMM can group slots per weight row; accuracy/
        confusion accumulation is order-invariant, and the DUMP sink
        receives the chunk's (sorted) window arrays plus per-slot ids."""
        dev = self.device
        W = task_row.numel()
        if mode == EV_ACC:
            total_out = torch.zeros(4 if want_mse else 3, n_tasks,
                                    dtype=torch.float64, device=dev)
        elif mode == EV_CONF:
            total_out = torch.zeros(n_tasks, self.O, self.O,
                                    dtype=torch.float64, device=dev)
        else:
            total_out = None
        if W == 0:
            return total_out
        order = torch.argsort(task_row, stable=True)
        task_row = task_row[order]
        task_id = task_id[order]
        win_off = win_off[order]
        win_len = win_len[order]
        if x_mask is not None and x_mask.dim() == 2:
            x_mask = x_mask[order]
        csum = torch.cumsum(win_len, 0)
        start = 0
        a2e = z1e = z1pe = x1e = z2e = None
        # reshaped conv2 weights for the eval MFMA B-operand:
        # [M, 9, 32, 64] from the [co][ci][ky][kx] flat slice (L2-hot,
        # shared across every block of a sweep)
        wtf_e = params[:, 320:18752].reshape(-1, 64, 32, 9) \
            .permute(0, 3, 2, 1).contiguous()
        while start < W:
            base = csum[start - 1] if start > 0 else csum.new_zeros(())
            end_idx = int(torch.searchsorted(
                csum, base + self.EVAL_SLOT_BUDGET, right=True))
            end = max(start + 1, min(W, end_idx))
            tr = task_row[start:end].contiguous()
            ti = task_id[start:end].contiguous()
            wo = win_off[start:end].contiguous()
            wl = win_len[start:end].contiguous()
            slot = torch.cumsum(wl, 0) - wl
            slots = int(wl.sum())
            max_len = int(wl.max()) if wl.numel() else 0
            if a2e is None or a2e.shape[0] < slots:
                a2e = torch.empty(max(slots, 1), NF, device=dev)
                z1e = torch.empty(max(slots, 1), NH, device=dev)
                z1pe = torch.empty(max(slots, 1), 8 * NH, device=dev)
                x1e = torch.empty(max(slots, 1), X1N, device=dev)
                z2e = torch.empty(max(slots, 1), Z2N, device=dev)
            # per-slot metadata
            srow = tr.repeat_interleave(wl)
            stid = ti.repeat_interleave(wl)
            swin = torch.arange(tr.numel(), device=dev) \
                .repeat_interleave(wl)
            within = (torch.arange(slots, device=dev)
                      - slot.repeat_interleave(wl))
            soff = wo.repeat_interleave(wl) + within
            sy = self._y_arena[soff]
            # fc1 GEMM blocks: runs of equal row, tiled by 64 slots
            blk_row, blk_s0, blk_len = self._fc1_blocks(tr, wl, slot)
            xm = None
            if x_mask is not None:
                xm = (x_mask[start:end].contiguous()
                      if x_mask.dim() == 2 else x_mask.contiguous())
            outp = None
            if mode == EV_DUMP:
                outp = torch.empty(slots, self.O, device=dev)
            out = self.mod.cnn_eval(
                params.contiguous(), tr, ti, wo, wl, slot,
                self._x_arena, self._y_arena, a2e, z1e, z1pe, x1e, z2e,
                wtf_e,
                blk_row, blk_s0, blk_len, srow, stid, sy.contiguous(),
                soff.contiguous(), swin.contiguous(),
                xm, n_tasks, self.O, mode, want_mse, max_len, slots, outp)
            if mode == EV_DUMP:
                dump_sink(stid, sy, outp)
            else:
                total_out += out
            start = end
        return total_out

    @staticmethod
    def _fc1_blocks(tr: torch.Tensor, wl: torch.Tensor, slot: torch.Tensor):
        """Tile the (row-sorted) slot space into <=64-slot blocks that
        never cross a model-row boundary."""
        dev = tr.device
        trc = tr.cpu().numpy()
        wlc = wl.cpu().numpy()
        slc = slot.cpu().numpy()
        rows, s0s, lens = [], [], []
        i = 0
        Wn = len(trc)
        while i < Wn:
            j = i
            while j < Wn and trc[j] == trc[i]:
                j += 1
            run_s0 = int(slc[i])
            run_end = int(slc[j - 1] + wlc[j - 1])
            for s in range(run_s0, run_end, 64):
                rows.append(int(trc[i]))
                s0s.append(s)
                lens.append(min(64, run_end - s))
            i = j
        t = lambda a: torch.as_tensor(a, dtype=torch.int64, device=dev)
        return t(rows), t(s0s), t(lens)

    @torch.no_grad()
    def eval_tasks_stacked(self, params: torch.Tensor, task_row, task_id,
                           win_off, win_len, n_tasks: int,
                           want_mse: bool = False, x_arena=None,
                           y_arena=None,
                           x_mask: Optional[torch.Tensor] = None,
                           **kw) -> torch.Tensor:
        self._x_arena, self._y_arena = x_arena, y_arena
        return self._eval_sweep(params, task_row, task_id, win_off, win_len,
                                n_tasks, EV_ACC, want_mse=want_mse,
                                x_mask=x_mask)

    @torch.no_grad()
    def confusion_tasks(self, params: torch.Tensor, x_arena, y_arena,
                        task_row, task_id, win_off, win_len, n_tasks: int,
                        n_classes: int,
                        x_mask: Optional[torch.Tensor] = None):
        assert n_classes == self.O
        self._x_arena, self._y_arena = x_arena, y_arena
        return self._eval_sweep(params, task_row, task_id, win_off, win_len,
                                n_tasks, EV_CONF, x_mask=x_mask)

    @torch.no_grad()
    def vote_multi(self, params: torch.Tensor, weights: torch.Tensor,
                   x_arena, y_arena, task_id, win_off, win_len,
                   n_tasks: int, mode: str = "hard",
                   masks: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Batched weighted ensemble vote over many tasks: per model, one
        probability-dump sweep; votes accumulate as batched torch ops.
        weights: [M] or [n_tasks, M]. Returns [2, n_tasks] correct/total."""
        dev = self.device
        self._x_arena, self._y_arena = x_arena, y_arena
        M = params.shape[0]
        W = task_id.numel()
        out = torch.zeros(2, n_tasks, dtype=torch.float64, device=dev)
        if W == 0:
            return out
        per_task = weights.dim() == 2
        active = [m for m in range(M)
                  if float(weights[:, m].abs().max() if per_task
                           else weights[m].abs()) > 0]
        if not active:
            return out
        slots_total = int(win_len.sum())
        votes = torch.zeros(slots_total, self.O, device=dev)
        wv = weights.to(dev)
        # per-model sweeps: all rows equal per sweep, so the row-sort in
        # _eval_sweep is the identity and slot order is consistent across
        # models; votes accumulate per slot in that order
        slot_meta = {}

        for m in active:
            rowv = torch.full_like(task_id, m)
            xm = masks[m] if masks is not None else None
            written = [0]

            def sink(stid, sy, probs, _m=m, _w=written):
                s0 = _w[0]
                ns = probs.shape[0]
                if mode == "hard":
                    contrib = torch.zeros_like(probs)
                    contrib.scatter_(1, probs.argmax(1, keepdim=True), 1.0)
                else:
                    contrib = probs
                if per_task:
                    wslice = wv[stid, _m]
                else:
                    wslice = wv[_m].expand(ns)
                votes[s0:s0 + ns] += wslice.unsqueeze(1).float() * contrib
                if _m == active[0]:
                    slot_meta[s0] = (stid, sy)
                _w[0] = s0 + ns

            self._eval_sweep(params, rowv, task_id, win_off, win_len,
                             n_tasks, EV_DUMP, x_mask=xm, dump_sink=sink)
        tids = torch.cat([slot_meta[k][0] for k in sorted(slot_meta)])
        y = torch.cat([slot_meta[k][1] for k in sorted(slot_meta)])
        pred = votes.argmax(1)
        corr = (pred == y).double()
        out[0].scatter_add_(0, tids, corr)
        out[1].scatter_add_(0, tids, torch.ones_like(corr))
        return out

    @torch.no_grad()
    def ens_vote_eval(self, params: torch.Tensor, weights: torch.Tensor,
                      x_arena, y_arena, windows, mode: str = "hard",
                      masks: Optional[torch.Tensor] = None):
        """Single-task vote (compat shim over vote_multi)."""
        wins = [(o, l) for (o, l) in windows if l > 0]
        if not wins:
            return 0.0, 0.0
        dev = self.device
        off = torch.as_tensor([o for o, _ in wins], dtype=torch.int64,
                              device=dev)
        ln = torch.as_tensor([l for _, l in wins], dtype=torch.int64,
                             device=dev)
        tid = torch.zeros(len(wins), dtype=torch.int64, device=dev)
        out = self.vote_multi(params, weights, x_arena, y_arena, tid, off,
                              ln, 1, mode=mode, masks=masks)
        return float(out[0, 0]), float(out[1, 0])
