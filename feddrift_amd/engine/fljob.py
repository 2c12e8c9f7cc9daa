"""FLJob: one training iteration (= one reference MPI job) on the new engine.

One process per GPU; worker slots are sharded across ranks
(w % world_size == rank). Each round:

  1. sync_replicas: every owned (worker, model) replica row <- global model
     row (a device-side broadcast copy; the reference re-sends pickled
     state_dicts to every worker process instead,
     FedAvgEnsServerManager.py:66-67).
  2. the algorithm plans which pairs train on which minibatch windows
     (host-side RNG, statistical parity with the reference's client-side
     np.random.choice picks, FedAvgEnsTrainer.py:67).
  3. ops.train_fused runs ALL local training for this rank in one batched
     op (HIP kernel on GPU / vectorized torch on CPU).
  4. aggregation: one fused [K, P+1] weighted-sum + all_reduce over
     RCCL/gloo (replaces FedAvgEnsAggregatorSoftCluster.py:148-195).
  5. algorithm post-aggregation hooks (clustering, ensemble weights, lr
     adaptation) run LOCKSTEP on every rank from identical allreduced
     inputs — no control-plane messages at all.
  6. prequential evaluation, sharded by client + allreduced.

Checkpoint layout is reference-compatible: model_params.pt =
dict(enumerate(state_dicts)) + per-algorithm state pickles
(FedAvgEnsServerManager.py:84-86).
"""

from __future__ import annotations

import os
import pickle
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from .. import ops
from ..comm import Communicator
from ..config import Config
from ..data.loader import ClientData, DriftDataset, RetrainView
from ..eval.metrics import MetricLogger
from ..models import packed, zoo
from .arena import DeviceArena, SegRef


@dataclass
class PlanTemplate:
    """Cached structure of a round's training plan: which pairs train, from
    which batch pools, with which aggregation weights. Only the random batch
    picks change between rounds — drawing them is one vectorized RNG call
    (the reference draws np.random.choice per model per step per worker,
    FedAvgEnsTrainer.py:67; same distribution)."""
    rows: np.ndarray         # [G]
    sample_num: np.ndarray   # [nW_local, K]
    pool_start: np.ndarray   # [G] into pool_off/pool_len
    pool_size: np.ndarray    # [G]
    pool_off: np.ndarray     # [sum pools]
    pool_len: np.ndarray     # [sum pools]

    def draw(self, rng: np.random.Generator, epochs: int) -> "TrainPlan":
        G = len(self.rows)
        if G == 0:
            return TrainPlan(self.rows, np.zeros((0, epochs), np.int64),
                             np.zeros((0, epochs), np.int64), self.sample_num,
                             template=self)
        # scratch reuse: at thousands of pairs the four fresh [G, E]
        # allocations cost more than the draw itself.  Generator.random
        # with `out=` consumes the identical bit stream as random(size),
        # so cached-buffer draws are bit-equal to the uncached path.
        sc = getattr(self, "_scratch", None)
        if sc is None or sc[0].shape != (G, epochs):
            sc = (np.empty((G, epochs), np.float64),
                  np.empty((G, epochs), np.int64),
                  np.empty((G, epochs), np.int64),
                  np.empty((G, epochs), np.int64))
            self._scratch = sc
        u, idx, off_o, len_o = sc
        rng.random(out=u)
        np.multiply(u, self.pool_size[:, None], out=u)
        idx[:] = u                          # float -> int64 truncation
        idx += self.pool_start[:, None]
        np.take(self.pool_off, idx, out=off_o)
        np.take(self.pool_len, idx, out=len_o)
        return TrainPlan(self.rows, off_o, len_o,
                         self.sample_num, template=self)


def build_template(pairs, sample_num) -> PlanTemplate:
    """pairs: list of (row, windows); windows = [(off, len), ...] pools."""
    rows, starts, sizes, offs, lens = [], [], [], [], []
    for row, wins in pairs:
        rows.append(row)
        starts.append(len(offs))
        sizes.append(len(wins))
        for o, l in wins:
            offs.append(o)
            lens.append(l)
    return PlanTemplate(np.asarray(rows, dtype=np.int64),
                        sample_num,
                        np.asarray(starts, dtype=np.int64),
                        np.asarray(sizes, dtype=np.int64),
                        np.asarray(offs, dtype=np.int64),
                        np.asarray(lens, dtype=np.int64))


@dataclass
class TrainPlan:
    """Task arrays for one round of local training on this rank."""
    rows: np.ndarray            # [G] row ids into the replica buffer
    step_off: np.ndarray        # [G, E]
    step_len: np.ndarray        # [G, E]
    # aggregation weights per (local worker, model): float (batch counts or
    # sample counts, per-algorithm — reference passes them as num_samples)
    sample_num: np.ndarray      # [nW_local, K]
    x_mask: Optional[torch.Tensor] = None   # [G, D] per-pair input mask (KUE)
    template: Optional["PlanTemplate"] = None


class TaskList:
    """Builder for batched evaluation sweeps."""

    def __init__(self):
        self.task_row: List[int] = []
        self.task_id: List[int] = []
        self.off: List[int] = []
        self.ln: List[int] = []
        self.n_tasks = 0

    def new_task(self) -> int:
        self.n_tasks += 1
        return self.n_tasks - 1

    # split long windows: more workgroups, short latency chains.  With the
    # single-wave eval kernel each block's fixed cost (weight staging +
    # reduce + atomics) is amortized over CHUNK samples; env-tunable for
    # same-box A/B of the block-count/blocking trade-off.
    CHUNK = int(os.environ.get("FEDDRIFT_EVAL_CHUNK", "128"))

    def add_windows(self, task_id: int, row: int,
                    windows: Sequence[Tuple[int, int]]) -> None:
        for o, l in windows:
            while l > 0:
                ln = min(l, self.CHUNK)
                self.task_row.append(row)
                self.task_id.append(task_id)
                self.off.append(o)
                self.ln.append(ln)
                o += ln
                l -= ln


class FLJob:
    def __init__(self, cfg: Config, comm: Communicator,
                 logger: Optional[MetricLogger] = None,
                 dataset: Optional[DriftDataset] = None):
        self.cfg = cfg
        self.comm = comm
        self.device = comm.device
        self.logger = logger or MetricLogger(
            cfg.log_dir, enabled=comm.is_root,
            use_wandb=bool(cfg.wandb),
            run_name=f"FedAvgCont-{cfg.dataset}-{cfg.concept_drift_algo}"
                     f"-iter{cfg.curr_train_iteration}")
        self.backend = ops.backend_for(self.device, cfg.use_hip_kernels)

        # seeds: np partition/batch RNG + torch init seed keyed on dummy_arg
        # (reference main_fedavg.py:292-298)
        np.random.seed(cfg.dummy_arg)
        torch.manual_seed(cfg.dummy_arg)
        zoo.set_torch_seed(cfg.dummy_arg)

        self.curr_iter = cfg.curr_train_iteration
        self.dataset = dataset or DriftDataset(
            cfg.data_dir, cfg.dataset, cfg.client_num_in_total,
            partition=cfg.change_points)  # fmow: partition letter = CP name
                                          # (fmow/data_loader.py:65)
        # two model paths: the MLP family runs on the fused HIP kernels /
        # batched torch ops; convolutional models (cnn / resnet) run as one
        # live torch module per rank over flat state rows (MIOpen convs)
        self.is_module_path = cfg.model not in ("lr", "fnn")
        if self.is_module_path:
            from ..models.generic_packer import ModulePacker
            proto0 = zoo.create_model(cfg.model, self.dataset.class_num,
                                      self.dataset.feature_num)
            self.packer = ModulePacker(proto0)
            self.spec = None
            from ..ops.module_vmap import VmapEngine, vmap_compatible
            has_buffers = len(list(proto0.buffers())) > 0
            # CNN_DropOut on a GPU: the hand-written CDNA4 kernel engine
            # (ops/cnn_hip.py, ops/hip/cnn_kernels.hip) owns the hot path
            # end-to-end. Required on GPU devices unless explicitly
            # disabled (FEDDRIFT_CNN_HIP=0 forces the vmap/MIOpen engine
            # for A/B) — a missing .so fails loudly rather than silently
            # falling back (ops/hip_loader.py policy).
            from ..ops.cnn_hip import CnnHipEngine, is_cnn_dropout
            use_cnn_hip = (self.device.type == "cuda"
                           and is_cnn_dropout(proto0)
                           and os.environ.get("FEDDRIFT_CNN_HIP") != "0"
                           and cfg.use_hip_kernels != "never")
            # Buffer-free modules (CNN_DropOut): vmap-batched autograd is
            # the measured winner at every size.  BN models (ResNet) are
            # supported by the vmap engine too (batched buffer pytrees,
            # parity-tested), but MIOpen's grouped convs lose to
            # sequential eager at SMALL fleets (1264 vs 499 ms/round at
            # 20 pairs) and win at scale (763 vs 1420 ms/round at 100
            # clients, same accuracy) -> dispatch BN models by fleet
            # size; FEDDRIFT_VMAP_BN=1/0 forces either way.
            force = os.environ.get("FEDDRIFT_VMAP_BN")
            bn_vmap = (force == "1" if force in ("0", "1")
                       else cfg.client_num_per_round >= 64)
            use_vmap = vmap_compatible(proto0) and (
                not has_buffers or bn_vmap)
            if use_cnn_hip:
                self.mod_engine = CnnHipEngine(proto0, self.packer,
                                               self.device)
            elif use_vmap:
                self.mod_engine = VmapEngine(proto0, self.packer,
                                             self.device)
            else:
                from ..ops.module_engine import ModuleEngine
                self.mod_engine = ModuleEngine(proto0, self.packer,
                                               self.device)
        else:
            self.spec = packed.spec_for(cfg.model, self.dataset.feature_num,
                                        self.dataset.class_num)
            self.packer = packed.PackedMLP(self.spec)
            self.mod_engine = None

        self.n_params = self.packer.n_params if self.is_module_path \
            else self.spec.n_params

        from . import algorithms
        self.algo = algorithms.make(cfg)

        # data: all_data history + the algorithm's per-model retrain views
        self.cdata = ClientData(self.dataset.store, self.curr_iter,
                                cfg.batch_size,
                                seed=cfg.dummy_arg * 1000003 + self.curr_iter)
        self.views: List[RetrainView] = self.algo.build_views(self)
        self.n_models = len(self.views)

        # device arena
        self.arena = DeviceArena(self.dataset.feature_num, self.device)
        C, T = cfg.client_num_in_total, self.curr_iter + 1
        self.all_ref: List[List[SegRef]] = [
            [self.arena.add(self.cdata.all_data[c][t]) for t in range(T)]
            for c in range(C)]
        self.view_train_ref: List[Dict[int, SegRef]] = [
            {c: self.arena.add(v.train[c]) for c in v.train}
            for v in self.views]
        # test set is iteration t+1 for every view — upload once (accuracy
        # sums are shuffle-invariant, so one view's shuffle suffices)
        self.test_ref: Dict[int, SegRef] = {
            c: self.arena.add(self.views[0].test[c])
            for c in self.views[0].test}
        self.algo.add_extra_segments(self)
        self.arena.freeze()

        # model bank: K global rows + canonical init row
        P = self.n_params
        proto = zoo.create_model(cfg.model, self.dataset.class_num,
                                 self.dataset.feature_num)
        self.init_flat = self.packer.flatten(proto.state_dict()).to(self.device)
        self.global_params = self.init_flat.unsqueeze(0).repeat(
            self.n_models, 1).contiguous()
        self.algo.load_checkpoint(self)

        # replicas + optimizer state for owned worker slots
        self.n_workers = cfg.client_num_per_round
        self.owned_workers = comm.owned_workers(self.n_workers)
        nW = len(self.owned_workers)
        self.replicas = torch.zeros(nW * self.n_models, P, device=self.device)
        if self.is_module_path:
            self.opt = self.mod_engine.make_opt_state(
                cfg.client_optimizer, nW * self.n_models, cfg.lr, cfg.wd)
        else:
            self.opt = ops.mlp_torch.make_opt_state(
                cfg.client_optimizer, nW * self.n_models, P, cfg.lr, cfg.wd,
                self.device)

        # per-rank batch-pick RNG (statistical parity; reference draws on
        # each worker process's own global np RNG)
        self.pick_rng = np.random.default_rng(
            (cfg.dummy_arg * 7919 + self.curr_iter) * 1009 + comm.rank)

        self._eval_cache: Dict = {}
        self._eval_fast: Optional[tuple] = None
        self._partial: Optional[torch.Tensor] = None
        self._partial_fused = False
        self.algo.init_iteration(self)

    # ------------------------------------------------------------------
    # helpers
    # ------------------------------------------------------------------
    def row(self, wi: int, m: int) -> int:
        """Replica row for owned-worker index wi, model m."""
        return wi * self.n_models + m

    def sync_replicas(self) -> None:
        nW = len(self.owned_workers)
        if nW:
            self.replicas.copy_(
                self.global_params.unsqueeze(0).expand(nW, -1, -1)
                .reshape(-1, self.n_params))

    def eval_tensors(self, tl: TaskList):
        """Upload one task list as device tensors (cacheable)."""
        dev = self.device
        idx = torch.as_tensor(
            np.stack([np.asarray(tl.task_row, dtype=np.int64),
                      np.asarray(tl.task_id, dtype=np.int64),
                      np.asarray(tl.off, dtype=np.int64),
                      np.asarray(tl.ln, dtype=np.int64)]), device=dev)
        return idx

    def run_eval_dev(self, params: torch.Tensor, tl: TaskList,
                     want_mse: bool = False,
                     idx: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Batched eval sweep; returns a stacked DEVICE tensor
        [correct; total; loss (; mse)] x n_tasks (float64) — callers
        all_reduce it and download once."""
        if idx is None:
            idx = self.eval_tensors(tl)
        if self.is_module_path:
            return self.mod_engine.eval_tasks_stacked(
                params, idx[0], idx[1], idx[2], idx[3], tl.n_tasks,
                want_mse=want_mse, x_arena=self.arena.x,
                y_arena=self.arena.y)
        return self.backend.eval_tasks_stacked(
            self.spec, params, self.arena.x, self.arena.y,
            idx[0], idx[1], idx[2], idx[3], tl.n_tasks, want_mse=want_mse)

    def run_eval(self, params: torch.Tensor, tl: TaskList,
                 want_mse: bool = False):
        out = self.run_eval_dev(params, tl, want_mse).cpu().numpy()
        return [out[0], out[1], out[2], out[3] if want_mse else None]

    def ens_vote_multi(self, weights: torch.Tensor, tl: TaskList,
                       idx: torch.Tensor, mode: str,
                       masks: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Batched many-task ensemble vote ([2, n_tasks] correct/total)."""
        if not self.is_module_path:
            return self.backend.ens_vote_multi(
                self.spec, self.global_params, weights, self.arena.x,
                self.arena.y, idx[1], idx[2], idx[3], tl.n_tasks,
                mode=mode, masks=masks)
        if hasattr(self.mod_engine, "vote_multi"):
            # batched kernel path (CnnHipEngine): one dump sweep per active
            # model, vote accumulation in batched torch ops — replaces the
            # per-task Python loop below
            return self.mod_engine.vote_multi(
                self.global_params, weights, self.arena.x, self.arena.y,
                idx[1], idx[2], idx[3], tl.n_tasks, mode=mode, masks=masks)
        out = torch.zeros(2, tl.n_tasks, dtype=torch.float64,
                          device=self.device)
        tid = idx[1].cpu().numpy()
        off = idx[2].cpu().numpy()
        ln = idx[3].cpu().numpy()
        for t in range(tl.n_tasks):
            wins = [(int(o), int(l)) for o, l, ti in zip(off, ln, tid)
                    if ti == t]
            if not wins:
                continue
            w_t = weights[t] if weights.dim() == 2 else weights
            c, n = self.mod_engine.ens_vote_eval(
                self.global_params, w_t, self.arena.x, self.arena.y, wins,
                mode=mode, masks=masks)
            out[0, t] = c
            out[1, t] = n
        return out

    def confusion(self, task_row, task_id, win_off, win_len, n_tasks: int,
                  n_classes: int, x_mask: Optional[torch.Tensor] = None):
        if self.is_module_path:
            return self.mod_engine.confusion_tasks(
                self.global_params, self.arena.x, self.arena.y, task_row,
                task_id, win_off, win_len, n_tasks, n_classes,
                x_mask=x_mask)
        return self.backend.confusion_tasks(
            self.spec, self.global_params, self.arena.x, self.arena.y,
            task_row, task_id, win_off, win_len, n_tasks, n_classes,
            x_mask=x_mask)

    def score_models_on_segments(self, flats, segs) -> tuple:
        """Batched accuracy of M flat parameter vectors on per-client
        host segments, used by the DriftSurf / MultiModelAcc data-load-
        time scoring (reference runs these as per-(model, client) CPU
        eager forwards at data-load time, DriftSurf_data_loader:269-314,
        FedAvgEnsDataLoader.py:350-390 — at FEMNIST scale that is
        minutes of host time per iteration). Here: one throwaway device
        arena + ONE batched eval sweep, sharded by client + allreduced.

        Returns ([M, C] correct, [M, C] total) numpy arrays."""
        M = len(flats)
        C = self.cfg.client_num_in_total
        dev = self.device
        if M == 0:
            return (np.zeros((0, C)), np.zeros((0, C)))
        params = torch.as_tensor(
            np.ascontiguousarray(np.stack(flats), dtype=np.float32),
            device=dev)
        xs, ys, wins = [], [], {}
        off = 0
        for c, seg in segs.items():
            if not self.comm.owns_client(c) or seg.n == 0:
                continue
            xs.append(np.asarray(seg.x, dtype=np.float32))
            ys.append(np.asarray(seg.y, dtype=np.int64))
            wins[c] = (off, seg.n)
            off += seg.n
        tl = TaskList()
        ids = {}
        for m in range(M):
            for c in range(C):
                tid = tl.new_task()
                ids[(m, c)] = tid
                if c in wins:
                    tl.add_windows(tid, m, [wins[c]])
        if xs:
            x_t = torch.as_tensor(np.concatenate(xs, 0), device=dev)
            y_t = torch.as_tensor(np.concatenate(ys, 0), device=dev)
        else:
            x_t = torch.zeros(1, self.dataset.feature_num, device=dev)
            y_t = torch.zeros(1, dtype=torch.int64, device=dev)
        idx = self.eval_tensors(tl)
        if self.is_module_path:
            res = self.mod_engine.eval_tasks_stacked(
                params, idx[0], idx[1], idx[2], idx[3], tl.n_tasks,
                x_arena=x_t, y_arena=y_t)
        else:
            res = self.backend.eval_tasks_stacked(
                self.spec, params, x_t, y_t, idx[0], idx[1], idx[2],
                idx[3], tl.n_tasks)
        self.comm.all_reduce_(res)
        cv = res.cpu().numpy()
        correct = np.zeros((M, C))
        total = np.zeros((M, C))
        for (m, c), tid in ids.items():
            correct[m, c] = cv[0][tid]
            total[m, c] = cv[1][tid]
        return correct, total

    def train(self, plan: TrainPlan) -> None:
        """All local training of this round in ONE fused launch. On the HIP
        path the kernel also (a) stages each pair's initial weights straight
        from the global model row (fusing the server->client broadcast) and
        (b) accumulates the weighted aggregation partial sums on its way out
        — so the whole round's compute is train-kernel + all_reduce +
        apply-kernel."""
        K, P = self.n_models, self.n_params
        dev = self.device
        robust = self.cfg.robust_norm_bound > 0
        hip = (not self.is_module_path) and self.backend is not ops.mlp_torch \
            and not robust   # clipping happens between train and aggregate
        if self._partial is None or self._partial.shape[0] != K:
            self._partial = torch.zeros(K, P + 1, device=dev)
            self._totals = torch.zeros(K, device=dev)
            self._partial_clean = True
        if hip:
            # the apply kernel drains partial to zero on its way out, so
            # steady-state rounds skip this fill entirely
            if not getattr(self, "_partial_clean", False):
                self._partial.zero_()
            self._partial_clean = False
        if plan.rows.size == 0:
            self._partial_fused = hip
            if not hip and not self.is_module_path:
                self.sync_replicas()
            return
        if self.is_module_path:
            self.mod_engine.train(self.global_params, self.replicas, plan,
                                  self.opt, self.arena.x, self.arena.y,
                                  K, x_mask=plan.x_mask)
            self._partial_fused = False
            self._robust_clip(plan)
            return
        off_t, len_t = self._upload_steps(plan, dev)
        if hip:
            # per-template constants (rows / model / aggregation weights)
            # are uploaded once and reused every round
            tmpl = plan.template
            cache = getattr(tmpl, "_dev", None) if tmpl is not None else None
            if cache is None:
                wi = plan.rows // K
                mo = (plan.rows % K).astype(np.int32)
                sw = plan.sample_num[wi, plan.rows % K].astype(np.float32)
                cache = (
                    torch.as_tensor(plan.rows, dtype=torch.int64, device=dev),
                    torch.as_tensor(mo, device=dev),
                    torch.as_tensor(sw, device=dev))
                if tmpl is not None:
                    tmpl._dev = cache
            rows_t, mo_t, sw_t = cache
            self.backend.train_fused(
                self.spec, self.replicas, rows_t, self.arena.x, self.arena.y,
                off_t, len_t, self.opt, x_mask=plan.x_mask,
                in_params=self.global_params,
                model_of=mo_t, sample_w=sw_t,
                partial=self._partial)
            self._partial_fused = True
        else:
            self.sync_replicas()
            rows_t = torch.as_tensor(plan.rows, dtype=torch.int64, device=dev)
            self.backend.train_fused(
                self.spec, self.replicas, rows_t, self.arena.x, self.arena.y,
                off_t, len_t, self.opt, x_mask=plan.x_mask)
            self._partial_fused = False
            self._robust_clip(plan)

    def _upload_steps(self, plan: TrainPlan, dev: torch.device):
        """Per-round upload of the [G, E] step windows. On GPU, a pageable
        torch.as_tensor costs tens of us per array in alloc + staged copy;
        a per-template pinned double buffer turns both arrays into ONE
        non-blocking DMA (the buffer slot is event-guarded against reuse
        while a previous round's copy is still in flight)."""
        tmpl = plan.template
        if dev.type != "cuda" or tmpl is None:
            return (torch.as_tensor(plan.step_off, dtype=torch.int64,
                                    device=dev),
                    torch.as_tensor(plan.step_len, dtype=torch.int64,
                                    device=dev))
        G, E = plan.step_off.shape
        st = getattr(tmpl, "_stage", None)
        if st is None or st["pin"].shape[2:] != (G, E):
            pin = torch.empty(2, 2, G, E, dtype=torch.int64,
                              pin_memory=True)
            tmpl._stage = st = {
                "pin": pin, "np": pin.numpy(),
                "buf": torch.empty(2, G, E, dtype=torch.int64, device=dev),
                "ev": [torch.cuda.Event(), torch.cuda.Event()], "slot": 0}
        slot = st["slot"]
        st["slot"] = slot ^ 1
        st["ev"][slot].synchronize()
        np.copyto(st["np"][slot][0], plan.step_off)
        np.copyto(st["np"][slot][1], plan.step_len)
        st["buf"].copy_(st["pin"][slot], non_blocking=True)
        st["ev"][slot].record()
        return st["buf"][0], st["buf"][1]

    def _robust_clip(self, plan: TrainPlan) -> None:
        """Optional defense: clip client updates to an L2 ball before they
        enter the aggregation (fedavg_robust equivalent,
        fedml_core robustness/robust_aggregation.py:38)."""
        if self.cfg.robust_norm_bound <= 0 or plan.rows.size == 0:
            return
        from ..comm.robust import robustify_replicas
        rows_t = torch.as_tensor(plan.rows, dtype=torch.int64,
                                 device=self.device)
        mo = torch.as_tensor(plan.rows % self.n_models, device=self.device)
        robustify_replicas(self.replicas, self.global_params, rows_t, mo,
                           self.cfg.robust_norm_bound)

    def aggregate(self, plan: TrainPlan,
                  model_mask: Optional[np.ndarray] = None) -> torch.Tensor:
        """Per-model sample-weighted averaging across all workers: one
        all_reduce of the fused [K, P+1] partial-sum tensor + one apply op.

        model_mask[m]=False skips the model entirely (its global params are
        left untouched), matching the softcluster skip rule
        (FedAvgEnsAggregatorSoftCluster.py:151-153). Returns total weights
        per model (device tensor). Models with zero total weight are
        skipped (:167-169)."""
        K, P = self.n_models, self.n_params
        partial = self._partial
        if not self._partial_fused:
            nW = len(self.owned_workers)
            partial.zero_()
            if nW:
                w = torch.as_tensor(plan.sample_num, dtype=torch.float32,
                                    device=self.device)      # [nW, K]
                reps = self.replicas.reshape(nW, K, P)
                partial[:, :P] = torch.einsum("wk,wkp->kp", w, reps)
                partial[:, P] = w.sum(dim=0)
        if self.cfg.secure_agg:
            self._apply_secure_masks(plan, partial)
        self.comm.all_reduce_(partial)
        totals = partial[:, P]
        mask_t = None
        if model_mask is not None:
            if isinstance(model_mask, torch.Tensor):
                mask_t = model_mask
            else:
                mask_t = torch.as_tensor(
                    np.ascontiguousarray(model_mask, dtype=np.uint8),
                    device=self.device)
        plain = (self.cfg.server_optimizer == "avg" and
                 self.cfg.robust_noise <= 0)
        if plain and not self.is_module_path and \
                self.backend is not ops.mlp_torch:
            from ..ops import mlp_hip
            totals = mlp_hip.apply_aggregate(self.global_params, partial,
                                             mask_t, self._totals)
            self._partial_clean = True
            return totals
        upd = totals > 0
        if mask_t is not None:
            upd &= mask_t.bool()
        newp = partial[:, :P] / totals.clamp(min=1e-30).unsqueeze(1)
        averaged = torch.where(upd.unsqueeze(1), newp, self.global_params)
        if self.cfg.robust_noise > 0:
            from ..comm.robust import add_noise
            if getattr(self, "_noise_gen", None) is None:
                # dedicated generator: every rank draws identical noise,
                # keeping the lockstep-replicated control flow intact
                self._noise_gen = torch.Generator(device=self.device)
                self._noise_gen.manual_seed(self.cfg.dummy_arg + 999331)
            noisy = add_noise(averaged, self.cfg.robust_noise,
                              self._noise_gen)
            averaged = torch.where(upd.unsqueeze(1), noisy, averaged)
        if self.cfg.server_optimizer != "avg":
            if getattr(self, "_server_opt", None) is None:
                from .server_opt import ServerOptimizer
                self._server_opt = ServerOptimizer(
                    self.global_params, self.cfg.server_optimizer,
                    self.cfg.server_lr)
            self._server_opt.step(self.global_params, averaged, upd)
        else:
            self.global_params.copy_(averaged)
        return totals

    def _apply_secure_masks(self, plan: TrainPlan,
                            partial: torch.Tensor) -> None:
        """Secure aggregation (reference turboaggregate equivalent,
        fedml_api/distributed/turboaggregate/): each active worker pair
        sharing a model adds +/- a seed-derived pairwise mask to its
        weighted upload, so every rank's all_reduce contribution is
        additively masked while the GLOBAL per-model sum is exact. The
        active (worker, model) set is agreed with one tiny all_reduce;
        pair seeds advance per aggregation so masks never repeat."""
        from ..comm.secure_agg import mask_for
        K, P = self.n_models, self.n_params
        W = self.n_workers
        act = torch.zeros(W, K, device=self.device)
        for wi, w in enumerate(self.owned_workers):
            for m in range(K):
                if plan.sample_num[wi, m] > 0:
                    act[w, m] = 1.0
        self.comm.all_reduce_(act)
        act_np = act.cpu().numpy() > 0
        self._secure_ctr = getattr(self, "_secure_ctr", 0) + 1
        base = (self.cfg.dummy_arg * 1009 + self.curr_iter) * 65537 \
            + self._secure_ctr
        owned = set(self.owned_workers)
        for m in range(K):
            ws = np.nonzero(act_np[:, m])[0]
            if len(ws) < 2:
                continue
            for w in ws:
                if int(w) in owned:
                    partial[m, :P] += mask_for(
                        int(w), [int(o) for o in ws], P, base * K + m,
                        self.device)

    def client_sampling(self, round_idx: int) -> np.ndarray:
        """Reference client_sampling (FedAvgEnsAggregatorSoftCluster.py:197-204):
        identity when every client participates, else a per-round seeded
        draw without replacement."""
        cfg = self.cfg
        if cfg.client_num_in_total == cfg.client_num_per_round:
            return np.arange(cfg.client_num_in_total)
        rs = np.random.RandomState(round_idx)
        return rs.choice(range(cfg.client_num_in_total),
                         min(cfg.client_num_per_round,
                             cfg.client_num_in_total), replace=False)

    # ------------------------------------------------------------------
    # model-bank operations (EngineHooks for the drift state machines)
    # ------------------------------------------------------------------
    def merge_models(self, base: int, second: int, w1: float, w2: float):
        self.global_params[base].mul_(w1).add_(self.global_params[second],
                                               alpha=w2)

    def reinit_model(self, m: int) -> None:
        self.global_params[m] = self.init_flat

    def copy_model(self, dst: int, src: int) -> None:
        self.global_params[dst] = self.global_params[src]

    def randomize_models_unseeded(self) -> None:
        """reference 'hard' iter-0 quirk: reset_parameters() WITHOUT
        re-seeding, so the K models diverge (needed for IFCA;
        FedAvgEnsAggregatorSoftCluster.py:64-69)."""
        proto = zoo.create_model(self.cfg.model, self.dataset.class_num,
                                 self.dataset.feature_num)
        # the reference resets the model's TOP-LEVEL children only
        # (FedAvgEnsAggregatorSoftCluster.py:66-69): on torchvision
        # models that means conv1/bn1/fc reset while the Sequential
        # block stages keep the shared init — PARTIAL divergence. Our
        # FlatImageModel wrapper adds one level, so unwrap to the
        # backbone first (children() on the wrapper itself reset
        # nothing, leaving all K models bit-identical — the round-1 bug;
        # full modules() recursion over-diverges relative to the
        # reference and measurably hurts IFCA on the CIFAR config).
        target = getattr(proto, "backbone", proto)
        for m in range(self.n_models):
            for layer in target.children():
                if hasattr(layer, "reset_parameters"):
                    layer.reset_parameters()
            self.global_params[m] = self.packer.flatten(
                proto.state_dict()).to(self.device)

    def train_acc_matrix_rows(self, model_rows: List[int]) -> np.ndarray:
        """[len(rows), C] acc of each global model row on each client's
        current-iteration data; sharded by client + allreduced."""
        C = self.cfg.client_num_in_total
        tl = TaskList()
        ids = {}
        for i, m in enumerate(model_rows):
            for c in range(C):
                tid = tl.new_task()
                ids[(i, c)] = tid
                if self.comm.owns_client(c):
                    tl.add_windows(tid, m,
                                   self.all_ref[c][self.curr_iter].windows)
        res = self.run_eval_dev(self.global_params, tl)
        self.comm.all_reduce_(res)
        cv = res.cpu().numpy()
        acc = np.zeros((len(model_rows), C))
        for (i, c), tid in ids.items():
            if cv[1][tid] != 0:
                acc[i][c] = cv[0][tid] / cv[1][tid]
        return acc

    def pooled_cluster_windows(self, weights: Dict[int, np.ndarray], m: int,
                               curr_iter: int) -> list:
        """Reference pooling order (FedAvgEnsDataLoader.py:901-910):
        client-major, then time."""
        out = []
        for c in range(self.cfg.client_num_in_total):
            for t in range(curr_iter + 1):
                if t in weights and weights[t][m][c] == 1:
                    out.extend(self.all_ref[c][t].windows)
        return out

    def cluster_pair_acc_windows(self, model_rows: List[int],
                                 cluster_windows: Dict[int, list],
                                 cap: int = 21) -> np.ndarray:
        """[K, K] acc of model i on cluster j's pooled (already shuffled)
        windows, capped at `cap` batches per pair (reference :923-931 with
        _infer_subset's 21-batch cap, :1111-1138). Evaluated identically on
        every rank (inputs are lockstep-identical)."""
        k = len(model_rows)
        tl = TaskList()
        ids = {}
        for i, mrow in enumerate(model_rows):
            for j, mj in enumerate(model_rows):
                tid = tl.new_task()
                ids[(i, j)] = tid
                tl.add_windows(tid, mrow, cluster_windows[mj][:cap])
        correct, total, _, _ = self.run_eval(self.global_params, tl)
        acc = np.zeros((k, k))
        for (i, j), tid in ids.items():
            if total[tid] != 0:
                acc[i][j] = correct[tid] / total[tid]
        return acc

    # per-client eval of chosen model rows on train + test (t+1) data
    def client_eval(self, model_idx_per_client: np.ndarray,
                    train_model_per_client: Optional[np.ndarray] = None,
                    train_on_view: bool = False):
        """Returns per-client (train_correct, train_total, train_loss) and
        (test_correct, test_total, test_loss) arrays, allreduced.

        train data is the current-iteration all_data (softcluster,
        FedAvgEnsAggregatorSoftCluster.py:227-231) or the per-model retrain
        view (the other aggregators) when train_on_view=True.
        ci==1 keeps only client 0 (reference CI shortcut, :259-264).
        The task tensors are cached per model assignment — the assignment
        changes only at clustering events, so steady-state rounds skip the
        task-list rebuild + upload entirely."""
        if train_model_per_client is None:
            train_model_per_client = model_idx_per_client
        # steady-state fast path: hashing two C-sized byte keys costs tens
        # of us at thousands of clients; an array compare against the last
        # assignment is ~5x cheaper and hits every round between
        # clustering events
        fast = self._eval_fast
        if fast is not None and fast[2] == train_on_view and \
                np.array_equal(fast[0], model_idx_per_client) and \
                np.array_equal(fast[1], train_model_per_client):
            cached = fast[3]
        else:
            key = (model_idx_per_client.tobytes(),
                   train_model_per_client.tobytes(), train_on_view)
            cached = self._eval_cache.get(key)
        C = self.cfg.client_num_in_total
        clients = range(C) if self.cfg.ci != 1 else range(1)
        if cached is None:
            tl = TaskList()
            for c in clients:
                tid_tr = tl.new_task()
                tid_te = tl.new_task()
                if self.comm.owns_client(c):
                    mt = int(train_model_per_client[c])
                    me = int(model_idx_per_client[c])
                    if train_on_view:
                        if c in self.view_train_ref[mt]:
                            tl.add_windows(tid_tr, mt,
                                           self.view_train_ref[mt][c].windows)
                    else:
                        tl.add_windows(tid_tr, mt,
                                       self.all_ref[c][self.curr_iter].windows)
                    if c in self.test_ref:
                        tl.add_windows(tid_te, me, self.test_ref[c].windows)
            cached = (tl, self.eval_tensors(tl))
            self._eval_cache[key] = cached
        if fast is None or cached is not fast[3]:
            self._eval_fast = (model_idx_per_client.copy(),
                               train_model_per_client.copy(),
                               train_on_view, cached)
        tl, idx = cached
        res = self.run_eval_dev(self.global_params, tl, idx=idx)
        self.comm.all_reduce_(res)
        correct, total, loss = res.cpu().numpy()
        n = len(list(clients))
        tr = (correct[0::2][:n], total[0::2][:n], loss[0::2][:n])
        te = (correct[1::2][:n], total[1::2][:n], loss[1::2][:n])
        return tr, te

    def log_round_stats(self, round_idx: int, tr, te) -> None:
        (trc, trt, trl) = tr
        (tec, tet, tel) = te
        if self.cfg.report_client == 1 and self.comm.is_root:
            for c in range(len(trc)):
                self.logger.log(
                    {f"Train/Acc-CL-{c}": trc[c] / trt[c] if trt[c] else -1},
                    round_idx)
                self.logger.log(
                    {f"Test/Acc-CL-{c}": tec[c] / tet[c] if tet[c] else -1},
                    round_idx)
        train_acc = trc.sum() / trt.sum() if trt.sum() else 0.0
        train_loss = trl.sum() / trt.sum() if trt.sum() else 0.0
        test_acc = tec.sum() / tet.sum() if tet.sum() else 0.0
        test_loss = tel.sum() / tet.sum() if tet.sum() else 0.0
        if self.comm.is_root:
            self.logger.log({"Train/Acc": train_acc, "Train/Loss": train_loss},
                            round_idx)
            self.logger.log({"Test/Acc": test_acc, "Test/Loss": test_loss},
                            round_idx)

    # ------------------------------------------------------------------
    # checkpointing (reference-compatible layout)
    # ------------------------------------------------------------------
    def ckpt_path(self, name: str) -> str:
        return os.path.join(self.cfg.log_dir, name)

    def save_model_params(self) -> None:
        if not self.comm.is_root:
            return
        sds = {m: self.packer.unflatten(self.global_params[m])
               for m in range(self.n_models)}
        torch.save(sds, self.ckpt_path("model_params.pt"))

    def load_model_params_general(self) -> None:
        """General reload rule: load models in saved order
        (main_fedavg.py:354-357)."""
        mp = torch.load(self.ckpt_path("model_params.pt"))
        for m_idx, sd in mp.items():
            if m_idx < self.n_models:
                self.global_params[m_idx] = self.packer.flatten(sd).to(
                    self.device)

    def save_state_pickle(self, name: str, state) -> None:
        if self.cfg.bench_mode:
            return
        # barrier BEFORE the write: every rank must be done reading the
        # previous state file before root truncates it (at world size 4
        # the iteration-start load on slow ranks raced root's re-save and
        # read a partial pickle); write-to-temp + atomic rename so a
        # reader never sees a half-written file; barrier after publishes.
        self.comm.barrier()
        if self.comm.is_root:
            tmp = self.ckpt_path(name) + ".tmp"
            with open(tmp, "wb") as f:
                pickle.dump(state, f)
            os.replace(tmp, self.ckpt_path(name))
        self.comm.barrier()

    def load_state_pickle(self, name: str):
        with open(self.ckpt_path(name), "rb") as f:
            return pickle.load(f)

    # ------------------------------------------------------------------
    # the round loop
    # ------------------------------------------------------------------
    def run(self) -> None:
        from .profiling import PhaseTimer
        timer = PhaseTimer()
        client_idx = self.client_sampling(0)
        for r in range(self.cfg.comm_round):
            with timer.phase("plan"):
                plan = self.algo.plan(self, r, client_idx)
            with timer.phase("train"):
                self.train(plan)
            with timer.phase("aggregate"):
                self.algo.aggregate(self, r, plan, client_idx)
                self.algo.post_aggregate(self, r)
            with timer.phase("test"):
                self.algo.test(self, r)
            client_idx = self.client_sampling(r + 1)
        self.save_model_params()
        self.algo.finalize(self)
        self.comm.barrier()
        if self.comm.is_root:
            self.logger.set_summary("phase_times", timer.summary())
        self.logger.flush()


def run_iteration(cfg: Config, comm: Optional[Communicator] = None,
                  logger: Optional[MetricLogger] = None) -> FLJob:
    comm = comm or Communicator(backend=cfg.backend if cfg.backend != "auto"
                                else "auto")
    job = FLJob(cfg, comm, logger)
    job.run()
    return job
