"""Per-algorithm policies: local-training plans, aggregation rules,
prequential testing, checkpoint wiring.

Each class re-expresses one reference (data_loader, trainer, aggregator)
triple (fedml_api/distributed/fedavg_ens/*) on the batched engine. The
DRIFT_ALGO surface matches the reference dispatch
(FedAvgEnsAPI.py:32-60, :99-141, :150-173):

  aue / auepc / driftsurf / mmacc / mmgeni / mmgeniex / softcluster(+win-1,
  +reset) / ada / exp / lin / kue, plus 'single' (the fedavg_cont_one
  single-model baselines win-1/win-2/all driven by --retrain_data).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import numpy as np
import torch

from ..config import (Config, driftsurf_delta, parse_ada_arg,
                      parse_softcluster_arg)
from ..data.generators import load_change_points
from ..drift.ada import AdaState
from ..drift.driftsurf import DriftSurfState
from ..drift.kue import KueState
from ..drift.mmacc import MultiModelAccState
from ..drift.softcluster import SoftClusterState
from .fljob import FLJob, TaskList, TrainPlan, build_template


def make(cfg: Config):
    algo = cfg.concept_drift_algo
    if algo in ("aue", "auepc"):
        return AueAlgo(per_client=(algo == "auepc"))
    if algo in ("driftsurf", "dsurf"):   # reference README spells it dsurf
        return DriftSurfAlgo()
    if algo in ("win-1", "win-2", "all"):
        # reference cont_one surface: the window is the DRIFT_ALGO itself
        cfg.retrain_data = algo
        return SingleAlgo()
    if algo == "clusterfl":
        raise NameError(
            "clusterfl is obsolete in the reference too — use "
            "concept_drift_algo=softcluster with "
            "concept_drift_algo_arg=cfl_{gamma}_{win-1|all}")
    if algo in ("mmacc", "mmgeni", "mmgeniex"):
        return MultiModelAlgo(algo)
    if cfg.is_softcluster:
        return SoftClusterAlgo()
    if algo == "ada":
        return AdaAlgo()
    if algo in ("exp", "lin"):
        return DecayAlgo(algo)
    if algo == "kue":
        return KueAlgo()
    if algo == "single":
        return SingleAlgo()
    raise NameError(algo)


class AlgoBase:
    n_models_hint: Optional[int] = None

    # ---- data / model setup ----
    def build_views(self, job: FLJob) -> list:
        raise NotImplementedError

    def add_extra_segments(self, job: FLJob) -> None:
        pass

    def load_checkpoint(self, job: FLJob) -> None:
        """General model_params.pt reload rule (main_fedavg.py:354-357)."""
        if job.curr_iter != 0 and not job.cfg.reset_models and \
                os.path.exists(job.ckpt_path("model_params.pt")):
            job.load_model_params_general()

    def init_iteration(self, job: FLJob) -> None:
        pass

    # ---- round phases ----
    def plan(self, job: FLJob, round_idx: int,
             client_idx: np.ndarray) -> TrainPlan:
        return self.default_plan(job, client_idx)

    def aggregate(self, job: FLJob, round_idx: int, plan: TrainPlan,
                  client_idx: np.ndarray) -> None:
        job.aggregate(plan)

    def post_aggregate(self, job: FLJob, round_idx: int) -> None:
        pass

    def test(self, job: FLJob, round_idx: int) -> None:
        if round_idx % job.cfg.frequency_of_the_test == 0 or \
                round_idx == job.cfg.comm_round - 1:
            C = job.cfg.client_num_in_total
            tr, te = self.client_eval_views(job, np.zeros(C, dtype=int))
            job.log_round_stats(round_idx, tr, te)

    def finalize(self, job: FLJob) -> None:
        pass

    # ---- shared helpers ----
    _tmpl = None
    _tmpl_key = None

    def default_plan(self, job: FLJob, client_idx: np.ndarray,
                     lr: Optional[float] = None) -> TrainPlan:
        """Reference FedAvgEnsTrainer semantics: per model, E random batch
        picks from its retrain view; num_samples = the view's per-client
        sample count (FedAvgEnsTrainer.py:47-95). The plan structure is
        cached per client assignment; only the picks are drawn per round."""
        if lr is not None:
            job.opt["lr"].fill_(lr)
        key = client_idx.tobytes()
        if self._tmpl is None or self._tmpl_key != key:
            K = job.n_models
            nW = len(job.owned_workers)
            sample = np.zeros((nW, K))
            pairs = []
            for wi, w in enumerate(job.owned_workers):
                c = int(client_idx[w])
                for m in range(K):
                    n = job.views[m].train_n(c)
                    sample[wi, m] = n
                    if n == 0:
                        continue
                    pairs.append((job.row(wi, m),
                                  job.view_train_ref[m][c].windows))
            self._tmpl = build_template(pairs, sample)
            self._tmpl_key = key
        return self._tmpl.draw(job.pick_rng, job.cfg.epochs)

    def client_eval_views(self, job: FLJob, model_per_client: np.ndarray,
                          train_model_per_client: Optional[np.ndarray] = None,
                          train_on_view: bool = True):
        """test_on_all_clients core: per-client _infer of the chosen model on
        the view train data (or current-iter all_data) + the test set."""
        return job.client_eval(model_per_client, train_model_per_client,
                               train_on_view=train_on_view)


# ---------------------------------------------------------------------------
# hooks adapter: drift state machines -> engine
# ---------------------------------------------------------------------------
class Hooks:
    """EngineHooks implementation over an FLJob (see drift/softcluster.py)."""

    def __init__(self, job: FLJob):
        self.job = job

    def train_acc_matrix(self, models_in_use):
        return self.job.train_acc_matrix_rows(list(models_in_use))

    def cluster_pair_acc(self, models_in_use, cluster_batches):
        return self.job.cluster_pair_acc_windows(list(models_in_use),
                                                 cluster_batches)

    def pooled_cluster_batches(self, weights, model, curr_iter):
        return self.job.pooled_cluster_windows(weights, model, curr_iter)

    def merge_models(self, base, second, w1, w2):
        self.job.merge_models(base, second, w1, w2)

    def reinit_model(self, m):
        self.job.reinit_model(m)

    def copy_model(self, dst, src):
        self.job.copy_model(dst, src)

    def log_client(self, key_fmt, client, value, round_idx):
        if self.job.comm.is_root:
            self.job.logger.log({key_fmt.format(client): value}, round_idx)

    def log_summary(self, key, value):
        if self.job.comm.is_root:
            self.job.logger.set_summary(key, value)


# ---------------------------------------------------------------------------
# SoftCluster family: FedDrift (H_*), FedDrift-Eager (mmacc), IFCA (hard),
# CFL (cfl_*), softmax, gmm, geni
# ---------------------------------------------------------------------------
class SoftClusterAlgo(AlgoBase):
    _mask = None

    def build_views(self, job: FLJob) -> list:
        # views are win-1 placeholders; the trainer uses all_data
        # (reference SoftCluster_data_loader:1334-1341)
        v = job.cdata.view("win-1")
        return [v for _ in range(job.cfg.concept_num)]

    def _state_path(self, job):
        return job.ckpt_path("sc_state.pkl")

    _mpc = None      # cached per-client test-model argmax (invalidated on
                     # any clustering event; weights are static in between)

    def init_iteration(self, job: FLJob) -> None:
        cfg = job.cfg
        hooks = Hooks(job)
        self._mpc = None
        if cfg.bench_mode:
            # synthetic steady-state: client c trains/tests model c % K at
            # every iteration — K active clusters, full per-round load,
            # no iteration-start clustering (bench.py)
            st = SoftClusterState(cfg.client_num_in_total, cfg.concept_num,
                                  "hard", seed=cfg.dummy_arg)
            for t in range(job.curr_iter + 1):
                w = np.zeros((cfg.concept_num, cfg.client_num_in_total))
                for c in range(cfg.client_num_in_total):
                    w[c % cfg.concept_num][c] = 1.0
                st.train_data_weights[t] = w
            self.state = st
            return
        if job.curr_iter == 0 or not os.path.exists(self._state_path(job)):
            p = parse_softcluster_arg(cfg.concept_drift_algo_arg, cfg.dataset_norm)
            st = SoftClusterState(
                cfg.client_num_in_total, cfg.concept_num, p.cluster_alg,
                p.mmacc_delta, p.softmax_alpha,
                load_change_points(cfg.data_dir, cfg.change_points)
                if p.cluster_alg == "geni" else None,
                cfg.time_stretch, p.h_delta, p.h_deltap, p.h_w, p.h_distance,
                p.h_cluster, p.cfl_gamma, p.cfl_retrain, seed=cfg.dummy_arg)
        else:
            st = job.load_state_pickle("sc_state.pkl")
        self.state = st
        arg = cfg.concept_drift_algo_arg

        # iteration-start clustering (FedAvgEnsAggregatorSoftCluster.py:46-118)
        if "H" in arg:
            if job.curr_iter == 0:
                st.cluster_init(hooks)
            else:
                st.cluster_hierarchical(hooks, job.curr_iter)
        elif "cfl" in arg:
            if job.curr_iter == 0:
                st.cluster_init(hooks)
            else:
                st.cluster_cfl_init(hooks, job.curr_iter)
        elif "hard" in arg:
            if job.curr_iter == 0:
                job.randomize_models_unseeded()
            acc = job.train_acc_matrix_rows(list(range(job.n_models)))
            st.cluster(hooks, acc, job.curr_iter, 0)
        elif "mmacc" in arg:
            if job.curr_iter == 0:
                st.cluster_init(hooks)
            else:
                st.cluster_mmacc2(hooks, job.curr_iter)
        else:
            if job.curr_iter == 0:
                st.cluster_init(hooks)
            else:
                if cfg.concept_drift_algo == "softclusterreset":
                    # delete epsilon-dominated models
                    # (FedAvgEnsAggregatorSoftCluster.py:85-97)
                    acc = job.train_acc_matrix_rows(list(range(job.n_models)))
                    deleted: List[int] = []
                    for m in reversed(range(job.n_models)):
                        rest = np.delete(acc, deleted + [m], axis=0)
                        if rest.shape[0] > 0 and \
                                np.all(acc[m] < np.max(rest, axis=0) + 0.01):
                            deleted.append(m)
                            hooks.log_summary(f"Reset-{m}", 1)
                            st.set_weights_zero_model(m)
                            job.reinit_model(m)
                acc = job.train_acc_matrix_rows(list(range(job.n_models)))
                st.cluster(hooks, acc, job.curr_iter, 0)

        if cfg.concept_drift_algo == "softclusterwin-1":
            st.set_weights_win1(job.curr_iter)

        # record accuracy at first iteration so the drift detector starts
        # initialized (FedAvgEnsAggregatorSoftCluster.py:105-116)
        if job.curr_iter == 0:
            accs = job.train_acc_matrix_rows(list(range(job.n_models)))
            for c in range(cfg.client_num_in_total):
                st.set_acc(c, accs[st.get_test_model_idx(0, c)][c])

        job.save_state_pickle("sc_state.pkl", st)

    def invalidate_plan(self) -> None:
        """Weights changed mid-iteration (hard-r recluster / CFL split):
        rebuild the plan structure next round."""
        self._tmpl = None
        self._mask = None
        self._mpc = None

    def plan(self, job: FLJob, round_idx: int,
             client_idx: np.ndarray) -> TrainPlan:
        """TrainerSoftCluster: pairs weighted by sc_weights x iteration batch
        counts; unit-weight fast path pools batches across selected
        iterations (FedAvgEnsTrainerSoftCluster.py:63-135). NOTE the
        reference weighs by len(all_local_data[t]) = the BATCH count of
        iteration t, not the sample count — preserved. Structure cached;
        only the random picks are drawn per round."""
        key = client_idx.tobytes()
        if self._tmpl is None or self._tmpl_key != key:
            st = self.state
            K = job.n_models
            T = job.curr_iter + 1
            nW = len(job.owned_workers)
            sample = np.zeros((nW, K))
            pairs = []
            w_iter = st.get_weights()
            active = [bool(np.any(w_iter[job.curr_iter][m]))
                      for m in range(K)]
            for wi, w in enumerate(job.owned_workers):
                c = int(client_idx[w])
                for m in range(K):
                    if not active[m]:
                        continue
                    unnorm = np.array([
                        w_iter[t][m][c] * len(job.all_ref[c][t].windows)
                        for t in range(T)])
                    tot = unnorm.sum()
                    if tot == 0:
                        continue
                    sample[wi, m] = tot
                    pool = []
                    for t in range(T):
                        if unnorm[t] > 0:
                            pool.extend(job.all_ref[c][t].windows)
                    pairs.append((job.row(wi, m), pool))
            self._tmpl = build_template(pairs, sample)
            self._tmpl_key = key
            self._mask = torch.as_tensor(
                np.asarray(active, dtype=np.uint8), device=job.device)
        return self._tmpl.draw(job.pick_rng, job.cfg.epochs)

    def aggregate(self, job: FLJob, round_idx: int, plan: TrainPlan,
                  client_idx: np.ndarray) -> None:
        st = self.state
        if "cfl" in job.cfg.concept_drift_algo_arg:
            if self._cfl_round(job, round_idx, plan, client_idx):
                self.invalidate_plan()
                return  # split: skip this round's aggregation (:141-148)
        job.aggregate(plan, model_mask=self._mask)
        if job.cfg.concept_drift_algo_arg == "hard-r":
            acc = job.train_acc_matrix_rows(list(range(job.n_models)))
            st.cluster(Hooks(job), acc, job.curr_iter, round_idx + 1)
            self.invalidate_plan()

    def _cfl_round(self, job: FLJob, round_idx: int, plan: TrainPlan,
                   client_idx: np.ndarray) -> bool:
        """CFL split check needs each client's raw weight update
        (reference cluster_cfl, FedAvgEnsDataLoader.py:1159-1223).

        Redesigned for scale: the active (worker, model) set is agreed via
        one small all_reduce, then the flat deltas travel as ONE dense
        device tensor all_reduce over RCCL/xGMI (each rank owns disjoint
        rows, sum = gather) — no pickled-object gathers; at ResNet-18 x
        100 clients the old all_gather_object serialized O(pairs x 11M)
        floats through the host per split check. The collective is
        chunked to bound peak traffic per call."""
        st = self.state
        K, P = job.n_models, job.n_params   # module path: full state size
        W = job.n_workers
        nW = len(job.owned_workers)
        reps = job.replicas.reshape(nW, K, P)
        act = torch.zeros(W, K, device=job.device)
        for wi, w in enumerate(job.owned_workers):
            for m in range(K):
                if plan.sample_num[wi, m] > 0:
                    act[w, m] = 1.0
        job.comm.all_reduce_(act)
        act_np = act.cpu().numpy() > 0
        pairs = [(w, m) for w in range(W) for m in range(K)
                 if act_np[w, m]]
        idx_of = {p: i for i, p in enumerate(pairs)}
        deltas = torch.zeros(len(pairs), P, device=job.device)
        for wi, w in enumerate(job.owned_workers):
            for m in range(K):
                if plan.sample_num[wi, m] > 0:
                    deltas[idx_of[(w, m)]] = \
                        reps[wi, m] - job.global_params[m]
        rows_per_chunk = max(1, (256 << 20) // max(4 * P, 1))
        for i0 in range(0, len(pairs), rows_per_chunk):
            job.comm.all_reduce_(deltas[i0:i0 + rows_per_chunk])
        row_of = {(int(client_idx[w]), m): i
                  for i, (w, m) in enumerate(pairs)}
        models_in_use = [m for m in range(K)
                         if np.any(st.get_weights()[job.curr_iter][m] > 0)]
        clients_by_model = {
            m: np.nonzero(st.get_weights()[job.curr_iter][m])[0]
            for m in models_in_use}
        # keep client lists aligned with the updates actually present
        clients_aligned = {
            m: np.array([c for c in clients_by_model[m]
                         if (c, m) in row_of])
            for m in models_in_use}
        # norm + pairwise-cosine statistics on the DEVICE (O(n^2 * P)
        # GEMM — host numpy at ResNet scale was the split check's cost);
        # only the tiny per-cluster matrices cross to the host
        stats = {}
        for m in models_in_use:
            cl = clients_aligned[m]
            if len(cl) == 0:
                continue
            idx = torch.as_tensor([row_of[(int(c), m)] for c in cl],
                                  device=job.device)
            D = deltas[idx]
            norms = D.norm(dim=1)
            mean_norm = float(D.mean(dim=0).norm())
            sims = (D @ D.T) / (torch.outer(norms, norms) + 1e-12)
            stats[m] = (float(norms.max()), mean_norm,
                        sims.cpu().numpy())
        return st.cluster_cfl(Hooks(job), job.curr_iter, round_idx + 1,
                              None, clients_aligned,
                              stats_by_model=stats)

    def test(self, job: FLJob, round_idx: int) -> None:
        cfg = job.cfg
        if round_idx % cfg.frequency_of_the_test == 0 or \
                round_idx == cfg.comm_round - 1:
            st = self.state
            mpc = self._mpc
            if mpc is None:
                # np.argmax over the [K, C] weight matrix costs >100 us at
                # thousands of clients; the assignment only changes at
                # clustering events, so cache it per iteration/recluster
                mpc = self._mpc = st.get_test_model_idx_all(job.curr_iter)
            # softcluster tests train data on the CURRENT-iteration all_data
            # (FedAvgEnsAggregatorSoftCluster.py:227-231)
            tr, te = self.client_eval_views(job, mpc, train_on_view=False)
            job.log_round_stats(round_idx, tr, te)
        if round_idx > (cfg.comm_round - 5):
            job.save_state_pickle("sc_state.pkl", self.state)

    def finalize(self, job: FLJob) -> None:
        job.save_state_pickle("sc_state.pkl", self.state)


# ---------------------------------------------------------------------------
# Single-model baselines (fedavg_cont_one: win-1 / win-2 / all via
# --retrain_data; reference fedml_experiments/distributed/fedavg_cont_one)
# ---------------------------------------------------------------------------
class SingleAlgo(AlgoBase):
    def build_views(self, job: FLJob) -> list:
        return [job.cdata.view(job.cfg.retrain_data)]


# ---------------------------------------------------------------------------
# AUE / AUE-PC
# ---------------------------------------------------------------------------
class AueAlgo(AlgoBase):
    EPS = 1e-20

    def __init__(self, per_client: bool):
        self.per_client = per_client

    def build_views(self, job: FLJob) -> list:
        # model m trains on win-(m+1) (reference AUE_data_loader:20-29)
        n = min(job.curr_iter + 1, job.cfg.ensemble_window)
        return [job.cdata.view(f"win-{m + 1}") for m in range(n)]

    def load_checkpoint(self, job: FLJob) -> None:
        # circular shift: model m loads previous model m-1
        # (main_fedavg.py:342-344)
        if job.curr_iter != 0 and not job.cfg.reset_models and \
                os.path.exists(job.ckpt_path("model_params.pt")):
            mp = torch.load(job.ckpt_path("model_params.pt"))
            for m_idx in range(1, job.n_models):
                if m_idx - 1 in mp:
                    job.global_params[m_idx] = job.packer.flatten(
                        mp[m_idx - 1]).to(job.device)

    def init_iteration(self, job: FLJob) -> None:
        # all models start "perfect" (FedAvgEnsAggregatorAue.py:45-54)
        py = 1.0 / job.dataset.class_num
        mser = (1 - py) ** 2
        M, C = job.n_models, job.cfg.client_num_in_total
        if self.per_client:
            w = np.full((C, M), 1.0 / (mser + self.EPS))
            self.ens_weights = w / w.sum(axis=1, keepdims=True)
        else:
            w = np.full(M, 1.0 / (mser + self.EPS))
            self.ens_weights = w / w.sum()

    def _update_ens_weights(self, job: FLJob) -> None:
        """AUE weights w_m = 1/(MSE_r + MSE_m + eps) over the newest batch;
        newest model gets the perfect score. The reference's index shift
        (enumerate over models[1:] writing ens_weights[m_idx]) is preserved
        verbatim (FedAvgEnsAggregatorAue.py:55-87)."""
        py = 1.0 / job.dataset.class_num
        mser = (1 - py) ** 2
        M, C = job.n_models, job.cfg.client_num_in_total
        # per-(model>=1, client) MSE on the win-1 view train data
        tl = TaskList()
        ids = {}
        for mi in range(1, M):
            for c in range(C):
                tid = tl.new_task()
                ids[(mi, c)] = tid
                if job.comm.owns_client(c) and c in job.view_train_ref[0]:
                    tl.add_windows(tid, mi, job.view_train_ref[0][c].windows)
        res = job.run_eval_dev(job.global_params, tl, want_mse=True)
        job.comm.all_reduce_(res)
        _, total, _, mse = res.cpu().numpy()

        if self.per_client:
            for c in range(C):
                for mi in range(1, M):
                    tid = ids[(mi, c)]
                    msei = mse[tid] / total[tid] if total[tid] else 0.0
                    self.ens_weights[c][mi - 1] = 1.0 / (mser + msei + self.EPS)
                self.ens_weights[c][0] = 1.0 / (mser + self.EPS)
                self.ens_weights[c] = self.ens_weights[c] / \
                    self.ens_weights[c].sum()
        else:
            for mi in range(1, M):
                msei = sum(mse[ids[(mi, c)]] for c in range(C))
                tot = sum(total[ids[(mi, c)]] for c in range(C))
                msei = msei / tot if tot else 0.0
                self.ens_weights[mi - 1] = 1.0 / (mser + msei + self.EPS)
            self.ens_weights[0] = 1.0 / (mser + self.EPS)
            self.ens_weights = self.ens_weights / self.ens_weights.sum()

    def post_aggregate(self, job: FLJob, round_idx: int) -> None:
        if round_idx % 10 == 0 or round_idx > (job.cfg.comm_round - 10):
            self._update_ens_weights(job)

    _test_lists = None

    def _get_test_lists(self, job: FLJob):
        if self._test_lists is None:
            C = job.cfg.client_num_in_total
            clients = range(C) if job.cfg.ci != 1 else range(1)
            tl_tr = TaskList()
            tl_te = TaskList()
            for c in clients:
                t1 = tl_tr.new_task()
                t2 = tl_te.new_task()
                if job.comm.owns_client(c):
                    if c in job.view_train_ref[0]:
                        tl_tr.add_windows(t1, 0,
                                          job.view_train_ref[0][c].windows)
                    if c in job.test_ref:
                        tl_te.add_windows(t2, 0, job.test_ref[c].windows)
            self._test_lists = (tl_tr, job.eval_tensors(tl_tr),
                                tl_te, job.eval_tensors(tl_te))
        return self._test_lists

    def test(self, job: FLJob, round_idx: int) -> None:
        cfg = job.cfg
        if not (round_idx % cfg.frequency_of_the_test == 0 or
                round_idx == cfg.comm_round - 1):
            return
        tl_tr, idx_tr, tl_te, idx_te = self._get_test_lists(job)
        # train: model 0 on view-0 train data (FedAvgEnsAggregatorAue.py:172)
        res_tr = job.run_eval_dev(job.global_params, tl_tr, idx=idx_tr)
        # test: weighted-vote ensemble, batched over all clients (:256-283)
        M = job.n_models
        w = self.ens_weights[:, :M] if self.per_client \
            else self.ens_weights[:M]
        res_te = job.ens_vote_multi(
            torch.as_tensor(w, dtype=torch.float32, device=job.device),
            tl_te, idx_te, mode="hard")
        buf = torch.cat([res_tr, res_te])
        job.comm.all_reduce_(buf)
        trc, trt, trl, tec, tet = buf.cpu().numpy()
        job.log_round_stats(round_idx, (trc, trt, trl),
                            (tec, tet, np.zeros_like(tec)))


# ---------------------------------------------------------------------------
# DriftSurf
# ---------------------------------------------------------------------------
class DriftSurfAlgo(AlgoBase):
    def build_views(self, job: FLJob) -> list:
        cfg = job.cfg
        if job.curr_iter == 0:
            self.state = DriftSurfState(
                delta=driftsurf_delta(cfg.concept_drift_algo_arg,
                                      cfg.dataset_norm))
            views = [job.cdata.view("sel-0") for _ in range(2)]
        else:
            self.state = job.load_state_pickle("ds_state.pkl")
            newest = job.cdata.view("win-1")
            self._score_newest(job, newest)
            views = []
            for key in self.state.get_train_keys():
                sel = ",".join(str(x) for x in self.state.get_train_data(key))
                views.append(job.cdata.view(f"sel-{sel}"))
        job.save_state_pickle("ds_state.pkl", self.state)
        return views

    def _score_newest(self, job: FLJob, view) -> None:
        """run_ds_algo at data-load time (DriftSurf_data_loader:269-314):
        score the pickled models on the newest global batch. One batched
        device sweep for ALL candidate models, sharded by client +
        allreduced (job.score_models_on_segments) — replaces the
        reference's per-(model, client) CPU eager forwards."""
        keys = [k for k in ("pred", "stab", "reac")
                if self.state.models.get(k) is not None]
        flats = [np.asarray(self.state.models[k]) for k in keys]
        correct, total = job.score_models_on_segments(flats, view.train)
        accs = {}
        for i, k in enumerate(keys):
            t = total[i].sum()
            accs[k] = float(correct[i].sum() / t) if t else 0.0

        self.state.run_ds_algo(lambda key: accs.get(key, 0.0),
                               job.curr_iter)

    def load_checkpoint(self, job: FLJob) -> None:
        # handled via ds_state (FedAvgEnsAggregatorDriftSurf.py:45-64)
        st = self.state
        self.test_model_idx = 0
        if job.curr_iter != 0 and not job.cfg.reset_models:
            for idx, key in enumerate(st.get_train_keys()):
                flat = st.models.get(key)
                if flat is not None and idx < job.n_models:
                    job.global_params[idx] = torch.from_numpy(
                        np.asarray(flat, dtype=np.float32)).to(job.device)
        for idx, key in enumerate(st.get_train_keys()):
            if key == st.get_model_key():
                self.test_model_idx = idx

    def test(self, job: FLJob, round_idx: int) -> None:
        cfg = job.cfg
        if round_idx % cfg.frequency_of_the_test == 0 or \
                round_idx == cfg.comm_round - 1:
            C = cfg.client_num_in_total
            mpc = np.full(C, self.test_model_idx)
            tr, te = self.client_eval_views(job, mpc, mpc)
            job.log_round_stats(round_idx, tr, te)

    def _sync_state_models(self, job: FLJob) -> None:
        for idx, key in enumerate(self.state.get_train_keys()):
            if idx < job.n_models:
                self.state.set_model(
                    key, job.global_params[idx].cpu().numpy())

    def aggregate(self, job: FLJob, round_idx: int, plan: TrainPlan,
                  client_idx: np.ndarray) -> None:
        job.aggregate(plan)
        self._sync_state_models(job)
        job.save_state_pickle("ds_state.pkl", self.state)

    def finalize(self, job: FLJob) -> None:
        self._sync_state_models(job)
        job.save_state_pickle("ds_state.pkl", self.state)


# ---------------------------------------------------------------------------
# MultiModelAcc (mmacc) + oracles (mmgeni / mmgeniex)
# ---------------------------------------------------------------------------
class MultiModelAlgo(AlgoBase):
    def __init__(self, variant: str):
        self.variant = variant

    def build_views(self, job: FLJob) -> list:
        cfg = job.cfg
        from ..config import DEFAULT_DELTAS
        if job.curr_iter == 0:
            self.state = MultiModelAccState(
                cfg.client_num_in_total, cfg.concept_num,
                DEFAULT_DELTAS.get(cfg.dataset_norm, 0.1))
        else:
            self.state = job.load_state_pickle("mm_state.pkl")

        if self.variant == "mmacc":
            if job.curr_iter == 0:
                self.state.run_model_select(None, 0)
            else:
                newest = job.cdata.view("win-1")
                self._run_select(job, newest)
        else:
            cps = load_change_points(cfg.data_dir, cfg.change_points)
            if self.variant == "mmgeni":
                self.state.model_select_geni(job.curr_iter, cps,
                                             cfg.time_stretch)
            else:
                self.state.model_select_geniex(job.curr_iter, cps,
                                               cfg.time_stretch)

        views = []
        self.model_ids = []   # original model id per view position
        for m in range(cfg.concept_num):
            td = self.state.get_train_data_by_model(m)
            if td != "":
                views.append(job.cdata.view("clientsel-" + td))
                self.model_ids.append(m)
        job.save_state_pickle("mm_state.pkl", self.state)
        return views

    def _run_select(self, job: FLJob, newest_view) -> None:
        """run_model_select with per-(model, client) scoring on the newest
        local batch (reference FedAvgEnsDataLoader.py:350-390). All
        (model, client) accuracies come from ONE batched device sweep
        (job.score_models_on_segments), replacing the reference's fresh
        per-call CPU eager forwards."""
        present = [m for m in range(job.cfg.concept_num)
                   if self.state.models.get(m) is not None]
        flats = [np.asarray(self.state.models[m]) for m in present]
        correct, total = job.score_models_on_segments(flats,
                                                      newest_view.train)
        acc = {}
        for i, m in enumerate(present):
            for c in range(job.cfg.client_num_in_total):
                if total[i, c] > 0:
                    acc[(m, c)] = float(correct[i, c] / total[i, c])

        self.state.run_model_select(lambda m, c: acc.get((m, c), 0.0),
                                    job.curr_iter)

    def load_checkpoint(self, job: FLJob) -> None:
        cfg = job.cfg
        if job.curr_iter != 0 and not cfg.reset_models and \
                os.path.exists(job.ckpt_path("model_params.pt")):
            mp = torch.load(job.ckpt_path("model_params.pt"))
            if self.variant in ("mmacc", "mmgeni", "mmgeniex") and \
                    job.n_models == 1 and len(mp) == 2:
                # post-transition special case (main_fedavg.py:337-339)
                job.global_params[0] = job.packer.flatten(mp[1]).to(job.device)
            else:
                for m_idx, sd in mp.items():
                    if m_idx < job.n_models:
                        job.global_params[m_idx] = job.packer.flatten(sd).to(
                            job.device)

    def _sync_state_models(self, job: FLJob) -> None:
        for pos, m in enumerate(self.model_ids):
            self.state.set_model(m, job.global_params[pos].cpu().numpy())

    def init_iteration(self, job: FLJob) -> None:
        # record accuracy for the drift detector at iteration 0
        # (reference records via test_on_all_clients; mmacc uses acc_dict)
        if self.variant == "mmacc" and job.curr_iter == 0:
            accs = job.train_acc_matrix_rows([0])
            for c in range(job.cfg.client_num_in_total):
                self.state.set_acc(c, accs[0][c])
            job.save_state_pickle("mm_state.pkl", self.state)

    def test(self, job: FLJob, round_idx: int) -> None:
        cfg = job.cfg
        if round_idx % cfg.frequency_of_the_test == 0 or \
                round_idx == cfg.comm_round - 1:
            C = cfg.client_num_in_total
            pos_of = {m: i for i, m in enumerate(self.model_ids)}
            mpc = np.array([pos_of.get(self.state.get_test_model_idx(c), 0)
                            for c in range(C)])
            tpc = np.array([pos_of.get(self.state.get_train_model_idx(c), 0)
                            for c in range(C)])
            tr, te = self.client_eval_views(job, mpc, tpc)
            job.log_round_stats(round_idx, tr, te)
        if round_idx > (cfg.comm_round - 5):
            self._sync_state_models(job)
            job.save_state_pickle("mm_state.pkl", self.state)

    def finalize(self, job: FLJob) -> None:
        self._sync_state_models(job)
        job.save_state_pickle("mm_state.pkl", self.state)


# ---------------------------------------------------------------------------
# Adaptive-FedAvg
# ---------------------------------------------------------------------------
class AdaAlgo(AlgoBase):
    def build_views(self, job: FLJob) -> list:
        retrain, gran = parse_ada_arg(job.cfg.concept_drift_algo_arg)
        self.update_each_round = (gran == "round")
        return [job.cdata.view(retrain)]

    def init_iteration(self, job: FLJob) -> None:
        if job.curr_iter == 0 or not os.path.exists(
                job.ckpt_path("ada_state.pkl")):
            self.state = AdaState(init_lr=job.cfg.lr)
        else:
            self.state = job.load_state_pickle("ada_state.pkl")
        job.save_state_pickle("ada_state.pkl", self.state)

    def plan(self, job: FLJob, round_idx: int,
             client_idx: np.ndarray) -> TrainPlan:
        # trainer uses the server-determined lr (FedAvgEnsTrainerAda.py:62-65)
        return self.default_plan(job, client_idx,
                                 lr=self.state.current_lr())

    def aggregate(self, job: FLJob, round_idx: int, plan: TrainPlan,
                  client_idx: np.ndarray) -> None:
        job.aggregate(plan)
        theta = job.global_params[0].cpu().numpy()
        if self.update_each_round:
            t = round_idx + job.curr_iter * job.cfg.comm_round
            self.state.update(theta, t)
        else:
            if round_idx == job.cfg.comm_round - 5:
                self.state.update(theta, job.curr_iter)

    def test(self, job: FLJob, round_idx: int) -> None:
        super().test(job, round_idx)
        if round_idx > (job.cfg.comm_round - 5):
            job.save_state_pickle("ada_state.pkl", self.state)


# ---------------------------------------------------------------------------
# Exp / Lin time-decay single-model baselines
# ---------------------------------------------------------------------------
class DecayAlgo(AlgoBase):
    def __init__(self, kind: str):
        self.kind = kind

    def build_views(self, job: FLJob) -> list:
        return [job.cdata.view("win-1")]

    def plan(self, job: FLJob, round_idx: int,
             client_idx: np.ndarray) -> TrainPlan:
        """Single model trained on all history with time-decay sampling:
        probs ~ 2**t (exp) or t+1 (lin); per step sample an iteration then a
        uniform batch within it (FedAvgEnsTrainerExp.py:47-96 / Lin:66)."""
        E = job.cfg.epochs
        T = job.curr_iter + 1
        rows, offs, lens = [], [], []
        nW = len(job.owned_workers)
        sample = np.zeros((nW, 1))
        unnorm = np.array([2.0 ** t if self.kind == "exp" else t + 1.0
                           for t in range(T)])
        probs = unnorm / unnorm.sum()
        for wi, w in enumerate(job.owned_workers):
            c = int(client_idx[w])
            total_batches = sum(len(job.all_ref[c][t].windows)
                                for t in range(T))
            if total_batches == 0:
                continue
            sample[wi, 0] = total_batches
            so, sl = [], []
            for _ in range(E):
                t = int(job.pick_rng.choice(T, p=probs))
                wins = job.all_ref[c][t].windows
                if len(wins) == 0:
                    so.append(0)
                    sl.append(0)   # reference skips the step (:73-74)
                    continue
                o, l = wins[int(job.pick_rng.integers(0, len(wins)))]
                so.append(o)
                sl.append(l)
            rows.append(job.row(wi, 0))
            offs.append(so)
            lens.append(sl)
        return TrainPlan(np.asarray(rows, dtype=np.int64),
                         np.asarray(offs, dtype=np.int64).reshape(-1, E),
                         np.asarray(lens, dtype=np.int64).reshape(-1, E),
                         sample)


# ---------------------------------------------------------------------------
# KUE
# ---------------------------------------------------------------------------
class KueAlgo(AlgoBase):
    def build_views(self, job: FLJob) -> list:
        # poisson bootstrap view per ensemble member
        # (Kue_data_loader:58-72)
        return [job.cdata.view("poisson")
                for _ in range(job.cfg.concept_num)]

    def init_iteration(self, job: FLJob) -> None:
        cfg = job.cfg
        if job.curr_iter == 0 or not os.path.exists(
                job.ckpt_path("kue_state.pkl")):
            self.state = KueState(cfg.concept_num, job.dataset.feature_num,
                                  rng=np.random.RandomState(cfg.dummy_arg))
        else:
            self.state = job.load_state_pickle("kue_state.pkl")
            # re-draw the worst model's mask + params
            # (FedAvgEnsAggregatorKue.py:47-57)
            worst = self.state.get_worst_idx()
            self.state.initialize_mask(worst)
            job.reinit_model(worst)
        job.save_state_pickle("kue_state.pkl", self.state)
        self.ens_weights = np.ones(job.n_models)

    def _masks_tensor(self, job: FLJob) -> torch.Tensor:
        return torch.as_tensor(
            self.state.get_masks().astype(np.float32), device=job.device)

    _row_mask = None

    def plan(self, job: FLJob, round_idx: int,
             client_idx: np.ndarray) -> TrainPlan:
        plan = self.default_plan(job, client_idx)
        if plan.rows.size:
            if self._row_mask is None or \
                    self._row_mask.shape[0] != plan.rows.size:
                masks = self._masks_tensor(job)
                model_of_row = torch.as_tensor(plan.rows % job.n_models,
                                               dtype=torch.int64,
                                               device=job.device)
                self._row_mask = masks[model_of_row].contiguous()
            plan.x_mask = self._row_mask
        return plan

    def aggregate(self, job: FLJob, round_idx: int, plan: TrainPlan,
                  client_idx: np.ndarray) -> None:
        job.aggregate(plan)
        if round_idx % 10 == 0 or round_idx > (job.cfg.comm_round - 10):
            self._update_ens_weights(job)

    def _update_ens_weights(self, job: FLJob) -> None:
        """Cohen's kappa per model from the global confusion matrix over the
        model's own (masked) train view (FedAvgEnsAggregatorKue.py:59-76)."""
        O = job.dataset.class_num
        masks = self._masks_tensor(job)
        for m in range(job.n_models):
            tl = TaskList()
            tid = tl.new_task()
            for c in range(job.cfg.client_num_in_total):
                if job.comm.owns_client(c) and c in job.view_train_ref[m]:
                    tl.add_windows(tid, m, job.view_train_ref[m][c].windows)
            A = job.confusion(
                torch.as_tensor(tl.task_row, dtype=torch.int64,
                                device=job.device),
                torch.as_tensor(tl.task_id, dtype=torch.int64,
                                device=job.device),
                torch.as_tensor(tl.off, dtype=torch.int64, device=job.device),
                torch.as_tensor(tl.ln, dtype=torch.int64, device=job.device),
                1, O, x_mask=masks[m])
            job.comm.all_reduce_(A)
            An = A[0].cpu().numpy()
            n = An.sum()
            left = np.trace(An)
            right = sum(An[i, :].sum() * An[:, i].sum() for i in range(O))
            denom = n ** 2 - right
            self.ens_weights[m] = (n * left - right) / denom if denom else 0.0
        if job.curr_iter != 0:
            self.state.set_worst_idx(int(np.argmin(self.ens_weights)))

    _test_lists = None

    def _get_test_lists(self, job: FLJob):
        if self._test_lists is None:
            C = job.cfg.client_num_in_total
            clients = range(C) if job.cfg.ci != 1 else range(1)
            tl_tr = TaskList()
            tl_te = TaskList()
            for c in clients:
                t1 = tl_tr.new_task()
                t2 = tl_te.new_task()
                if job.comm.owns_client(c):
                    if c in job.view_train_ref[0]:
                        tl_tr.add_windows(t1, 0,
                                          job.view_train_ref[0][c].windows)
                    if c in job.test_ref:
                        tl_te.add_windows(t2, 0, job.test_ref[c].windows)
            self._test_lists = (tl_tr, job.eval_tensors(tl_tr),
                                tl_te, job.eval_tensors(tl_te))
        return self._test_lists

    def test(self, job: FLJob, round_idx: int) -> None:
        cfg = job.cfg
        if not (round_idx % cfg.frequency_of_the_test == 0 or
                round_idx == cfg.comm_round - 1):
            return
        tl_tr, idx_tr, tl_te, idx_te = self._get_test_lists(job)
        res_tr = job.run_eval_dev(job.global_params, tl_tr, idx=idx_tr)
        # soft-vote ensemble with masks, excluding the worst model, batched
        # over all clients (FedAvgEnsAggregatorKue.py:234-264)
        masks = self._masks_tensor(job)
        include = np.array([m != self.state.get_worst_idx() and
                            self.ens_weights[m] > 0
                            for m in range(job.n_models)])
        w = np.where(include, self.ens_weights, 0.0)
        res_te = job.ens_vote_multi(
            torch.as_tensor(w, dtype=torch.float32, device=job.device),
            tl_te, idx_te, mode="soft", masks=masks)
        buf = torch.cat([res_tr, res_te])
        job.comm.all_reduce_(buf)
        trc, trt, trl, tec, tet = buf.cpu().numpy()
        job.log_round_stats(round_idx, (trc, trt, trl),
                            (tec, tet, np.zeros_like(tec)))
        if round_idx > (cfg.comm_round - 5):
            job.save_state_pickle("kue_state.pkl", self.state)

    def finalize(self, job: FLJob) -> None:
        job.save_state_pickle("kue_state.pkl", self.state)
