"""SoftClusterState — the FedDrift / FedDrift-Eager / IFCA / CFL umbrella.

Semantic re-implementation of the reference state machine
(fedml_api/distributed/fedavg_ens/FedAvgEnsDataLoader.py:581-1269), decoupled
from torch modules: all model operations (accuracy matrices, merges,
re-inits) go through an EngineHooks object, so the GPU engine supplies fused
batched kernels and the tests supply numpy fakes.

Weight bookkeeping: train_data_weights[t] is a (model_num x client_num)
matrix; entry [m][c] is the weight of client c's iteration-t data when
training model m (reference :588-589).

Key reference behaviors preserved (file:line into FedAvgEnsDataLoader.py):
  * cluster_init (:616-638): everyone on model 0, or per-client models when
    h_cluster == 'F'.
  * cluster dispatch (:640-669) — hard / hard-r (IFCA), softmax, mmacc
    (FedDrift-Eager), gmm, geni.
  * cluster_hierarchical (:840-978, FedDrift): drift detection -> new
    isolated model marked for W iterations; pooled per-cluster data;
    21-batch-capped cluster-pair accuracy (:929 with the break at
    batch_count > 20 in :1131-1134); distance variant 'A'/'B' (:933-944);
    scipy complete/average linkage + fcluster(t=delta') (:946-951); weighted
    model merge (:1048-1072); LRU eviction with same-iteration veto
    (:1008-1036).
  * CFL split machinery (:1150-1249).
"""

from __future__ import annotations

import random
from typing import Dict, List, Optional, Protocol

import numpy as np
import scipy.cluster.hierarchy as sch
from scipy.spatial.distance import squareform


class EngineHooks(Protocol):
    """What the state machine needs from the engine (or a test fake)."""

    def train_acc_matrix(self, models_in_use: List[int]) -> np.ndarray:
        """[len(models_in_use), client_num] accuracy of each model on each
        client's CURRENT-iteration data (reference :1074-1085)."""
        ...

    def cluster_pair_acc(self, models_in_use: List[int],
                         cluster_batches: Dict[int, list]) -> np.ndarray:
        """[K,K]: acc of model i on cluster j's pooled batch list, capped at
        21 batches per pair (reference :923-931, :1111-1138)."""
        ...

    def merge_models(self, base: int, second: int, w1: float, w2: float) -> None:
        """models[base] <- w1*models[base] + w2*models[second] (:1059-1065)."""
        ...

    def reinit_model(self, m: int) -> None:
        """reinitialize(models[m]) — back to the seeded common init (:1066)."""
        ...

    def copy_model(self, dst: int, src: int) -> None:
        """models[dst].load_state_dict(models[src].state_dict()) (:1033)."""
        ...

    def pooled_cluster_batches(self, weights: Dict[int, np.ndarray],
                               model: int, curr_iter: int) -> list:
        """All batches (opaque handles) of clients x iterations whose weight
        for `model` is 1, pooled client-major then time (reference
        :901-921)."""
        ...

    def log_client(self, key: str, client: int, value, round_idx: int) -> None:
        ...

    def log_summary(self, key: str, value) -> None:
        ...


class SoftClusterState:
    def __init__(self, client_num: int, model_num: int = 2,
                 cluster_alg: str = "softmax_0", mmacc_delta: float = 0.1,
                 softmax_alpha: int = 0, geni_change_points=None,
                 geni_stretch: int = 1, h_delta: float = 0.1,
                 h_deltap: float = 0.1, h_w: int = 1, h_distance: str = "A",
                 h_cluster: str = "C", cfl_gamma: float = 0.1,
                 cfl_retrain: str = "win-1", seed: int = 0):
        self.client_num = client_num
        self.model_num = model_num
        self.train_data_weights: Dict[int, np.ndarray] = {}
        self.cluster_alg = cluster_alg
        self.mmacc_delta = mmacc_delta
        self.mmacc_acc_dict: Dict[int, float] = {}
        self.softmax_alpha = softmax_alpha
        self.geni_change_points = geni_change_points
        self.geni_stretch = geni_stretch
        self.h_delta = h_delta
        self.h_deltap = h_deltap
        self.h_w = h_w
        self.h_distance = h_distance
        self.h_cluster = h_cluster
        # clients -> (model_idx, iteration at which to unmark)
        self.h_marked: Dict[int, tuple] = {}
        self.h_next_free_model = 1
        self.cfl_gamma = cfl_gamma
        self.cfl_retrain = cfl_retrain
        self.cfl_norm = 0.0
        self.cfl_eps1 = 0.0
        self.cfl_eps2 = 10000.0
        # dedicated RNG streams (reference uses the process-global np/python
        # RNGs; we keep deterministic lockstep across ranks instead)
        self._np_rng = np.random.RandomState(seed + 11173)
        self._py_rng = random.Random(seed + 20359)

    # -- initial clustering (reference :616-638) ---------------------------
    def cluster_init(self, hooks: EngineHooks) -> None:
        self.train_data_weights[0] = np.zeros((self.model_num, self.client_num))
        if self.h_cluster == "F":
            # per-client initial models: one model PER CLIENT, so the model
            # cap must cover the fleet (the reference crashes with a raw
            # IndexError here, FedAvgEnsDataLoader.py:630; fail clearly)
            if self.model_num < self.client_num:
                raise ValueError(
                    f"H_*_F (per-client init) needs concept_num >= "
                    f"client_num: got {self.model_num} models for "
                    f"{self.client_num} clients — raise CONCEPT_NUM (the "
                    f"model cap) to at least the client count")
            for c in range(self.client_num):
                self.train_data_weights[0][c][c] = 1.0
            for c in range(self.client_num):
                hooks.log_client("Plurality/CL-{}", c, c, 0)
                hooks.log_summary(f"Contribute/CL-{c}", 1)
            hooks.log_summary("num_models", self.client_num)
            hooks.log_summary("local_models", self.client_num)
            return
        for c in range(self.client_num):
            self.train_data_weights[0][0][c] = 1.0
        for c in range(self.client_num):
            hooks.log_client("Plurality/CL-{}", c, 0, 0)
            hooks.log_summary(f"Contribute/CL-{c}", 1)
        hooks.log_summary("num_models", 1)
        hooks.log_summary("local_models", 0)

    # -- per-round / per-iteration clustering dispatch (reference :640-669) -
    def cluster(self, hooks: EngineHooks, acc_matrix: np.ndarray,
                curr_iter: int, round_idx: int) -> None:
        if self.cluster_alg in ("hard", "hard-r"):
            self.cluster_hard(acc_matrix, curr_iter)
        elif "softmax" in self.cluster_alg:
            self.cluster_softmax(acc_matrix, curr_iter)
        elif "mmacc" in self.cluster_alg:
            if round_idx == 0:
                self.cluster_mmacc(hooks, acc_matrix, curr_iter)
            else:
                self.cluster_hard_among_existing(acc_matrix, curr_iter)
        elif self.cluster_alg == "gmm":
            self.cluster_gmm(acc_matrix, curr_iter)
        elif self.cluster_alg == "geni":
            if round_idx == 0:
                self.cluster_geni(curr_iter)
        else:
            raise NameError("cluster alg")
        for c in range(self.client_num):
            hooks.log_client("Plurality/CL-{}", c,
                             self.get_test_model_idx(curr_iter, c), round_idx)
            if "softmax" in self.cluster_alg:
                hooks.log_client(
                    "Weight-All/CL-{}", c,
                    np.array2string(self.train_data_weights[curr_iter][:, c]),
                    round_idx)

    def cluster_hard(self, acc_matrix: np.ndarray, curr_iter: int) -> None:
        self.train_data_weights[curr_iter] = np.zeros(
            (self.model_num, self.client_num))
        best = np.argmax(acc_matrix, axis=0)
        for c in range(self.client_num):
            self.train_data_weights[curr_iter][best[c]][c] = 1.0

    def cluster_softmax(self, acc_matrix: np.ndarray, curr_iter: int) -> None:
        from scipy.special import softmax as sp_softmax
        self.train_data_weights[curr_iter] = sp_softmax(
            acc_matrix * (2 ** self.softmax_alpha), axis=0)

    def cluster_mmacc(self, hooks: EngineHooks, acc_matrix: np.ndarray,
                      curr_iter: int) -> None:
        """FedDrift-Eager on a full-K acc matrix (reference :685-721)."""
        models_in_use = [m for m in range(self.model_num)
                         if any(np.any(self.train_data_weights[t][m] > 0)
                                for t in range(curr_iter))]
        self.train_data_weights[curr_iter] = np.zeros(
            (self.model_num, self.client_num))
        for c in range(self.client_num):
            best = models_in_use[int(np.argmax(acc_matrix[models_in_use, c]))]
            self.train_data_weights[curr_iter][best][c] = 1.0
        next_free = -42
        for c in range(self.client_num):
            bi = int(np.argmax(acc_matrix[models_in_use, c]))
            best = models_in_use[bi]
            newest_acc = acc_matrix[best][c]
            if self.mmacc_acc_dict[c] - acc_matrix[best][c] > self.mmacc_delta:
                if next_free == -42:
                    next_free = self.find_unused_model_lru(curr_iter)
                if next_free != -1:
                    self.train_data_weights[curr_iter][:, c] = 0.0
                    self.train_data_weights[curr_iter][next_free][c] = 1.0
            self.set_acc(c, newest_acc)
        self.log_models(hooks, curr_iter)

    def cluster_mmacc2(self, hooks: EngineHooks, curr_iter: int) -> None:
        """FedDrift-Eager at iteration start (reference :796-837)."""
        acc_matrix = hooks.train_acc_matrix(list(range(self.model_num)))
        models_in_use = [m for m in range(self.model_num)
                         if any(np.any(self.train_data_weights[t][m] > 0)
                                for t in range(curr_iter))]
        self.train_data_weights[curr_iter] = np.zeros(
            (self.model_num, self.client_num))
        for c in range(self.client_num):
            best = models_in_use[int(np.argmax(acc_matrix[models_in_use, c]))]
            self.train_data_weights[curr_iter][best][c] = 1.0
        next_free = -42
        for c in range(self.client_num):
            bi = int(np.argmax(acc_matrix[models_in_use, c]))
            best = models_in_use[bi]
            newest_acc = acc_matrix[best][c]
            if self.mmacc_acc_dict[c] - acc_matrix[best][c] > self.mmacc_delta:
                if next_free == -42:
                    next_free = self.find_unused_model_lru(
                        curr_iter, use_models=True, original_model=best,
                        hooks=hooks)
                if next_free != -1:
                    self.train_data_weights[curr_iter][:, c] = 0.0
                    self.train_data_weights[curr_iter][next_free][c] = 1.0
            self.set_acc(c, newest_acc)
        for c in range(self.client_num):
            hooks.log_client("Plurality/CL-{}", c,
                             self.get_test_model_idx(curr_iter, c), 0)
        self.log_models(hooks, curr_iter)

    def cluster_gmm(self, acc_matrix: np.ndarray, curr_iter: int) -> None:
        from sklearn.mixture import GaussianMixture
        self.train_data_weights[curr_iter] = np.zeros(
            (self.model_num, self.client_num))
        gm = GaussianMixture(n_components=2, random_state=0).fit(acc_matrix.T)
        probs = gm.predict_proba(acc_matrix.T).T
        if gm.means_[0][0] > gm.means_[0][1]:
            self.train_data_weights[curr_iter][0] = probs[0]
            self.train_data_weights[curr_iter][1] = probs[1]
        else:
            self.train_data_weights[curr_iter][0] = probs[1]
            self.train_data_weights[curr_iter][1] = probs[0]

    def cluster_geni(self, curr_iter: int) -> None:
        self.train_data_weights[curr_iter] = np.zeros(
            (self.model_num, self.client_num))
        for c in range(self.client_num):
            best = int(self.geni_change_points[
                curr_iter // self.geni_stretch][c])
            self.train_data_weights[curr_iter][best][c] = 1.0

    def cluster_hard_among_existing(self, acc_matrix: np.ndarray,
                                    curr_iter: int) -> None:
        models_in_use = [m for m in range(self.model_num)
                         if np.any(self.train_data_weights[curr_iter][m] > 0)]
        self.train_data_weights[curr_iter] = np.zeros(
            (self.model_num, self.client_num))
        for c in range(self.client_num):
            best = models_in_use[int(np.argmax(acc_matrix[models_in_use, c]))]
            self.train_data_weights[curr_iter][best][c] = 1.0

    # -- FedDrift hierarchical clustering (reference :840-978) -------------
    def cluster_hierarchical(self, hooks: EngineHooks, curr_iter: int) -> None:
        # FedDrift-C ('E'): keep only one of the models created last iter
        if self.h_cluster == "E":
            marked_models = [m for (m, t) in self.h_marked.values()]
            if marked_models:
                keep = self._np_rng.choice(marked_models)
                for mm in marked_models:
                    if mm != keep:
                        hooks.reinit_model(mm)
                        self.set_weights_zero_model(mm)

        self.update_marking(curr_iter)

        marked_models = [m for (m, t) in self.h_marked.values()]
        models_in_use = [m for m in range(self.model_num)
                         if any(np.any(self.train_data_weights[t][m] > 0)
                                for t in range(curr_iter))
                         and m not in marked_models]

        acc_matrix = hooks.train_acc_matrix(models_in_use)

        self.train_data_weights[curr_iter] = np.zeros(
            (self.model_num, self.client_num))

        # marked clients stay on their isolated local model
        for c, (m, t) in self.h_marked.items():
            self.train_data_weights[curr_iter][m][c] = 1.0

        # everyone else provisionally on their best existing model (so LRU
        # does not evict it)
        for c in range(self.client_num):
            if c not in self.h_marked:
                best = models_in_use[int(np.argmax(acc_matrix[:, c]))]
                self.train_data_weights[curr_iter][best][c] = 1.0

        # drift detection: leave for a fresh isolated model on acc drop
        for c in range(self.client_num):
            if c in self.h_marked:
                continue
            bi = int(np.argmax(acc_matrix[:, c]))
            best = models_in_use[bi]
            newest_acc = acc_matrix[bi][c]
            if self.mmacc_acc_dict[c] - acc_matrix[bi][c] > self.h_delta:
                next_free = self.find_unused_model_lru(
                    curr_iter, use_models=True, original_model=best,
                    hooks=hooks)
                if next_free != -1:
                    best = next_free
                    self.h_marked[c] = (best, curr_iter + self.h_w)
                    self.train_data_weights[curr_iter][:, c] = 0.0
                    self.train_data_weights[curr_iter][best][c] = 1.0
            self.set_acc(c, newest_acc)

        if len(models_in_use) > 1:
            # pooled data per cluster, shuffled (reference :901-921)
            cluster_data = {}
            for m in models_in_use:
                batches = hooks.pooled_cluster_batches(
                    self.train_data_weights, m, curr_iter)
                self._py_rng.shuffle(batches)
                cluster_data[m] = batches

            cluster_acc = hooks.cluster_pair_acc(models_in_use, cluster_data)

            k = len(models_in_use)
            dist = np.zeros((k, k))
            for i in range(k):
                for j in range(k):
                    if self.h_distance == "A":
                        dist[i][j] = max(cluster_acc[i][i] - cluster_acc[i][j],
                                         cluster_acc[j][j] - cluster_acc[j][i],
                                         0)
                    elif self.h_distance == "B":
                        dist[i][j] = max(cluster_acc[i][i] - cluster_acc[j][i],
                                         cluster_acc[j][j] - cluster_acc[i][j],
                                         0)

            method = "average" if self.h_cluster == "D" else "complete"
            Z = sch.linkage(squareform(dist), method=method)
            T = sch.fcluster(Z, t=self.h_deltap, criterion="distance")

            clusters: Dict[int, List[int]] = {}
            for i in range(k):
                clusters.setdefault(T[i], []).append(models_in_use[i])

            merged_log = ["(" + ", ".join(str(e) for e in grp) + ")"
                          for grp in clusters.values() if len(grp) > 1]
            if merged_log:
                hooks.log_summary("Merge", ", ".join(merged_log))

            for grp in clusters.values():
                base = grp[0]
                for second in grp[1:]:
                    self.merge(hooks, curr_iter, base, second)

        for c in range(self.client_num):
            hooks.log_client("Plurality/CL-{}", c,
                             self.get_test_model_idx(curr_iter, c), 0)
        self.log_models(hooks, curr_iter)

    # -- model-slot allocation policies (reference :981-1036) --------------
    def find_unused_model_capped(self) -> int:
        if self.h_next_free_model < self.model_num:
            nf = self.h_next_free_model
            self.h_next_free_model += 1
            return nf
        return -1

    def find_unused_model_lru(self, curr_iter: int, use_models: bool = False,
                              original_model: int = 0,
                              hooks: Optional[EngineHooks] = None) -> int:
        if self.h_next_free_model < self.model_num:
            next_free = self.h_next_free_model
            self.h_next_free_model += 1
        else:
            time_last_used = -1 * np.ones(self.model_num)
            for t in range(curr_iter + 1):
                for m in range(self.model_num):
                    if t in self.train_data_weights and \
                            np.any(self.train_data_weights[t][m]):
                        time_last_used[m] = t
            lru = np.where(time_last_used == time_last_used.min())[0]
            next_free = int(self._np_rng.choice(lru))
            if time_last_used[next_free] == curr_iter:
                return -1
            self.set_weights_zero_model(next_free)
        if use_models and hooks is not None:
            # new model starts from the parameters of the drifted client's
            # previous best model (reference :1031-1033)
            hooks.copy_model(next_free, original_model)
        return next_free

    def update_marking(self, curr_iter: int) -> None:
        for c in [c for c, (m, t) in self.h_marked.items() if t == curr_iter]:
            del self.h_marked[c]

    def merge(self, hooks: EngineHooks, curr_iter: int, base: int,
              second: int) -> None:
        w1 = w2 = 0.0
        for c in range(self.client_num):
            for t in range(curr_iter + 1):
                w1 += self.train_data_weights[t][base][c]
                w2 += self.train_data_weights[t][second][c]
        s = w1 + w2
        hooks.merge_models(base, second, w1 / s, w2 / s)
        hooks.reinit_model(second)
        for c in range(self.client_num):
            for t in range(curr_iter + 1):
                self.train_data_weights[t][base][c] += \
                    self.train_data_weights[t][second][c]
        self.set_weights_zero_model(second)

    # -- CFL inside softcluster (reference :1150-1249) ---------------------
    def cluster_cfl_init(self, hooks: EngineHooks, curr_iter: int) -> None:
        self.train_data_weights[curr_iter] = np.copy(
            self.train_data_weights[curr_iter - 1])
        if self.cfl_retrain == "win-1":
            self.set_weights_win1(curr_iter)
        for c in range(self.client_num):
            hooks.log_client("Plurality/CL-{}", c,
                             self.get_test_model_idx(curr_iter, c), 0)

    def cluster_cfl(self, hooks: EngineHooks, curr_iter: int, round_idx: int,
                    weight_updates_by_model: Dict[int, List[np.ndarray]],
                    clients_by_model: Dict[int, np.ndarray],
                    stats_by_model: Optional[Dict[int, tuple]] = None
                    ) -> bool:
        """One CFL round: binary split check per active cluster.

        weight_updates_by_model[m] = flattened (local - global) updates of
        the clients (with data) in cluster m, same order as
        clients_by_model[m] (reference :1159-1223). The engine may pass
        stats_by_model[m] = (max_norm, mean_norm, sims) precomputed on
        the GPU (norms + the pairwise cosine matrix are O(n^2 * P) — at
        ResNet scale that is a GEMM, not host numpy work); the raw
        updates are then not consulted at all.
        """
        did_split = False
        for model_idx, clients in clients_by_model.items():
            if stats_by_model is not None:
                if model_idx not in stats_by_model:
                    continue
                max_norm, mean_norm, sims_pre = stats_by_model[model_idx]
            else:
                updates = weight_updates_by_model[model_idx]
                if not updates:
                    continue
                stack = np.stack(updates)
                norms = np.linalg.norm(stack, axis=1)
                max_norm = float(np.max(norms))
                mean_norm = float(np.linalg.norm(np.mean(stack, axis=0)))
                sims_pre = None
            if mean_norm > self.cfl_norm:
                self.cfl_norm = mean_norm
                self.cfl_eps1 = self.cfl_norm / 10.0
                self.cfl_eps2 = 6 * self.cfl_eps1
            else:
                if mean_norm < self.cfl_eps1 and max_norm > self.cfl_eps2:
                    sims = sims_pre if sims_pre is not None \
                        else self._pairwise_cos(stack)
                    cl1, cl2 = self._bipartition(sims)
                    alpha_cross = max(max(sims[i, j] for j in cl2)
                                      for i in cl1)
                    if ((1 - alpha_cross) / 2.0) ** 0.5 > self.cfl_gamma:
                        next_free = self.find_unused_model_capped()
                        if next_free != -1:
                            did_split = True
                            hooks.reinit_model(model_idx)
                            self.train_data_weights[curr_iter][model_idx] = \
                                np.zeros(self.client_num)
                            for i in cl1:
                                self.train_data_weights[curr_iter][
                                    model_idx][clients[i]] = 1.0
                            for i in cl2:
                                self.train_data_weights[curr_iter][
                                    next_free][clients[i]] = 1.0
        if did_split:
            for c in range(self.client_num):
                hooks.log_client("Plurality/CL-{}", c,
                                 self.get_test_model_idx(curr_iter, c),
                                 round_idx)
            if self.cfl_retrain == "all":
                for t in range(curr_iter):
                    self.train_data_weights[t] = np.copy(
                        self.train_data_weights[curr_iter])
        return did_split

    @staticmethod
    def _pairwise_cos(stack: np.ndarray) -> np.ndarray:
        norms = np.linalg.norm(stack, axis=1)
        dots = stack @ stack.T
        return dots / (np.outer(norms, norms) + 1e-12)

    @staticmethod
    def _bipartition(S: np.ndarray):
        from sklearn.cluster import AgglomerativeClustering
        try:
            cl = AgglomerativeClustering(metric="precomputed",
                                         linkage="complete").fit(-S)
        except TypeError:  # older sklearn uses affinity=
            cl = AgglomerativeClustering(affinity="precomputed",
                                         linkage="complete").fit(-S)
        return (np.argwhere(cl.labels_ == 0).flatten(),
                np.argwhere(cl.labels_ == 1).flatten())

    # -- bookkeeping (reference :723-764, :1253-1269) ----------------------
    def log_models(self, hooks: EngineHooks, curr_iter: int) -> None:
        num_models = 0
        if self.h_cluster == "E":
            for m in range(self.model_num):
                if any(np.any(self.train_data_weights[t][m] > 0)
                       for t in range(curr_iter)):
                    num_models += 1
            if self.h_marked:
                num_models += 1
        else:
            for m in range(self.model_num):
                if any(np.any(self.train_data_weights[t][m] > 0)
                       for t in range(curr_iter + 1)):
                    num_models += 1
        hooks.log_summary("num_models", num_models)

        trained_by = {m: set() for m in range(self.model_num)}
        for t in range(curr_iter + 1):
            if t not in self.train_data_weights:
                continue
            for m in range(self.model_num):
                for c in range(self.client_num):
                    if self.train_data_weights[t][m][c] > 0:
                        trained_by[m].add(c)
        local_models = 0
        for m in list(trained_by):
            if len(trained_by[m]) == 1:
                local_models += 1
                del trained_by[m]
        hooks.log_summary("local_models", local_models)
        for c in range(self.client_num):
            hooks.log_summary(
                f"Contribute/CL-{c}",
                sum(1 for clients in trained_by.values() if c in clients))

    def set_acc(self, client: int, acc: float) -> None:
        self.mmacc_acc_dict[client] = acc

    def get_test_model_idx(self, curr_iter: int, client_idx: int) -> int:
        return int(np.argmax(self.train_data_weights[curr_iter][:, client_idx]))

    def get_test_model_idx_all(self, curr_iter: int) -> np.ndarray:
        """Vectorized per-client test-model assignment (argmax of each
        weight column) — the per-round per-client python loop is O(C) and
        dominates host time at thousands of clients."""
        return np.argmax(self.train_data_weights[curr_iter], axis=0)

    def get_weights(self) -> Dict[int, np.ndarray]:
        return self.train_data_weights

    def set_weights_win1(self, curr_iter: int) -> None:
        for t in range(curr_iter):
            self.train_data_weights[t] = np.zeros(
                (self.model_num, self.client_num))

    def set_weights_zero_model(self, m_idx: int) -> None:
        for t in self.train_data_weights:
            self.train_data_weights[t][m_idx] = np.zeros(self.client_num)
