"""Auxiliary components: optimizer repo, topology, robust aggregation,
partitioning, server optimizer, decentralized/hierarchical FL."""

import numpy as np
import pytest
import torch

from feddrift_amd.comm.robust import add_noise, norm_diff_clipping
from feddrift_amd.comm.topology import (AsymmetricTopologyManager,
                                        SymmetricTopologyManager)
from feddrift_amd.data.partition import partition
from feddrift_amd.engine.decentralized import DecentralizedDSGD, HierarchicalFL
from feddrift_amd.engine.server_opt import ServerOptimizer
from feddrift_amd.models.packed import spec_for
from feddrift_amd.utils.optrepo import OptRepo


def test_optrepo_lookup():
    # same behaviors the reference's only unit test checks
    # (tests/fedml_api/standalone/fedavg/test_optrepo.py)
    assert OptRepo.name2cls("sgd") is torch.optim.SGD
    assert OptRepo.name2cls("Adam") is torch.optim.Adam
    assert OptRepo.name2cls("aDaGRad") is torch.optim.Adagrad
    with pytest.raises(KeyError):
        OptRepo.name2cls("nope")
    assert "lr" in OptRepo.supported_parameters("sgd")


def test_topologies_row_stochastic():
    for mgr in [SymmetricTopologyManager(8, 2), SymmetricTopologyManager(7, 3),
                AsymmetricTopologyManager(8, 3)]:
        assert np.allclose(mgr.topology.sum(axis=1), 1.0)
        assert all(len(mgr.get_out_neighbor_idx_list(i)) >= 2
                   for i in range(mgr.n))
    sym = SymmetricTopologyManager(8, 2)
    # symmetric adjacency
    assert np.array_equal(sym.topology > 0, (sym.topology > 0).T)


def test_norm_clipping():
    g = torch.zeros(1, 10)
    local = torch.zeros(1, 10)
    local[0, 0] = 10.0
    clipped = norm_diff_clipping(local, g, 1.0)
    assert abs(torch.linalg.vector_norm(clipped - g).item() - 1.0) < 1e-5
    # inside the ball: untouched
    local2 = torch.full((1, 10), 0.01)
    assert torch.allclose(norm_diff_clipping(local2, g, 1.0), local2)


def test_add_noise_deterministic_generator():
    gen1 = torch.Generator().manual_seed(5)
    gen2 = torch.Generator().manual_seed(5)
    x = torch.zeros(4, 8)
    assert torch.equal(add_noise(x, 0.1, gen1), add_noise(x, 0.1, gen2))


def test_partition_homo_and_dirichlet():
    labels = np.repeat(np.arange(10), 100)
    homo = partition("homo", labels, 8, seed=0)
    assert sum(len(v) for v in homo.values()) == 1000
    het = partition("hetero", labels, 8, alpha=0.3, seed=0)
    assert sum(len(v) for v in het.values()) == 1000
    # hetero skews label distributions: some client misses some class
    counts = np.array([[np.sum(labels[v] == k) for k in range(10)]
                       for v in het.values()])
    assert (counts == 0).any()
    # all indices unique
    allidx = np.concatenate(list(het.values()))
    assert len(np.unique(allidx)) == 1000


def test_server_optimizer_fedavg_equivalence():
    g = torch.randn(3, 20)
    avg = torch.randn(3, 20)
    upd = torch.tensor([True, False, True])
    so = ServerOptimizer(g, "sgd", lr=1.0)
    g2 = g.clone()
    so.step(g2, avg, upd)
    assert torch.allclose(g2[0], avg[0], atol=1e-6)
    assert torch.allclose(g2[1], g[1])   # non-updated row kept


def _toy_data(n_workers, n=64, d=3, o=2, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n * n_workers, d) * 8
    y = ((x[:, 1] + x[:, 2] > 8).long())
    windows = [[(w * n, n)] for w in range(n_workers)]
    return x, y, windows


def test_decentralized_dsgd_learns_and_converges():
    spec = spec_for("fnn", 3, 2)
    torch.manual_seed(0)
    from feddrift_amd.models.zoo import FeedForwardNN
    from feddrift_amd.models.packed import PackedMLP
    init = PackedMLP(spec).flatten(FeedForwardNN(3, 2, 6).state_dict())
    x, y, wins = _toy_data(6)
    topo = SymmetricTopologyManager(6, 2)
    eng = DecentralizedDSGD(spec, 6, topo, init, x, y, wins, lr=0.05,
                            epochs=2, device=torch.device("cpu"))
    for _ in range(60):
        eng.round()
    # mixing drives consensus; training drives accuracy
    assert eng.consensus_distance() < 1.0
    from feddrift_amd.ops import mlp_torch
    logits = mlp_torch.forward_logits(spec, eng.params[:1], x.unsqueeze(0))
    acc = (logits.squeeze(0).argmax(-1) == y).float().mean().item()
    assert acc > 0.7, acc


def test_pushsum_dsgd_directed_consensus():
    from feddrift_amd.engine.decentralized import PushSumDSGD
    spec = spec_for("fnn", 3, 2)
    torch.manual_seed(0)
    from feddrift_amd.models.zoo import FeedForwardNN
    from feddrift_amd.models.packed import PackedMLP
    init = PackedMLP(spec).flatten(FeedForwardNN(3, 2, 6).state_dict())
    x, y, wins = _toy_data(6)
    topo = AsymmetricTopologyManager(6, 2)       # directed out-neighbors
    eng = PushSumDSGD(spec, 6, topo, init, x, y, wins, lr=0.05,
                      epochs=2, device=torch.device("cpu"))
    # mixing matrix is column-stochastic (push-sum requirement)
    assert torch.allclose(eng.mix.sum(dim=0), torch.ones(6), atol=1e-6)
    for _ in range(60):
        eng.round()
    # total weight mass is conserved by column-stochastic mixing
    assert abs(eng.w.sum().item() - 6.0) < 1e-3
    assert eng.consensus_distance() < 1.0
    from feddrift_amd.ops import mlp_torch
    logits = mlp_torch.forward_logits(spec, eng.estimates()[:1],
                                      x.unsqueeze(0))
    acc = (logits.squeeze(0).argmax(-1) == y).float().mean().item()
    assert acc > 0.7, acc


def test_hierarchical_fl_runs():
    spec = spec_for("fnn", 3, 2)
    torch.manual_seed(0)
    from feddrift_amd.models.zoo import FeedForwardNN
    from feddrift_amd.models.packed import PackedMLP
    init = PackedMLP(spec).flatten(FeedForwardNN(3, 2, 6).state_dict())
    x, y, wins = _toy_data(6)
    h = HierarchicalFL(spec, [[0, 1, 2], [3, 4, 5]], init, x, y, wins,
                       lr=0.05, epochs=2, group_comm_round=2,
                       device=torch.device("cpu"))
    for _ in range(10):
        h.round()
    # after a global sync round, both groups share the model
    assert torch.allclose(h.group_params[0], h.group_params[1])


def test_robust_aggregation_end_to_end(tmp_path, sea_dataset_factory=None):
    """Engine round with clipping+noise enabled still trains."""
    import dataclasses
    from feddrift_amd.comm import Communicator
    from feddrift_amd.config import Config
    from feddrift_amd.data.generators import sample_sea
    from feddrift_amd.data.loader import DriftDataset
    from feddrift_amd.engine.fljob import FLJob
    from feddrift_amd.eval.metrics import MetricLogger

    ds = DriftDataset(data_dir="/nonexistent", dataset="sea", num_client=4)
    rng = np.random.default_rng(0)
    for c in range(4):
        for t in range(3):
            arr = sample_sea(200, 0, rng)
            ds.store.put(c, t, arr[:, :3], arr[:, 3])
    cfg = Config(model="fnn", dataset="sea", data_dir="/nonexistent",
                 client_num_in_total=4, client_num_per_round=4,
                 batch_size=100, lr=0.01, epochs=3, comm_round=30,
                 total_train_iteration=2, curr_train_iteration=1,
                 concept_num=2, concept_drift_algo="single",
                 log_dir=str(tmp_path), report_client=0,
                 robust_norm_bound=5.0, robust_noise=1e-4)
    comm = Communicator()
    logger = MetricLogger(str(tmp_path), enabled=True, to_file=False)
    job = FLJob(cfg, comm, logger, dataset=ds)
    job.run()
    assert logger.series("Test/Acc")[-1] > 0.7
