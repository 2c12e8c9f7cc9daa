"""Module execution path (CNN / ResNet): numerics vs the fused MLP path,
and end-to-end drift timelines on convolutional models."""

import dataclasses
import os

import numpy as np
import pytest
import torch

from feddrift_amd.comm import Communicator
from feddrift_amd.config import Config
from feddrift_amd.data.generators import (sample_cifar, sample_femnist,
                                           sample_mnist)
from feddrift_amd.data.loader import DriftDataset
from feddrift_amd.engine.fljob import FLJob, TrainPlan
from feddrift_amd.engine.timeline import clean_state_files
from feddrift_amd.eval.metrics import MetricLogger
from feddrift_amd.models.generic_packer import ModulePacker
from feddrift_amd.models.packed import PackedMLP, spec_for
from feddrift_amd.models.zoo import FeedForwardNN, create_model
from feddrift_amd.ops import mlp_torch
from feddrift_amd.ops.module_engine import ModuleEngine


def test_module_engine_matches_mlp_path():
    """Training an FNN through the generic module engine must match the
    batched MLP op exactly (same Adam math, same batches)."""
    torch.manual_seed(0)
    d, o, h = 4, 3, 8
    spec = spec_for("fnn", d, o)
    mlp_packer = PackedMLP(spec)
    model = FeedForwardNN(d, o, h)
    flat0 = mlp_packer.flatten(model.state_dict())

    n, E = 200, 4
    x = torch.rand(n, d) * 5
    y = torch.randint(0, o, (n,))
    off = torch.tensor([[0, 50, 100, 150]])
    ln = torch.full((1, E), 50)

    # MLP path
    p_ref = flat0.unsqueeze(0).clone()
    opt = mlp_torch.make_opt_state("adam", 1, spec.n_params, 0.01, 0.001,
                                   "cpu")
    mlp_torch.train_fused(spec, p_ref, torch.tensor([0]), x, y, off, ln, opt)

    # module path (generic packer flattens the same state_dict layout)
    packer = ModulePacker(model)
    eng = ModuleEngine(model, packer, torch.device("cpu"))
    glob = flat0.unsqueeze(0).clone()
    reps = torch.zeros(1, packer.n_params)
    mopt = eng.make_opt_state("adam", 1, 0.01, 0.001)
    plan = TrainPlan(np.array([0]), off.numpy(), ln.numpy(),
                     np.ones((1, 1)))
    eng.train(glob, reps, plan, mopt, x, y, n_models=1)
    assert torch.allclose(reps[0], p_ref[0], atol=1e-6), \
        (reps[0] - p_ref[0]).abs().max()


def _mini_dataset(dataset, sampler, n_clients=3, iters=3, n=96, seed=0):
    ds = DriftDataset(data_dir="/nonexistent", dataset=dataset,
                      num_client=n_clients)
    rng = np.random.default_rng(seed)
    for c in range(n_clients):
        for t in range(iters + 1):
            arr = sampler(n, 0 if t < 2 else 1, rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])
    return ds


def test_cnn_drift_timeline(tmp_path):
    ds = _mini_dataset("MNIST", sample_mnist)
    comm = Communicator()
    cfg = Config(model="cnn", dataset="MNIST", data_dir="/nonexistent",
                 client_num_in_total=3, client_num_per_round=3,
                 batch_size=48, lr=0.01, epochs=3, comm_round=4,
                 total_train_iteration=2, concept_num=2,
                 concept_drift_algo="softcluster",
                 concept_drift_algo_arg="mmacc_06", log_dir=str(tmp_path),
                 report_client=0)
    clean_state_files(cfg)
    accs = []
    for it in range(2):
        icfg = dataclasses.replace(cfg, curr_train_iteration=it)
        logger = MetricLogger(str(tmp_path), enabled=True, to_file=False)
        job = FLJob(icfg, comm, logger, dataset=ds)
        job.run()
        accs.append(logger.mean("Test/Acc"))
    # prototypes are well separated: the CNN should beat chance (0.1)
    # within a few rounds
    assert accs[-1] > 0.3, accs
    assert os.path.exists(str(tmp_path / "model_params.pt"))


def test_femnist_cnn62_shapes():
    m = create_model("cnn", 62, 784)
    out = m(torch.rand(4, 784))
    assert out.shape == (4, 62)


def test_resnet_forward_backward():
    m = create_model("resnet", 10, 3072)
    x = torch.rand(2, 3072)
    y = torch.tensor([1, 3])
    loss = torch.nn.functional.cross_entropy(m(x), y)
    loss.backward()
    assert np.isfinite(loss.item())
    # packer round-trip
    p = ModulePacker(m)
    flat = p.flatten(m.state_dict())
    assert flat.numel() == p.n_params
    sd = p.unflatten(flat)
    assert set(sd.keys()) == set(m.state_dict().keys())
    # buffers included (BN running stats), params subset marked
    assert p.n_train_params < p.n_params


def test_vmap_engine_matches_sequential():
    """Batched vmap training must match the sequential module engine for a
    deterministic (dropout-free) module."""
    from feddrift_amd.ops.module_vmap import VmapEngine, vmap_compatible
    torch.manual_seed(3)
    model = FeedForwardNN(4, 3, 8)
    assert vmap_compatible(model)
    packer = ModulePacker(model)
    P = packer.n_params
    dev = torch.device("cpu")

    n, G, E, K = 240, 4, 3, 2
    x = torch.rand(n, 4) * 5
    y = torch.randint(0, 3, (n,))
    glob = torch.stack([packer.flatten(FeedForwardNN(4, 3, 8).state_dict())
                        for _ in range(K)])
    offs = np.array([[0, 60, 120], [60, 0, 180], [120, 180, 0],
                     [180, 60, 120]])
    lens = np.full((G, E), 60)
    lens[2, 1] = 0   # a skipped step
    plan = TrainPlan(np.arange(G), offs, lens, np.ones((2, K)))

    eng_v = VmapEngine(FeedForwardNN(4, 3, 8), ModulePacker(model), dev)
    reps_v = torch.zeros(G, P)
    opt_v = eng_v.make_opt_state("adam", G, 0.01, 0.001)
    eng_v.train(glob.clone(), reps_v, plan, opt_v, x, y, n_models=K)

    eng_s = ModuleEngine(FeedForwardNN(4, 3, 8), ModulePacker(model), dev)
    reps_s = torch.zeros(G, P)
    opt_s = eng_s.make_opt_state("adam", G, 0.01, 0.001)
    # sequential engine's opt covers only trainable params = all here
    eng_s.train(glob.clone(), reps_s, plan, opt_s, x, y, n_models=K)

    assert torch.allclose(reps_v, reps_s, atol=1e-5), \
        (reps_v - reps_s).abs().max()
    assert torch.equal(opt_v["t"], opt_s["t"])


def test_vmap_engine_eval_matches():
    from feddrift_amd.ops.module_vmap import VmapEngine
    torch.manual_seed(4)
    model = FeedForwardNN(3, 2, 6)
    packer = ModulePacker(model)
    dev = torch.device("cpu")
    eng_v = VmapEngine(FeedForwardNN(3, 2, 6), packer, dev)
    eng_s = ModuleEngine(FeedForwardNN(3, 2, 6), packer, dev)
    params = torch.stack([packer.flatten(FeedForwardNN(3, 2, 6).state_dict())
                          for _ in range(3)])
    x = torch.rand(400, 3) * 8
    y = (x[:, 1] + x[:, 2] > 8).long()
    tr = torch.tensor([0, 1, 2, 0])
    ti = torch.tensor([0, 1, 2, 1])
    wo = torch.tensor([0, 100, 200, 300])
    wl = torch.tensor([100, 100, 100, 100])
    a = eng_v.eval_tasks_stacked(params, tr, ti, wo, wl, 3, want_mse=True,
                                 x_arena=x, y_arena=y)
    b = eng_s.eval_tasks_stacked(params, tr, ti, wo, wl, 3, want_mse=True,
                                 x_arena=x, y_arena=y)
    assert torch.allclose(a, b, atol=1e-4), (a - b).abs().max()

def _bn_model():
    from feddrift_amd.models.resnet import FlatImageModel
    torch.manual_seed(9)
    backbone = torch.nn.Sequential(
        torch.nn.Conv2d(1, 4, 3, padding=1), torch.nn.BatchNorm2d(4),
        torch.nn.ReLU(), torch.nn.Flatten(), torch.nn.Linear(4 * 6 * 6, 3))
    return FlatImageModel(backbone, (1, 6, 6))


def test_vmap_engine_batchnorm_matches_sequential():
    """BN models run on the vmap engine too: batched train-mode BN updates
    each pair's running stats in place and must match the sequential
    engine's eager per-pair training (params AND buffers)."""
    from feddrift_amd.ops.module_vmap import VmapEngine, vmap_compatible
    model = _bn_model()
    assert vmap_compatible(model)
    packer = ModulePacker(model)
    assert packer.n_train_params < packer.n_params   # BN buffers present
    dev = torch.device("cpu")

    n, G, E, K = 240, 4, 3, 2
    torch.manual_seed(11)
    x = torch.rand(n, 36)
    y = torch.randint(0, 3, (n,))
    glob = torch.stack([packer.flatten(_bn_model().state_dict())
                        for _ in range(K)])
    offs = np.array([[0, 60, 120], [60, 0, 180], [120, 180, 0],
                     [180, 60, 120]])
    lens = np.full((G, E), 60)
    lens[1, 2] = 0   # a skipped step
    plan = TrainPlan(np.arange(G), offs, lens, np.ones((2, K)))

    eng_v = VmapEngine(_bn_model(), packer, dev)
    reps_v = torch.zeros(G, packer.n_params)
    opt_v = eng_v.make_opt_state("adam", G, 0.01, 0.001)
    eng_v.train(glob.clone(), reps_v, plan, opt_v, x, y, n_models=K)

    eng_s = ModuleEngine(_bn_model(), ModulePacker(model), dev)
    reps_s = torch.zeros(G, packer.n_params)
    opt_s = eng_s.make_opt_state("adam", G, 0.01, 0.001)
    eng_s.train(glob.clone(), reps_s, plan, opt_s, x, y, n_models=K)

    assert torch.allclose(reps_v, reps_s, atol=1e-4), \
        (reps_v - reps_s).abs().max()

    # eval parity (uses the trained running stats through batched views)
    tr = torch.tensor([0, 1, 2, 3])
    ti = torch.tensor([0, 1, 2, 0])
    wo = torch.tensor([0, 60, 120, 180])
    wl = torch.tensor([60, 60, 60, 60])
    a = eng_v.eval_tasks_stacked(reps_v, tr, ti, wo, wl, 3, x_arena=x,
                                 y_arena=y)
    b = eng_s.eval_tasks_stacked(reps_s, tr, ti, wo, wl, 3, x_arena=x,
                                 y_arena=y)
    assert torch.allclose(a, b, atol=1e-4), (a - b).abs().max()


def test_resnet_timeline_uses_vmap(tmp_path, monkeypatch):
    """End-to-end: a ResNet drift timeline routes through the vmap engine
    when FEDDRIFT_VMAP_BN=1 (BN supported) and still learns."""
    monkeypatch.setenv("FEDDRIFT_VMAP_BN", "1")
    from feddrift_amd.ops.module_vmap import VmapEngine
    ds = _mini_dataset("cifar", sample_cifar, n=48)
    comm = Communicator()
    cfg = Config(model="resnet", dataset="cifar", data_dir="/nonexistent",
                 client_num_in_total=3, client_num_per_round=3,
                 batch_size=24, lr=0.01, epochs=2, comm_round=2,
                 total_train_iteration=1, concept_num=2,
                 concept_drift_algo="softcluster",
                 concept_drift_algo_arg="mmacc_06", log_dir=str(tmp_path),
                 report_client=0)
    clean_state_files(cfg)
    logger = MetricLogger(str(tmp_path), enabled=True, to_file=False)
    job = FLJob(cfg, comm, logger, dataset=ds)
    assert isinstance(job.mod_engine, VmapEngine)
    job.run()
    assert np.isfinite(logger.mean("Test/Acc"))


def test_cfl_split_on_module_path(tmp_path):
    """CFL's weight-update gathering must work on the module path too
    (full-state flat rows, not the MLP spec)."""
    ds = _mini_dataset("MNIST", sample_mnist, n=64)
    comm = Communicator()
    cfg = Config(model="cnn", dataset="MNIST", data_dir="/nonexistent",
                 client_num_in_total=3, client_num_per_round=3,
                 batch_size=32, lr=0.05, epochs=2, comm_round=3,
                 total_train_iteration=1, concept_num=2,
                 concept_drift_algo="softcluster",
                 concept_drift_algo_arg="cfl_0.2_all",
                 log_dir=str(tmp_path), report_client=0)
    clean_state_files(cfg)
    logger = MetricLogger(str(tmp_path), enabled=True, to_file=False)
    job = FLJob(cfg, comm, logger, dataset=ds)
    job.run()
    assert np.isfinite(logger.mean("Test/Acc"))


def test_cnn_hip_engine_not_selected_on_cpu(tmp_path):
    """CPU runs stay on the vmap engine (the HIP kernels need a GPU);
    use_hip_kernels='never' must also force the torch path on any
    device — the escape hatch for A/B runs."""
    import numpy as np
    from feddrift_amd.comm import Communicator
    from feddrift_amd.config import Config
    from feddrift_amd.data.generators import sample_mnist
    from feddrift_amd.data.loader import DriftDataset
    from feddrift_amd.engine.fljob import FLJob
    from feddrift_amd.eval.metrics import MetricLogger
    from feddrift_amd.ops.module_vmap import VmapEngine
    ds = DriftDataset(data_dir="/nonexistent", dataset="MNIST",
                      num_client=2)
    rng = np.random.default_rng(0)
    for c in range(2):
        for t in range(3):
            arr = sample_mnist(30, 0, rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])
    cfg = Config(model="cnn", dataset="MNIST", data_dir="/nonexistent",
                 client_num_in_total=2, client_num_per_round=2,
                 batch_size=15, epochs=1, comm_round=1,
                 total_train_iteration=2, curr_train_iteration=1,
                 concept_num=2, concept_drift_algo="softcluster",
                 concept_drift_algo_arg="mmacc_06", bench_mode=1,
                 report_client=0, log_dir=str(tmp_path))
    job = FLJob(cfg, Communicator(device=__import__("torch").device("cpu")),
                MetricLogger(enabled=False, to_file=False), dataset=ds)
    assert isinstance(job.mod_engine, VmapEngine), type(job.mod_engine)


def test_cnn_hip_engine_tunables(monkeypatch):
    """Host-side sizing contracts of the CNN kernel engine (pure
    functions, no GPU): the wgrad m-split fills >=1024 blocks capped at
    64 (FEDDRIFT_W2MS overrides), and pair chunking respects the
    workspace budget with the 2048-pair kernel cap."""
    from feddrift_amd.ops.cnn_hip import CnnHipEngine
    w2ms = CnnHipEngine._w2ms
    assert w2ms(1) == 64 and w2ms(13) == 64     # small fleets hit the cap
    assert w2ms(100) == 10                       # 1024 // 100
    assert w2ms(2048) == 1                       # scale: no extra split
    monkeypatch.setenv("FEDDRIFT_W2MS", "32")
    assert w2ms(13) == 32                        # A/B override
    monkeypatch.delenv("FEDDRIFT_W2MS")
    eng = CnnHipEngine.__new__(CnnHipEngine)     # no GPU init
    eng.WS_BUDGET = CnnHipEngine.WS_BUDGET
    eng.P = 1_199_882
    assert eng._chunk_pairs(1) >= eng._chunk_pairs(100)  # bigger B, fewer
    assert 1 <= eng._chunk_pairs(500) <= 2048
    eng.WS_BUDGET = 1
    assert eng._chunk_pairs(100) == 1            # floor at one pair
