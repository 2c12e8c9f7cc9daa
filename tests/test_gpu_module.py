"""GPU: module path (CNN via MIOpen), big-MLP size dispatch, scale smoke."""

import dataclasses

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs ROCm GPU")


@requires_gpu
def test_cnn_timeline_gpu(tmp_path):
    from feddrift_amd.comm import Communicator
    from feddrift_amd.config import Config
    from feddrift_amd.data.generators import sample_mnist
    from feddrift_amd.data.loader import DriftDataset
    from feddrift_amd.engine.fljob import FLJob
    from feddrift_amd.engine.timeline import clean_state_files
    from feddrift_amd.eval.metrics import MetricLogger

    ds = DriftDataset(data_dir="/nonexistent", dataset="MNIST", num_client=4)
    rng = np.random.default_rng(0)
    for c in range(4):
        for t in range(3):
            arr = sample_mnist(128, 0 if t < 2 else 1, rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])
    cfg = Config(model="cnn", dataset="MNIST", data_dir="/nonexistent",
                 client_num_in_total=4, client_num_per_round=4,
                 batch_size=64, lr=0.005, epochs=3, comm_round=6,
                 total_train_iteration=2, concept_num=2,
                 concept_drift_algo="softcluster",
                 concept_drift_algo_arg="mmacc_06", log_dir=str(tmp_path),
                 report_client=0)
    comm = Communicator()
    clean_state_files(cfg)
    accs = []
    for it in range(2):
        icfg = dataclasses.replace(cfg, curr_train_iteration=it)
        logger = MetricLogger(str(tmp_path), enabled=True, to_file=False)
        job = FLJob(icfg, comm, logger, dataset=ds)
        assert job.device.type == "cuda"
        job.run()
        accs.append(logger.mean("Test/Acc"))
    assert accs[-1] > 0.3, accs


@requires_gpu
def test_big_mlp_size_dispatch_matches_torch():
    """Models beyond the LDS budget must run on the torch-GPU dispatch and
    still produce the fused broadcast+partial semantics."""
    from feddrift_amd.models.packed import spec_for
    from feddrift_amd.ops import mlp_hip, mlp_torch

    spec = spec_for("lr", 784, 62)      # P = 48,670 > TRAIN_MAX_P check
    assert not mlp_hip._fits_train(spec) or spec.n_params < 40000
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    n, G, E, K = 512, 6, 3, 2
    x = torch.rand(n, 784, device=dev)
    y = torch.randint(0, 62, (n,), device=dev)
    glob = torch.randn(K, spec.n_params, device=dev) * 0.05
    rows = torch.arange(G, device=dev)
    model_of = (rows % K).to(torch.int32)
    sw = torch.ones(G, device=dev)
    off = torch.randint(0, n - 65, (G, E), device=dev)
    ln = torch.full((G, E), 64, device=dev)

    reps = torch.zeros(G, spec.n_params, device=dev)
    opt = mlp_torch.make_opt_state("adam", G, spec.n_params, 0.01, 0.001,
                                   dev)
    partial = torch.zeros(K, spec.n_params + 1, device=dev)
    mlp_hip.train_fused(spec, reps, rows, x, y, off, ln, opt,
                        in_params=glob, model_of=model_of, sample_w=sw,
                        partial=partial)
    torch.cuda.synchronize()
    # partial totals = number of pairs per model
    assert torch.allclose(partial[:, -1],
                          torch.tensor([3.0, 3.0], device=dev))
    assert float(partial[:, :-1].abs().max()) > 0


@requires_gpu
def test_scale_smoke_200_clients():
    """FEMNIST-scale direction: 200 clients x K=4 lr ensemble, one full
    round on the engine, everything resident."""
    import time
    from feddrift_amd.comm import Communicator
    from feddrift_amd.config import Config
    from feddrift_amd.data.generators import sample_femnist
    from feddrift_amd.data.loader import DriftDataset
    from feddrift_amd.engine.fljob import FLJob
    from feddrift_amd.eval.metrics import MetricLogger

    C = 200
    ds = DriftDataset(data_dir="/nonexistent", dataset="femnist",
                      num_client=C)
    rng = np.random.default_rng(0)
    for c in range(C):
        for t in range(3):
            arr = sample_femnist(100, 0, rng)
            ds.store.put(c, t, arr[:, :-1], arr[:, -1])
    cfg = Config(model="lr", dataset="femnist", data_dir="/nonexistent",
                 client_num_in_total=C, client_num_per_round=C,
                 batch_size=100, lr=0.01, epochs=5,
                 comm_round=10 ** 9, total_train_iteration=2,
                 curr_train_iteration=1, concept_num=4,
                 concept_drift_algo="softcluster",
                 concept_drift_algo_arg="H_A_F_1_06_0", bench_mode=1,
                 report_client=0)
    comm = Communicator()
    job = FLJob(cfg, comm, MetricLogger(enabled=True, to_file=False),
                dataset=ds)
    client_idx = np.arange(C)
    t0 = time.time()
    for r in range(3):
        plan = job.algo.plan(job, r, client_idx)
        job.train(plan)
        job.algo.aggregate(job, r, plan, client_idx)
        job.algo.test(job, r)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / 3
    acc = job.logger.mean("Test/Acc")
    assert acc > 0.0
    assert dt < 5.0, f"round too slow at 200 clients: {dt:.2f}s"
