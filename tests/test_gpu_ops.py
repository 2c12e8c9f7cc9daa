"""GPU numerics: the HIP kernels must match the torch (fp32) reference
implementation (ops/mlp_torch.py) — which itself matches the reference's
eager loops exactly (see test_ops.py)."""

import os

import numpy as np
import pytest
import torch

# Latched once per process by the C++ dispatch: make the templated
# small-eval kernel run even at tiny test grids (its production gate is
# W >= 4096 workgroups), so the fnn(3,6,2) parity case below exercises it.
# The lr(6,3) case is not in the template list and covers the generic
# eval kernel in the same process.
os.environ.setdefault("FEDDRIFT_FORCE_SMALL_EVAL", "1")

from feddrift_amd.models.packed import spec_for
from feddrift_amd.ops import mlp_torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs ROCm GPU")


def _setup(spec_kind, d, o, n=600, G=7, E=5, seed=0):
    torch.manual_seed(seed)
    spec = spec_for(spec_kind, d, o)
    x = torch.rand(n, d) * 10
    y = torch.randint(0, o, (n,))
    params = torch.randn(G + 2, spec.n_params) * 0.3
    rows = torch.randperm(G + 2)[:G]
    # windows of varying lengths, incl. one zero-length (skip) step
    off = torch.randint(0, n - 101, (G, E))
    ln = torch.randint(1, 100, (G, E))
    ln[1, 2] = 0
    return spec, x, y, params, rows, off, ln


@requires_gpu
@pytest.mark.parametrize("kind,d,o", [("fnn", 3, 2), ("fnn", 8, 5),
                                      ("lr", 6, 3)])
@pytest.mark.parametrize("optname", ["adam", "sgd"])
def test_hip_train_matches_torch(kind, d, o, optname):
    from feddrift_amd.ops import mlp_hip
    spec, x, y, params, rows, off, ln = _setup(kind, d, o)
    dev = torch.device("cuda:0")

    p_ref = params.clone()
    opt_ref = mlp_torch.make_opt_state(optname, params.shape[0],
                                       spec.n_params, 0.01, 0.001, "cpu")
    mlp_torch.train_fused(spec, p_ref, rows, x, y, off, ln, opt_ref)

    p_gpu = params.to(dev)
    opt_gpu = mlp_torch.make_opt_state(optname, params.shape[0],
                                       spec.n_params, 0.01, 0.001, dev)
    mlp_hip.train_fused(spec, p_gpu, rows.to(dev), x.to(dev), y.to(dev),
                        off.to(dev), ln.to(dev), opt_gpu)
    torch.cuda.synchronize()

    diff = (p_gpu.cpu() - p_ref).abs().max().item()
    assert diff < 5e-5, diff
    if optname == "adam":
        assert torch.equal(opt_gpu["t"].cpu(), opt_ref["t"])
        md = (opt_gpu["m"].cpu() - opt_ref["m"]).abs().max().item()
        assert md < 5e-5, md


@requires_gpu
def test_hip_train_with_mask():
    from feddrift_amd.ops import mlp_hip
    spec, x, y, params, rows, off, ln = _setup("fnn", 5, 3)
    dev = torch.device("cuda:0")
    mask = (torch.rand(rows.shape[0], 5) > 0.4).float()

    p_ref = params.clone()
    opt_ref = mlp_torch.make_opt_state("adam", params.shape[0],
                                       spec.n_params, 0.01, 0.001, "cpu")
    mlp_torch.train_fused(spec, p_ref, rows, x, y, off, ln, opt_ref,
                          x_mask=mask)

    p_gpu = params.to(dev)
    opt_gpu = mlp_torch.make_opt_state("adam", params.shape[0],
                                       spec.n_params, 0.01, 0.001, dev)
    mlp_hip.train_fused(spec, p_gpu, rows.to(dev), x.to(dev), y.to(dev),
                        off.to(dev), ln.to(dev), opt_gpu, x_mask=mask.to(dev))
    torch.cuda.synchronize()
    assert (p_gpu.cpu() - p_ref).abs().max().item() < 5e-5


@requires_gpu
@pytest.mark.parametrize("kind,d,o", [("fnn", 3, 2), ("lr", 6, 3)])
def test_hip_eval_matches_torch(kind, d, o):
    from feddrift_amd.ops import mlp_hip
    torch.manual_seed(3)
    spec = spec_for(kind, d, o)
    n = 2000
    x = torch.rand(n, d) * 10
    y = torch.randint(0, o, (n,))
    params = torch.randn(4, spec.n_params) * 0.4
    W = 30
    task_row = torch.randint(0, 4, (W,))
    task_id = torch.randint(0, 6, (W,))
    off = torch.randint(0, n - 130, (W,))
    ln = torch.randint(1, 128, (W,))

    c0, t0, l0, m0 = mlp_torch.eval_tasks(spec, params, x, y, task_row,
                                          task_id, off, ln, 6, want_mse=True)
    dev = torch.device("cuda:0")
    c1, t1, l1, m1 = mlp_hip.eval_tasks(
        spec, params.to(dev), x.to(dev), y.to(dev), task_row.to(dev),
        task_id.to(dev), off.to(dev), ln.to(dev), 6, want_mse=True)
    torch.cuda.synchronize()
    assert torch.equal(c1.cpu(), c0)      # correct counts are integral
    assert torch.equal(t1.cpu(), t0)
    assert (l1.cpu() - l0).abs().max().item() < 2e-2
    assert (m1.cpu() - m0).abs().max().item() < 2e-2


@requires_gpu
def test_gpu_full_round_smoke():
    import __graft_entry__
    __graft_entry__.smoke()


@requires_gpu
def test_gpu_timeline_accuracy():
    """Full FedDrift timeline on GPU with HIP kernels: must learn."""
    import os
    import tempfile
    from feddrift_amd.config import Config
    from feddrift_amd.data.generators import generate_data
    from feddrift_amd.engine.timeline import run_timeline

    with tempfile.TemporaryDirectory() as td:
        d = os.path.join(td, "data")
        os.makedirs(os.path.join(d, "changepoints"))
        mat = np.zeros((4, 6), dtype=int)
        mat[2:, :3] = 1
        np.savetxt(os.path.join(d, "changepoints", "T.cp"), mat, fmt="%u")
        np.random.seed(0)
        generate_data("sea", d, 3, 6, 0, 300, 0.0, 1, "T")
        cfg = Config(model="fnn", dataset="sea", data_dir=d,
                     client_num_in_total=6, client_num_per_round=6,
                     batch_size=300, lr=0.01, epochs=5, comm_round=15,
                     total_train_iteration=3, concept_num=2,
                     concept_drift_algo="softcluster",
                     concept_drift_algo_arg="H_A_C_1_10_0",
                     change_points="T", dummy_arg=0, log_dir=td,
                     report_client=0, use_hip_kernels="always")
        out = run_timeline(cfg)
        assert out["per_iteration_test_acc"][-1] > 0.8, out


@requires_gpu
def test_hip_fused_broadcast_and_partial():
    """The in_params/partial fusion (stage from global rows + weighted
    aggregation sums in the train launch) must equal the unfused sequence:
    replica sync -> train -> einsum."""
    from feddrift_amd.ops import mlp_hip
    torch.manual_seed(2)
    spec = spec_for("fnn", 3, 2)
    dev = torch.device("cuda:0")
    K, nW, E, n = 3, 4, 4, 500
    P = spec.n_params
    x = (torch.rand(n, 3) * 10).to(dev)
    y = torch.randint(0, 2, (n,)).to(dev)
    glob = (torch.randn(K, P) * 0.3).to(dev)
    G = nW * K
    rows = torch.arange(G, device=dev)
    model_of = (rows % K).to(torch.int32)
    sample_w = torch.rand(G, device=dev) + 0.5
    off = torch.randint(0, n - 101, (G, E)).to(dev)
    ln = torch.randint(1, 100, (G, E)).to(dev)

    # reference: sync replicas then train then einsum
    reps_ref = glob[rows % K].clone()
    opt_ref = mlp_torch.make_opt_state("adam", G, P, 0.01, 0.001, dev)
    mlp_torch.train_fused(spec, reps_ref, rows, x, y, off, ln, opt_ref)
    part_ref = torch.zeros(K, P + 1, device=dev)
    for g in range(G):
        part_ref[g % K, :P] += sample_w[g] * reps_ref[g]
        part_ref[g % K, P] += sample_w[g]

    reps = torch.zeros(G, P, device=dev)
    opt = mlp_torch.make_opt_state("adam", G, P, 0.01, 0.001, dev)
    partial = torch.zeros(K, P + 1, device=dev)
    mlp_hip.train_fused(spec, reps, rows, x, y, off, ln, opt,
                        in_params=glob, model_of=model_of,
                        sample_w=sample_w, partial=partial)
    torch.cuda.synchronize()
    assert (reps - reps_ref).abs().max().item() < 5e-5
    assert (partial - part_ref).abs().max().item() < 3e-4

    # apply: average models, verify against manual; the kernel DRAINS
    # partial to zero on its way out (next round's fused accumulation
    # starts clean without a separate fill) and parks totals
    glob2 = glob.clone()
    part2 = partial.clone()
    totals = mlp_hip.apply_aggregate(glob2, part2, None)
    torch.cuda.synchronize()
    want = part_ref[:, :P] / part_ref[:, P:P + 1]
    assert (glob2 - want).abs().max().item() < 3e-4
    assert (totals - part_ref[:, P]).abs().max().item() < 1e-4
    assert part2.abs().max().item() == 0.0, "partial must be drained"

    # masked apply keeps masked-out rows (fresh partial: the buffer is
    # consumed by each apply)
    glob3 = glob.clone()
    part3 = partial.clone()
    mask = torch.tensor([1, 0, 1], dtype=torch.uint8, device=dev)
    mlp_hip.apply_aggregate(glob3, part3, mask)
    torch.cuda.synchronize()
    assert torch.equal(glob3[1], glob[1])
    assert (glob3[0] - want[0]).abs().max().item() < 3e-4
    assert part3.abs().max().item() == 0.0


@requires_gpu
@pytest.mark.parametrize("mode", ["hard", "soft"])
@pytest.mark.parametrize("per_task", [False, True])
def test_hip_vote_matches_torch(mode, per_task):
    from feddrift_amd.ops import mlp_hip
    torch.manual_seed(5)
    spec = spec_for("fnn", 4, 3)
    n, M, T, W = 1500, 4, 6, 40
    x = torch.rand(n, 4) * 4
    y = torch.randint(0, 3, (n,))
    params = torch.randn(M, spec.n_params) * 0.5
    task_id = torch.randint(0, T, (W,))
    off = torch.randint(0, n - 130, (W,))
    ln = torch.randint(1, 128, (W,))
    weights = (torch.rand(T, M) if per_task else torch.rand(M))
    weights[..., 1] = 0.0   # inactive model skipped
    ref = mlp_torch.ens_vote_multi(spec, params, weights, x, y, task_id,
                                   off, ln, T, mode=mode)
    dev = torch.device("cuda:0")
    got = mlp_hip.ens_vote_multi(
        spec, params.to(dev), weights.to(dev), x.to(dev), y.to(dev),
        task_id.to(dev), off.to(dev), ln.to(dev), T, mode=mode)
    torch.cuda.synchronize()
    assert torch.equal(got[1].cpu(), ref[1])
    # vote ties can break differently at fp ulp; require near-exact
    assert (got[0].cpu() - ref[0]).abs().max() <= 1.0


@requires_gpu
def test_hip_vote_with_masks():
    from feddrift_amd.ops import mlp_hip
    torch.manual_seed(6)
    spec = spec_for("lr", 5, 3)
    n, M, T, W = 800, 3, 4, 20
    x = torch.rand(n, 5) * 3
    y = torch.randint(0, 3, (n,))
    params = torch.randn(M, spec.n_params) * 0.5
    task_id = torch.randint(0, T, (W,))
    off = torch.randint(0, n - 80, (W,))
    ln = torch.randint(1, 64, (W,))
    weights = torch.rand(M)
    masks = (torch.rand(M, 5) > 0.3).float()
    ref = mlp_torch.ens_vote_multi(spec, params, weights, x, y, task_id,
                                   off, ln, T, mode="soft", masks=masks)
    dev = torch.device("cuda:0")
    got = mlp_hip.ens_vote_multi(
        spec, params.to(dev), weights.to(dev), x.to(dev), y.to(dev),
        task_id.to(dev), off.to(dev), ln.to(dev), T, mode="soft",
        masks=masks.to(dev))
    torch.cuda.synchronize()
    assert torch.equal(got[1].cpu(), ref[1])
    assert (got[0].cpu() - ref[0]).abs().max() <= 1.0


@requires_gpu
def test_hip_confusion_matches_torch():
    from feddrift_amd.ops import mlp_hip
    torch.manual_seed(7)
    spec = spec_for("fnn", 3, 2)
    n, M, T, W = 1200, 3, 5, 30
    x = torch.rand(n, 3) * 5
    y = torch.randint(0, 2, (n,))
    params = torch.randn(M, spec.n_params) * 0.5
    task_row = torch.randint(0, M, (W,))
    task_id = torch.randint(0, T, (W,))
    off = torch.randint(0, n - 130, (W,))
    ln = torch.randint(1, 128, (W,))
    ref = mlp_torch.confusion_tasks(spec, params, x, y, task_row, task_id,
                                    off, ln, T, 2)
    dev = torch.device("cuda:0")
    got = mlp_hip.confusion_tasks(
        spec, params.to(dev), x.to(dev), y.to(dev), task_row.to(dev),
        task_id.to(dev), off.to(dev), ln.to(dev), T, 2)
    torch.cuda.synchronize()
    assert torch.equal(got.cpu(), ref)
