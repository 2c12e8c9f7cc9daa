"""Property-based tests (hypothesis) for the string-DSL surfaces and the
flat-packing layer — the places where a malformed-but-plausible input is
most likely to corrupt state silently."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from feddrift_amd.config import (DEFAULT_DELTAS, DRIFTSURF_DELTAS,
                                 driftsurf_delta,
                                 parse_ada_arg, parse_softcluster_arg)
from feddrift_amd.data.loader import RawStore, resolve_retrain_rows
from feddrift_amd.models.generic_packer import ModulePacker
from feddrift_amd.models.packed import PackedMLP, spec_for
from feddrift_amd.models.zoo import FeedForwardNN, LogisticRegression


def _store(n_clients, iters, n, d=3, seed=0):
    s = RawStore(data_dir="/nonexistent", dataset="sea",
                 num_client=n_clients)
    rng = np.random.default_rng(seed)
    for c in range(n_clients):
        for t in range(iters + 1):
            s.put(c, t, rng.random((n, d)).astype(np.float32),
                  rng.integers(0, 2, n))
    return s


@settings(max_examples=30, deadline=None)
@given(curr=st.integers(0, 5), w=st.integers(1, 8), n=st.integers(1, 40))
def test_win_view_row_counts(curr, w, n):
    """win-W must return exactly the last min(W, t+1) iterations' rows."""
    s = _store(2, 6, n)
    rng = np.random.default_rng(0)
    x, y = resolve_retrain_rows(s, 0, curr, f"win-{w}", rng)
    assert len(y) == min(w, curr + 1) * n
    assert x.shape[1] == 3


@settings(max_examples=20, deadline=None)
@given(curr=st.integers(0, 5), n=st.integers(1, 20))
def test_weight_views_duplicate_rows(curr, n):
    """weight-linear duplicates iteration t's rows (t+1)x; weight-exp 2^t
    (the reference's row-duplication semantics, common/retrain.py)."""
    s = _store(1, 6, n)
    rng = np.random.default_rng(0)
    x, _ = resolve_retrain_rows(s, 0, curr, "weight-linear", rng)
    assert len(x) == n * sum(t + 1 for t in range(curr + 1))
    x, _ = resolve_retrain_rows(s, 0, curr, "weight-exp", rng)
    assert len(x) == n * sum(2 ** t for t in range(curr + 1))


@settings(max_examples=20, deadline=None)
@given(curr=st.integers(0, 4), n=st.integers(2, 30),
       seed=st.integers(0, 10))
def test_poisson_view_preserves_size_and_source(curr, n, seed):
    s = _store(1, 5, n)
    rng = np.random.default_rng(seed)
    x, y = resolve_retrain_rows(s, 0, curr, "poisson", rng)
    assert len(y) == n                      # bootstrap keeps the size
    src_x = s.get(0, curr)[0]
    # every drawn row exists in the source iteration
    assert all(any(np.array_equal(r, sr) for sr in src_x) for r in x[:5])


@settings(max_examples=30, deadline=None)
@given(w=st.integers(1, 9), d1=st.integers(0, 99), d2=st.integers(0, 99),
       dist=st.sampled_from("AB"), clust=st.sampled_from("CDF"))
def test_softcluster_h_dsl_roundtrip(w, d1, d2, dist, clust):
    """H_{dist}_{clust}_{W}_{100*delta}_{100*delta'} parses to the same
    numbers it encodes."""
    arg = f"H_{dist}_{clust}_{w}_{d1:02d}_{d2:02d}"
    p = parse_softcluster_arg(arg, "sea")
    assert p.h_w == w
    assert p.h_distance == dist and p.h_cluster == clust
    # 0 fields fall back: delta -> the dataset default, delta' -> delta
    want_d = d1 / 100.0 if d1 else DEFAULT_DELTAS["sea"]
    assert abs(p.h_delta - want_d) < 1e-9
    want_dp = d2 / 100.0 if d2 else p.h_delta
    assert abs(p.h_deltap - want_dp) < 1e-9


@settings(max_examples=20, deadline=None)
@given(alpha=st.integers(0, 20))
def test_softmax_and_mmacc_dsl(alpha):
    # cluster_alg keeps the raw arg (the dispatch matches on substrings,
    # engine/algorithms.py); only the numeric fields are parsed out
    p = parse_softcluster_arg(f"softmax_{alpha}", "sea")
    assert "softmax" in p.cluster_alg and p.softmax_alpha == alpha
    p = parse_softcluster_arg(f"mmacc_{alpha:02d}", "sea")
    assert "mmacc" in p.cluster_alg
    want = alpha / 100.0 if alpha else DEFAULT_DELTAS["sea"]
    assert abs(p.mmacc_delta - want) < 1e-9


def test_driftsurf_delta_defaults_and_override():
    # DriftSurf has its OWN default table (reference :273-278), distinct
    # from the softcluster/mmacc defaults
    for ds, v in DRIFTSURF_DELTAS.items():
        assert driftsurf_delta("", ds) == v
    assert driftsurf_delta("7", "sea") == 0.07


@settings(max_examples=15, deadline=None)
@given(win=st.sampled_from(["all", "win-1", "win-3"]),
       mode=st.sampled_from(["round", "iter"]))
def test_ada_dsl(win, mode):
    w, m = parse_ada_arg(f"{win}_{mode}")
    assert w == win and m == mode


@settings(max_examples=15, deadline=None)
@given(d=st.integers(2, 9), o=st.integers(2, 6), seed=st.integers(0, 99))
def test_packed_mlp_roundtrip(d, o, seed):
    torch.manual_seed(seed)
    spec = spec_for("fnn", d, o)
    m = FeedForwardNN(d, o, 2 * d)
    pk = PackedMLP(spec)
    flat = pk.flatten(m.state_dict())
    assert flat.numel() == spec.n_params
    sd = pk.unflatten(flat)
    for k, v in m.state_dict().items():
        assert torch.allclose(sd[k], v), k


@settings(max_examples=10, deadline=None)
@given(d=st.integers(2, 8), o=st.integers(2, 5), seed=st.integers(0, 99))
def test_module_packer_roundtrip_lr(d, o, seed):
    torch.manual_seed(seed)
    m = LogisticRegression(d, o)
    pk = ModulePacker(m)
    flat = pk.flatten(m.state_dict())
    sd = pk.unflatten(flat)
    for k, v in m.state_dict().items():
        assert torch.allclose(sd[k], v.float()), k
    m2 = LogisticRegression(d, o)
    pk.load_into(m2, flat)
    x = torch.rand(4, d)
    assert torch.allclose(m(x), m2(x), atol=1e-6)
