import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a ROCm GPU (run on MI355X only)")


@pytest.fixture
def sea_dataset_factory():
    """In-memory synthetic SEA dataset builder (no CSV IO)."""
    from feddrift_amd.data.generators import sample_sea
    from feddrift_amd.data.loader import DriftDataset

    def build(n_clients=4, n_iters=3, n=200, seed=0, concept_of=None):
        ds = DriftDataset(data_dir="/nonexistent", dataset="sea",
                          num_client=n_clients)
        rng = np.random.default_rng(seed)
        for c in range(n_clients):
            for t in range(n_iters + 1):
                k = concept_of(c, t) if concept_of else 0
                arr = sample_sea(n, k, rng)
                ds.store.put(c, t, arr[:, :3], arr[:, 3])
        return ds

    return build
