"""Run configuration: argparse surface + the string micro-DSL parsers.

Compatible with the reference CLI surface:
  * the argparse flags of fedml_experiments/distributed/fedavg_cont_ens/main_fedavg.py:42-139
  * the DRIFT_ALGO / DRIFT_ALGO_ARG mini-DSLs parsed in
    fedml_api/distributed/fedavg_ens/FedAvgEnsDataLoader.py:1272-1341 (softcluster),
    :269-314 (driftsurf), :128-143 (ada)
  * the retrain_data DSL of fedml_api/data_preprocessing/common/retrain.py:7-91
"""

from __future__ import annotations

import argparse
from dataclasses import dataclass, field, fields


# Per-dataset default drift-detection deltas
# (reference FedAvgEnsDataLoader.py:1274 and :455).
DEFAULT_DELTAS = {"sea": 0.04, "sine": 0.20, "circle": 0.10, "MNIST": 0.10}
# DriftSurf uses its own table (reference FedAvgEnsDataLoader.py:274).
DRIFTSURF_DELTAS = {"sea": 0.02, "sine": 0.10, "circle": 0.05}


@dataclass
class Config:
    model: str = "fnn"
    dataset: str = "sea"
    data_dir: str = "./data"
    client_num_in_total: int = 10
    client_num_per_round: int = 10
    batch_size: int = 500
    client_optimizer: str = "adam"       # reference default: adam (amsgrad, wd)
    lr: float = 0.01
    wd: float = 0.001
    epochs: int = 5                      # local steps per round
    comm_round: int = 200
    frequency_of_the_test: int = 1
    ci: int = 0
    total_train_iteration: int = 10
    curr_train_iteration: int = 0
    drift_together: int = 0
    report_client: int = 1
    retrain_data: str = "win-1"
    concept_drift_algo: str = "softcluster"
    concept_drift_algo_arg: str = ""
    ensemble_window: int = 4
    concept_num: int = 4
    change_points: str = "rand"
    time_stretch: int = 1
    reset_models: int = 0
    # data-GENERATION surface args (consumed by scripts/prepare_data.py /
    # generate_data at data-prep time, like the reference's main_fedavg
    # arg list; the engine itself reads the generated CSVs)
    noise_prob: float = 0.0
    dummy_arg: int = 0
    sample_num: int = 500                # samples per (client, iteration)

    # engine knobs (new; no reference equivalent)
    device: str = "auto"                 # auto|cpu|cuda
    backend: str = "auto"                # auto|gloo|nccl
    use_hip_kernels: str = "auto"        # auto|always|never
    wandb: int = 0                       # optional wandb mirror of metric logs
    log_dir: str = "."                   # where checkpoints/state files go
    bench_mode: int = 0                  # 1: synthetic steady-state (bench.py)

    # robust aggregation (fedavg_robust / fedml_core robustness equivalent)
    secure_agg: int = 0                  # 1: pairwise-mask uploads
                                         # (turboaggregate equivalent)
    robust_norm_bound: float = 0.0       # >0: clip client updates to this L2
    robust_noise: float = 0.0            # >0: Gaussian noise stddev on avg
    # FedOpt server optimizer ('avg' = plain FedAvg replacement)
    server_optimizer: str = "avg"
    server_lr: float = 1.0

    def __post_init__(self):
        self.dataset_norm = "MNIST" if self.dataset.lower() == "mnist" else self.dataset

    @property
    def is_softcluster(self) -> bool:
        return self.concept_drift_algo in (
            "softcluster", "softclusterwin-1", "softclusterreset")


def add_args(parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
    for f in fields(Config):
        if f.name in ("dataset_norm",):
            continue
        parser.add_argument("--" + f.name, type=type(f.default), default=f.default)
    return parser


def config_from_argv(argv=None) -> Config:
    parser = argparse.ArgumentParser()
    add_args(parser)
    ns, _ = parser.parse_known_args(argv)
    return Config(**{f.name: getattr(ns, f.name) for f in fields(Config)
                     if hasattr(ns, f.name)})


# ---------------------------------------------------------------------------
# DRIFT_ALGO_ARG parsers
# ---------------------------------------------------------------------------

@dataclass
class SoftClusterParams:
    """Parsed concept_drift_algo_arg for the softcluster family.

    Mirrors the parse in reference FedAvgEnsDataLoader.py:1279-1328:
      'hard' | 'hard-r' (IFCA)        -> cluster_alg as-is
      'softmax_{alpha}'               -> softmax_alpha = int
      'mmacc_{100*delta}'             -> mmacc_delta (FedDrift-Eager)
      'gmm', 'geni'                   -> as-is
      'H_{dist}_{clust}_{W}_{100d}_{100dp}' -> FedDrift hierarchical
      'cfl_{gamma}_{win-1|all}'       -> CFL inside softcluster
    """
    cluster_alg: str = ""
    mmacc_delta: float = 0.0
    softmax_alpha: int = 0
    h_delta: float = 0.0
    h_deltap: float = 0.0
    h_w: int = 0
    h_distance: str = ""
    h_cluster: str = ""
    cfl_gamma: float = 0.0
    cfl_retrain: str = ""


def parse_softcluster_arg(arg: str, dataset: str) -> SoftClusterParams:
    p = SoftClusterParams(cluster_alg=arg)
    if "mmacc" in arg:
        p.mmacc_delta = 0.01 * float(arg.split("_")[-1])
        if p.mmacc_delta == 0 and dataset in DEFAULT_DELTAS:
            p.mmacc_delta = DEFAULT_DELTAS[dataset]
    elif "softmax" in arg:
        p.softmax_alpha = int(arg.split("_")[-1])
    elif arg == "geni":
        pass
    elif "H" in arg:
        parts = arg.split("_")
        p.h_distance = parts[1]
        p.h_cluster = parts[2]
        p.h_w = int(parts[3])
        p.h_delta = 0.01 * float(parts[4])
        if p.h_delta == 0 and dataset in DEFAULT_DELTAS:
            p.h_delta = DEFAULT_DELTAS[dataset]
        p.h_deltap = 0.01 * float(parts[5])
        if p.h_deltap == 0:
            p.h_deltap = p.h_delta
    elif "cfl" in arg:
        parts = arg.split("_")
        p.cfl_gamma = float(parts[1])
        p.cfl_retrain = parts[2]
    return p


def driftsurf_delta(arg: str, dataset: str) -> float:
    """DriftSurf delta from its algo arg (reference FedAvgEnsDataLoader.py:273-278)."""
    d = 0.01 * float(arg) if arg else 0.0
    if d == 0:
        d = DRIFTSURF_DELTAS.get(dataset, 0.1)
    return d


def parse_ada_arg(arg: str):
    """'{win-1|all}_{round|iter}' (reference FedAvgEnsDataLoader.py:137-138,
    FedAvgEnsAggregatorAda.py)."""
    parts = arg.split("_")
    retrain = parts[0] if parts[0] else "win-1"
    granularity = parts[1] if len(parts) > 1 else "round"
    return retrain, granularity
