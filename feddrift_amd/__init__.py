"""feddrift-mi355x: MI355X-native federated learning under distributed concept drift.

A from-scratch engine with the capabilities of microsoft/FedDrift (AISTATS'23).
Architecture (nothing shared with the reference's MPI/FedML design):

  * one process per GPU; simulated clients are sharded across ranks
    (reference: one MPI process per client, fedml_core/distributed/...)
  * all K ensemble models, per-(client,model) replicas, optimizer state and
    client data are resident in HBM for the whole run (reference shuttles each
    model CPU<->GPU every round: fedml_api/distributed/fedavg_ens/FedAvgEnsTrainer.py:51,87)
  * aggregation is a fused weighted parameter sum + RCCL all_reduce over xGMI
    (reference: pickled state_dicts over mpi4py p2p + a Python triple loop,
    FedAvgEnsAggregatorSoftCluster.py:148-195)
  * the hot numerical loops (local train steps, model x client accuracy
    matrices, weighted averaging) are hand-written CDNA4 HIP kernels in
    feddrift_amd/ops/hip/, with a vectorized torch CPU path used as the
    numerics reference and for GPU-less testing.

Drift-algorithm surface (DRIFT_ALGO / DRIFT_ALGO_ARG), the 24-argument shell
entrypoint, and the per-iteration checkpoint layout stay compatible with the
reference.
"""

__version__ = "0.1.0"
