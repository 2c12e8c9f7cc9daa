from .generators import generate_data, load_change_points
from .loader import ClientData, DriftDataset, load_all_data, load_retrain_data

__all__ = [
    "generate_data", "load_change_points",
    "ClientData", "DriftDataset", "load_all_data", "load_retrain_data",
]
