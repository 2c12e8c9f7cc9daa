from .softcluster import SoftClusterState
from .driftsurf import DriftSurfState
from .mmacc import MultiModelAccState
from .ada import AdaState
from .kue import KueState

__all__ = ["SoftClusterState", "DriftSurfState", "MultiModelAccState",
           "AdaState", "KueState"]
