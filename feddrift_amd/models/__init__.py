from .zoo import (CNN_DropOut, FeedForwardNN, LogisticRegression, create_model,
                  reinitialize, set_torch_seed)
from .packed import MLPSpec, PackedMLP, spec_for

__all__ = ["CNN_DropOut", "FeedForwardNN", "LogisticRegression",
           "create_model", "reinitialize", "set_torch_seed",
           "MLPSpec", "PackedMLP", "spec_for"]
