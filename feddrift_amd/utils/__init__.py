from .optrepo import OptRepo

__all__ = ["OptRepo"]
