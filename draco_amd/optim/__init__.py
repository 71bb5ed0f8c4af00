from .flat_sgd import FlatAdam, FlatSGD

__all__ = ["FlatSGD", "FlatAdam"]
