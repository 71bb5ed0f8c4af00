from .real import PrefetchLoader, RealClassification, dataset_available
from .synthetic import GlobalBatchSource, GroupBatchSource, SyntheticClassification

__all__ = [
    "SyntheticClassification",
    "GroupBatchSource",
    "GlobalBatchSource",
    "RealClassification",
    "PrefetchLoader",
    "dataset_available",
]
