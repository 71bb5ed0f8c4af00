from .synthetic import GlobalBatchSource, GroupBatchSource, SyntheticClassification

__all__ = ["SyntheticClassification", "GroupBatchSource", "GlobalBatchSource"]
