from .synthetic import DeviceBatchLoader, SyntheticClassification

__all__ = ["SyntheticClassification", "DeviceBatchLoader"]
