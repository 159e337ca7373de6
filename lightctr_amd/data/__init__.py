from .libffm import load_libffm, LibffmDataset
from .synthetic import synthetic_criteo_batch, SyntheticCriteo

__all__ = [
    "load_libffm",
    "LibffmDataset",
    "synthetic_criteo_batch",
    "SyntheticCriteo",
]
