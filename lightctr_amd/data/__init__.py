from .libffm import load_libffm, LibffmDataset
from .synthetic import synthetic_criteo_batch, SyntheticCriteo
from .dense import load_dense_csv

__all__ = [
    "load_libffm",
    "LibffmDataset",
    "synthetic_criteo_batch",
    "SyntheticCriteo",
    "load_dense_csv",
]
