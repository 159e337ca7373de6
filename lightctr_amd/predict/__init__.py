from .ann_index import ANNIndex
from .predictor import BatchPredictor

__all__ = ["ANNIndex", "BatchPredictor"]
